"""Tensor-native image transforms (torchvision is not a dependency).

Implements exactly the transform set the reference uses:
  CIFAR train:  RandomCrop(32, pad 4) + RandomHorizontalFlip + Normalize
                (src/data_utils/custom_cifar10.py:47-50)
  CIFAR eval:   Normalize only (custom_cifar10.py:51-53)
  ImageNet train: RandomResizedCrop(224) + RandomHorizontalFlip + Normalize
                (src/data_utils/custom_imagenet.py:49-51)
  ImageNet eval:  Resize(256) + CenterCrop(224) + Normalize (custom_imagenet.py:52-53)

All transforms take and return float32 CHW tensors in [0, 1] (pre-normalize).
``to_chw_tensor`` converts PIL images / HWC uint8 arrays.
"""

import math
import random

import numpy as np
import torch
import torch.nn.functional as F

CIFAR10_MEAN = (0.4914, 0.4822, 0.4465)
CIFAR10_STD = (0.2023, 0.1994, 0.2010)
IMAGENET_MEAN = (0.485, 0.456, 0.406)
IMAGENET_STD = (0.229, 0.224, 0.225)


def to_chw_tensor(img) -> torch.Tensor:
    """PIL image / HWC uint8 ndarray / CHW tensor -> float32 CHW in [0,1]."""
    if isinstance(img, torch.Tensor):
        t = img
        if t.dtype == torch.uint8:
            t = t.float().div_(255.0)
        return t
    arr = np.asarray(img)
    if arr.ndim == 2:
        arr = arr[:, :, None]
    t = torch.from_numpy(np.ascontiguousarray(arr))
    if t.dtype == torch.uint8:
        t = t.float().div_(255.0)
    return t.permute(2, 0, 1).contiguous()


class Compose:
    def __init__(self, transforms):
        self.transforms = list(transforms)

    def __call__(self, x):
        for t in self.transforms:
            x = t(x)
        return x


class Normalize:
    def __init__(self, mean, std):
        self.mean = torch.tensor(mean).view(-1, 1, 1)
        self.std = torch.tensor(std).view(-1, 1, 1)

    def __call__(self, x):
        return (x - self.mean) / self.std


class RandomHorizontalFlip:
    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, x):
        if random.random() < self.p:
            return torch.flip(x, dims=[2])
        return x


class RandomCrop:
    def __init__(self, size, padding=0):
        self.size = size
        self.padding = padding

    def __call__(self, x):
        if self.padding:
            x = F.pad(x, (self.padding,) * 4)
        _, h, w = x.shape
        top = random.randint(0, h - self.size)
        left = random.randint(0, w - self.size)
        return x[:, top:top + self.size, left:left + self.size]


class CenterCrop:
    def __init__(self, size):
        self.size = size

    def __call__(self, x):
        _, h, w = x.shape
        top = max(0, (h - self.size) // 2)
        left = max(0, (w - self.size) // 2)
        return x[:, top:top + self.size, left:left + self.size]


class Resize:
    """Resize the shorter side to ``size`` (bilinear, antialiased)."""

    def __init__(self, size):
        self.size = size

    def __call__(self, x):
        _, h, w = x.shape
        if h <= w:
            nh, nw = self.size, max(1, round(w * self.size / h))
        else:
            nh, nw = max(1, round(h * self.size / w)), self.size
        return F.interpolate(x[None], size=(nh, nw), mode="bilinear",
                             align_corners=False, antialias=True)[0]


class RandomResizedCrop:
    """Scale (0.08, 1.0), ratio (3/4, 4/3) crop resized to ``size`` — the
    standard ImageNet augmentation the reference gets from torchvision."""

    def __init__(self, size, scale=(0.08, 1.0), ratio=(3 / 4, 4 / 3)):
        self.size = size
        self.scale = scale
        self.ratio = ratio

    def __call__(self, x):
        _, h, w = x.shape
        area = h * w
        for _ in range(10):
            target_area = area * random.uniform(*self.scale)
            log_ratio = (math.log(self.ratio[0]), math.log(self.ratio[1]))
            aspect = math.exp(random.uniform(*log_ratio))
            cw = int(round(math.sqrt(target_area * aspect)))
            ch = int(round(math.sqrt(target_area / aspect)))
            if 0 < cw <= w and 0 < ch <= h:
                top = random.randint(0, h - ch)
                left = random.randint(0, w - cw)
                crop = x[:, top:top + ch, left:left + cw]
                return F.interpolate(crop[None], size=(self.size, self.size),
                                     mode="bilinear", align_corners=False)[0]
        # fallback: center crop of the shorter side
        side = min(h, w)
        crop = CenterCrop(side)(x)
        return F.interpolate(crop[None], size=(self.size, self.size), mode="bilinear",
                             align_corners=False)[0]


def cifar_transforms():
    train = Compose([to_chw_tensor, RandomCrop(32, padding=4), RandomHorizontalFlip(),
                     Normalize(CIFAR10_MEAN, CIFAR10_STD)])
    evalt = Compose([to_chw_tensor, Normalize(CIFAR10_MEAN, CIFAR10_STD)])
    return train, evalt


def imagenet_transforms():
    train = Compose([to_chw_tensor, RandomResizedCrop(224), RandomHorizontalFlip(),
                     Normalize(IMAGENET_MEAN, IMAGENET_STD)])
    evalt = Compose([to_chw_tensor, Resize(256), CenterCrop(224),
                     Normalize(IMAGENET_MEAN, IMAGENET_STD)])
    return train, evalt
