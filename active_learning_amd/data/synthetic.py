"""Deterministic synthetic image datasets.

Used where the reference would download CIFAR/ImageNet (this environment has
no network) and by bench.py (BASELINE.json requires synthetic ImageNet-shaped
data). Images are generated per-index from a seeded generator with a
class-dependent mean so models can actually fit the data in tests.
"""

import numpy as np
import torch


class SyntheticImageDataset(torch.utils.data.Dataset):
    """Index-returning synthetic dataset: __getitem__ -> (x, y, index),
    matching the contract of every reference dataset
    (src/data_utils/custom_cifar10.py:23-25)."""

    def __init__(self, size, num_classes, img_shape=(3, 32, 32), seed=0, transform=None,
                 debug_mode=False):
        self.size = size
        self.num_classes = num_classes
        self.img_shape = tuple(img_shape)
        self.seed = seed
        self.transform = transform
        self.debug_mode = debug_mode
        rng = np.random.default_rng(seed)
        self.targets = rng.integers(0, num_classes, size=size).tolist()
        # low-dim class signature added to noise so the data is learnable
        g = torch.Generator().manual_seed(seed + 1)
        self._class_means = torch.randn(num_classes, self.img_shape[0], 4, 4, generator=g) * 0.5

    def __len__(self):
        if self.debug_mode:
            return min(50, self.size)
        return self.size

    def __getitem__(self, index):
        index = int(index)
        y = self.targets[index]
        g = torch.Generator().manual_seed(self.seed * 1_000_003 + index)
        x = torch.randn(*self.img_shape, generator=g) * 0.25
        mean = torch.nn.functional.interpolate(
            self._class_means[y][None], size=self.img_shape[1:], mode="nearest")[0]
        x = x + mean
        if self.transform is not None:
            x = self.transform(x)
        return x, y, index


def get_data_synthetic(num_classes, train_size, test_size, img_shape, seed=0,
                       debug_mode=False, train_transform=None, eval_transform=None):
    """Return (train_set, test_set, al_set) with the reference triple contract:
    al_set = train data under eval transforms (src/data_utils/custom_cifar10.py:36-38)."""
    train_set = SyntheticImageDataset(train_size, num_classes, img_shape, seed=seed,
                                      transform=train_transform, debug_mode=debug_mode)
    test_set = SyntheticImageDataset(test_size, num_classes, img_shape, seed=seed + 7,
                                     transform=eval_transform, debug_mode=debug_mode)
    al_set = SyntheticImageDataset(train_size, num_classes, img_shape, seed=seed,
                                   transform=eval_transform, debug_mode=debug_mode)
    return train_set, test_set, al_set
