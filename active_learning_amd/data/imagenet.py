"""ImageNet folder dataset (ImageFolder-style) + ImageNet-LT list dataset.

Reference: src/data_utils/custom_imagenet.py (ImageFolder subclass,
index-returning __getitem__ :24-26, debug cap :15-18) and
src/data_utils/custom_imbalanced_imagenet.py (path/label txt lists :22-26).
"""

import os

import torch

from .transforms import imagenet_transforms

_IMG_EXTS = {".jpg", ".jpeg", ".png", ".bmp", ".webp"}


def _load_image(path):
    from PIL import Image
    with Image.open(path) as img:
        return img.convert("RGB")


class CustomImageNet(torch.utils.data.Dataset):
    num_classes = 1000

    def __init__(self, root, transform=None, debug_mode=False):
        if not os.path.isdir(root):
            raise FileNotFoundError(f"ImageNet directory {root!r} does not exist. Use "
                                    f"--dataset synthetic_imagenet for synthetic data.")
        classes = sorted(d.name for d in os.scandir(root) if d.is_dir())
        self.class_to_idx = {c: i for i, c in enumerate(classes)}
        self.samples = []
        for c in classes:
            cdir = os.path.join(root, c)
            for fn in sorted(os.listdir(cdir)):
                if os.path.splitext(fn)[1].lower() in _IMG_EXTS:
                    self.samples.append((os.path.join(cdir, fn), self.class_to_idx[c]))
        self.targets = [y for _, y in self.samples]
        self.transform = transform
        self.debug_mode = debug_mode

    def __len__(self):
        if self.debug_mode:
            return 50
        return len(self.samples)

    def __getitem__(self, index):
        path, y = self.samples[index]
        x = _load_image(path)
        if self.transform is not None:
            x = self.transform(x)
        return x, y, index


class ImbalanceImagenet(torch.utils.data.Dataset):
    """ImageNet-LT: dataset defined by a txt file of `path label` lines.

    Reference: src/data_utils/custom_imbalanced_imagenet.py:17-40.
    """

    num_classes = 1000

    def __init__(self, root, list_file, transform=None, debug_mode=False):
        self.img_path = []
        self.targets = []
        with open(list_file) as fh:
            for line in fh:
                parts = line.split()
                if len(parts) >= 2:
                    self.img_path.append(os.path.join(root, parts[0]))
                    self.targets.append(int(parts[1]))
        self.transform = transform
        self.debug_mode = debug_mode

    def __len__(self):
        if self.debug_mode:
            return 50
        return len(self.targets)

    def __getitem__(self, index):
        x = _load_image(self.img_path[index])
        y = self.targets[index]
        if self.transform is not None:
            x = self.transform(x)
        return x, y, index


def get_data_imagenet(data_path, debug_mode=False):
    train_transform, val_transform = imagenet_transforms()
    traindir = os.path.join(data_path, "train")
    valdir = os.path.join(data_path, "val")
    train_set = CustomImageNet(traindir, transform=train_transform, debug_mode=debug_mode)
    test_set = CustomImageNet(valdir, transform=val_transform, debug_mode=debug_mode)
    al_set = CustomImageNet(traindir, transform=val_transform, debug_mode=debug_mode)
    return train_set, test_set, al_set


def get_data_imbalanced_imagenet(data_path, debug_mode=False):
    """ImageNet-LT splits: expects ImageNet_LT_{train,test}.txt under data_path."""
    train_transform, val_transform = imagenet_transforms()
    train_list = os.path.join(data_path, "ImageNet_LT_train.txt")
    test_list = os.path.join(data_path, "ImageNet_LT_test.txt")
    train_set = ImbalanceImagenet(data_path, train_list, transform=train_transform,
                                  debug_mode=debug_mode)
    test_set = ImbalanceImagenet(data_path, test_list, transform=val_transform,
                                 debug_mode=debug_mode)
    al_set = ImbalanceImagenet(data_path, train_list, transform=val_transform,
                               debug_mode=debug_mode)
    return train_set, test_set, al_set
