"""CIFAR-10 dataset reading the standard `cifar-10-batches-py` pickle format.

Reference: src/data_utils/custom_cifar10.py (torchvision CIFAR10 subclass with
index-returning __getitem__ :23-25 and debug-mode length cap :14-17). This
implementation reads the same on-disk format torchvision uses, without
torchvision, and cannot download (no network in this environment).
"""

import os
import pickle

import numpy as np
import torch

from .transforms import cifar_transforms

_TRAIN_FILES = [f"data_batch_{i}" for i in range(1, 6)]
_TEST_FILES = ["test_batch"]


def _load_batches(root, files):
    base = os.path.join(root, "cifar-10-batches-py")
    data, labels = [], []
    for fn in files:
        path = os.path.join(base, fn)
        with open(path, "rb") as fh:
            entry = pickle.load(fh, encoding="latin1")
        data.append(entry["data"])
        labels.extend(entry.get("labels", entry.get("fine_labels", [])))
    data = np.vstack(data).reshape(-1, 3, 32, 32).transpose(0, 2, 3, 1)  # HWC uint8
    return data, labels


class CustomCIFAR10(torch.utils.data.Dataset):
    num_classes = 10

    def __init__(self, root, train=True, transform=None, debug_mode=False, **_):
        if root is None or not os.path.isdir(os.path.join(str(root), "cifar-10-batches-py")):
            raise FileNotFoundError(
                f"CIFAR-10 batches not found under {root!r}. This environment cannot "
                f"download datasets; use --dataset synthetic_cifar10 for synthetic data.")
        self.data, self.targets = _load_batches(root, _TRAIN_FILES if train else _TEST_FILES)
        self.transform = transform
        self.debug_mode = debug_mode

    def __len__(self):
        if self.debug_mode:
            return 50  # custom_cifar10.py:14-17
        return len(self.data)

    def __getitem__(self, index):
        x, y = self.data[index], self.targets[index]
        if self.transform is not None:
            x = self.transform(x)
        return x, y, index


def get_data_cifar10(data_path, input_size=(32, 32), supervised=False, debug_mode=False):
    train_transform, val_transform = cifar_transforms()
    train_set = CustomCIFAR10(data_path, train=True, transform=train_transform,
                              debug_mode=debug_mode)
    test_set = CustomCIFAR10(data_path, train=False, transform=val_transform,
                             debug_mode=debug_mode)
    al_set = CustomCIFAR10(data_path, train=True, transform=val_transform,
                           debug_mode=debug_mode)
    return train_set, test_set, al_set
