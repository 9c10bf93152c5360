"""Long-tailed CIFAR-10 by per-class subsampling.

Reference: src/data_utils/custom_imbalanced_cifar10.py — exp decay
``img_max * factor^(c/(C-1))`` (:33-35), step profile (:36-40), seeded
per-class shuffle + truncate (:45-61).
"""

import numpy as np

from .cifar10 import CustomCIFAR10
from .transforms import cifar_transforms


def get_img_num_per_cls(total, num_classes, imbalance_type, imbalance_factor):
    img_max = total / num_classes
    if imbalance_type == "exp":
        return [int(img_max * imbalance_factor ** (c / (num_classes - 1.0)))
                for c in range(num_classes)]
    if imbalance_type == "step":
        half = num_classes // 2
        return [int(img_max)] * half + [int(img_max * imbalance_factor)] * half
    raise ValueError("Choose a valid imbalance_type: one of exp or step.")


class ImbalanceCifar10(CustomCIFAR10):
    def __init__(self, root, train=True, transform=None, debug_mode=False,
                 imbalance_args=None, **kw):
        super().__init__(root, train=train, transform=transform, debug_mode=debug_mode, **kw)
        imbalance_args = imbalance_args or {}
        self.imbalance_type = imbalance_args.get("imbalance_type")
        self.imbalance_factor = imbalance_args.get("imbalance_factor", 0.1)
        self.imbalance_seed = imbalance_args.get("imbalance_seed", 0)
        if self.imbalance_type in ("exp", "step"):
            self.img_num_list = get_img_num_per_cls(len(self.data), self.num_classes,
                                                    self.imbalance_type, self.imbalance_factor)
            self._gen_imbalanced_data(self.img_num_list)

    def _gen_imbalanced_data(self, img_num_per_cls):
        rng = np.random.RandomState(self.imbalance_seed)
        targets_np = np.asarray(self.targets, dtype=np.int64)
        new_data, new_targets = [], []
        self.num_per_cls_dict = {}
        for cls, n in zip(np.unique(targets_np), img_num_per_cls):
            self.num_per_cls_dict[int(cls)] = n
            idx = np.where(targets_np == cls)[0]
            rng.shuffle(idx)
            sel = idx[:n]
            new_data.append(self.data[sel, ...])
            new_targets.extend([int(cls)] * n)
        self.data = np.vstack(new_data)
        self.targets = new_targets

    def get_num_classes_list(self):
        if self.imbalance_type is None:
            return [len(self.data) // self.num_classes] * self.num_classes
        return [self.num_per_cls_dict[i] for i in range(self.num_classes)]


def get_data_imbalanced_cifar10(data_path, input_size=(32, 32), supervised=False,
                                debug_mode=False, imbalance_args=None):
    train_transform, val_transform = cifar_transforms()
    train_set = ImbalanceCifar10(data_path, train=True, transform=train_transform,
                                 debug_mode=debug_mode, imbalance_args=imbalance_args)
    test_set = CustomCIFAR10(data_path, train=False, transform=val_transform,
                             debug_mode=debug_mode)
    al_set = ImbalanceCifar10(data_path, train=True, transform=val_transform,
                              debug_mode=debug_mode, imbalance_args=imbalance_args)
    return train_set, test_set, al_set
