"""Data facade — dispatch by dataset name.

Reference: src/data_utils/top_level_data_utils.py:7-19. Adds two synthetic
datasets (CIFAR-/ImageNet-shaped) because this environment has no network for
downloads; BASELINE.json's bench configs use synthetic data explicitly.

Every builder returns (train_set, test_set, al_set) where al_set is the train
data under eval (non-augmenting) transforms, and every dataset's __getitem__
returns (x, y, index).
"""

import os

from .cifar10 import get_data_cifar10
from .imagenet import get_data_imagenet, get_data_imbalanced_imagenet
from .imbalanced_cifar10 import get_data_imbalanced_cifar10
from .synthetic import SyntheticImageDataset, get_data_synthetic  # noqa: F401


def get_data(data_path, data_name, supervised=False, debug_mode=False, imbalance_args=None):
    if data_name == "cifar10":
        return get_data_cifar10(data_path, input_size=(32, 32), supervised=supervised,
                                debug_mode=debug_mode)
    if data_name == "imagenet":
        return get_data_imagenet(data_path, debug_mode=debug_mode)
    if data_name == "imbalanced_cifar10":
        return get_data_imbalanced_cifar10(data_path, debug_mode=debug_mode,
                                           imbalance_args=imbalance_args)
    if data_name == "imbalanced_imagenet":
        return get_data_imbalanced_imagenet(data_path, debug_mode=debug_mode)
    if data_name == "synthetic_cifar10":
        train_size = int(os.environ.get("AL_AMD_SYNTH_TRAIN_SIZE", 50_000))
        test_size = int(os.environ.get("AL_AMD_SYNTH_TEST_SIZE", 10_000))
        return get_data_synthetic(10, train_size, test_size, (3, 32, 32),
                                  debug_mode=debug_mode)
    if data_name == "synthetic_imagenet":
        train_size = int(os.environ.get("AL_AMD_SYNTH_TRAIN_SIZE", 1_281_167))
        test_size = int(os.environ.get("AL_AMD_SYNTH_TEST_SIZE", 50_000))
        return get_data_synthetic(1000, train_size, test_size, (3, 224, 224),
                                  debug_mode=debug_mode)
    raise ValueError(f"Dataset {data_name!r} does not exist")
