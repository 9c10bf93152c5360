"""Model registry.

Reference: src/utils/get_networks.py:3-29 — {SSLResNet18, SSLResNet50} x
{cifar10, imbalanced_cifar10, imagenet, imbalanced_imagenet} -> ResNetSimCLR.
Synthetic datasets map to the same architectures.
"""

from .ssl_resnet import ResNetSimCLR

DATA_ARGS = {
    "cifar10": {"num_classes": 10},
    "imbalanced_cifar10": {"num_classes": 10},
    "imagenet": {"num_classes": 1000},
    "imbalanced_imagenet": {"num_classes": 1000},
    "synthetic_cifar10": {"num_classes": 10},
    "synthetic_imagenet": {"num_classes": 1000},
}

MODEL_ARGS = {
    "SSLResNet18": {"base_model": "resnet18"},
    "SSLResNet50": {"base_model": "resnet50"},
}


def get_networks(data_name, model_name):
    if data_name not in DATA_ARGS:
        raise ValueError(f"Unknown dataset {data_name!r}")
    if model_name not in MODEL_ARGS:
        raise ValueError(f"Unknown model {model_name!r}")
    return ResNetSimCLR(MODEL_ARGS[model_name]["base_model"],
                        num_classes=DATA_ARGS[data_name]["num_classes"])
