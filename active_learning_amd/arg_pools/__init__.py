"""Arg pools: per-experiment training-hyperparameter dicts.

Reference: src/arg_pools/*.py, loaded there by ``exec`` (src/main_al.py:48-49).
Here the pools are plain data modules resolved through an explicit registry —
same module names, same dict keys, no ``exec``.
"""

import importlib

_POOL_MODULES = (
    "default",
    "ssp_finetuning",
    "ssp_linear_evaluation",
    "ssp_finetuning_imbalanced_cifar10_imb_0_1",
    "ssp_finetuning_imbalanced_cifar10_imb_0_01",
)


def get_arg_pool(name: str) -> dict:
    """Return the ``args_pool`` dict of the named pool module."""
    if name not in _POOL_MODULES:
        raise ValueError(f"Unknown arg pool {name!r}; available: {_POOL_MODULES}")
    mod = importlib.import_module(f"{__name__}.{name}")
    return mod.args_pool


def available_pools():
    return _POOL_MODULES
