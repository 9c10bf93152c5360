"""Arg pools: every registered pool must provide complete, trainer-usable
config dicts (reference: the five src/arg_pools modules, exec-imported at
main_al.py:48; explicit registry here)."""

import pytest

from active_learning_amd.arg_pools import get_arg_pool
from active_learning_amd.ops.optim import OPTIMIZERS, SCHEDULERS

POOLS = ["default", "ssp_finetuning", "ssp_linear_evaluation",
         "ssp_finetuning_imbalanced_cifar10_imb_0_01",
         "ssp_finetuning_imbalanced_cifar10_imb_0_1"]

REQUIRED = {"eval_split", "loader_tr_args", "loader_te_args", "optimizer",
            "optimizer_args", "lr_scheduler", "lr_scheduler_args"}


@pytest.mark.parametrize("pool", POOLS)
def test_pool_entries_complete(pool):
    p = get_arg_pool(pool)
    assert p, f"{pool} empty"
    for dataset, cfg in p.items():
        missing = REQUIRED - set(cfg)
        assert not missing, f"{pool}[{dataset}] missing {missing}"
        assert 0 < cfg["eval_split"] < 1
        assert cfg["optimizer"] in OPTIMIZERS
        assert cfg["lr_scheduler"] in SCHEDULERS
        assert "batch_size" in cfg["loader_tr_args"]
        assert "lr" in cfg["optimizer_args"]


def test_unknown_pool_raises():
    with pytest.raises(Exception):
        get_arg_pool("nonexistent_pool")
