"""Shared test fixtures: tiny synthetic strategy harness."""

import numpy as np
import torch

from active_learning_amd.data.synthetic import get_data_synthetic
from active_learning_amd.models import get_networks
from active_learning_amd.utils.pool_init import generate_eval_idxs
from active_learning_amd.utils.tracking import Experiment


def default_kwargs(**over):
    kw = dict(early_stop_patience=2, n_epoch=1, world_size=1, model="SSLResNet18",
              freeze_feature=False, ckpt_path="/tmp/al_test_ckpt", exp_name="t",
              exp_hash="h", subset_labeled=None, subset_unlabeled=None, partitions=2,
              vae_latent_dim=8, vaal_adversary_param=1.0, lr_vae=5e-4,
              lr_discriminator=5e-4)
    kw.update(over)
    return kw


def tiny_train_args(batch=16):
    return {
        "eval_split": 0.1,
        "loader_tr_args": {"batch_size": batch, "num_workers": 0},
        "loader_te_args": {"batch_size": batch, "num_workers": 0},
        "optimizer": "SGD",
        "optimizer_args": {"lr": 0.05, "weight_decay": 5e-4, "momentum": 0.9},
        "lr_scheduler": "CosineAnnealingLR",
        "lr_scheduler_args": {"T_max": 5},
    }


def make_strategy(strategy_cls, n=60, img=16, seed=0, **kw_over):
    train_set, test_set, al_set = get_data_synthetic(10, n, 20, (3, img, img), seed=seed)
    net = get_networks("synthetic_cifar10", "SSLResNet18")
    eval_idxs = np.arange(n - 6, n)  # last 6 samples reserved for eval
    exp = Experiment(disabled=True)
    kw = default_kwargs(**kw_over)
    strat = strategy_cls(train_set, al_set, net, tiny_train_args(), eval_idxs, exp,
                         test_set, **kw)
    return strat
