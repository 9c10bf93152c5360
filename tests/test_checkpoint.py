"""Checkpoint format + resume + pretrained-weight key surgery
(reference: utils/resume_training.py, utils/load_pretrained_weights.py)."""

import os

import numpy as np
import torch

from active_learning_amd.models import get_networks
from active_learning_amd.strategies import RandomSampler
from active_learning_amd.utils.checkpoint import (LAYOUT_MARKER, load_experiment,
                                                  load_pretrained_weights,
                                                  save_experiment,
                                                  state_dict_with_marker)
from helpers import make_strategy


def test_own_ckpt_roundtrip(tmp_path):
    net = get_networks("synthetic_cifar10", "SSLResNet18")
    path = tmp_path / "w.pth"
    torch.save(state_dict_with_marker(net), path)
    net2 = get_networks("synthetic_cifar10", "SSLResNet18")
    load_pretrained_weights(net2, str(path))
    for (k1, v1), (k2, v2) in zip(net.state_dict().items(), net2.state_dict().items()):
        assert k1 == k2
        assert torch.equal(v1, v2), k1


def test_module_prefix_normalization(tmp_path):
    """DDP-wrapped checkpoints carry 'module.'; load into unwrapped net
    (load_pretrained_weights.py:27-60)."""
    net = get_networks("synthetic_cifar10", "SSLResNet18")
    sd = state_dict_with_marker(net)
    wrapped = {("module." + k if k != LAYOUT_MARKER else k): v for k, v in sd.items()}
    path = tmp_path / "w.pth"
    torch.save(wrapped, path)
    net2 = get_networks("synthetic_cifar10", "SSLResNet18")
    load_pretrained_weights(net2, str(path))
    assert torch.equal(net2.state_dict()["linear.weight"], net.state_dict()["linear.weight"])


def test_external_oihw_ckpt_is_permuted(tmp_path):
    """External (torchvision-layout OIHW) conv weights land permuted into the
    native KRSC layout; filters/renames apply (ssp_finetuning.py:34-37)."""
    net = get_networks("synthetic_cifar10", "SSLResNet18")
    target = net.state_dict()["encoder.conv1.weight"]  # (64, 3, 3, 3) KRSC
    src = torch.randn(64, 3, 3, 3)  # OIHW
    ckpt = {"state_dict": {
        "encoder_q.conv1.weight": src,
        "fc.weight": torch.randn(10, 512),       # must be skipped
        "unrelated.thing": torch.randn(3),        # must be dropped (required_key)
    }}
    path = tmp_path / "ssl.pth.tar"
    torch.save(ckpt, path)
    before_linear = net.state_dict()["linear.weight"].clone()
    load_pretrained_weights(net, str(path), replace_key={"encoder_q": "encoder"},
                            skip_key=["fc"], required_key=["encoder_q"])
    got = net.state_dict()["encoder.conv1.weight"]
    assert torch.equal(got, src.permute(0, 2, 3, 1))
    # the randomly initialized linear head survives the merge
    assert torch.equal(net.state_dict()["linear.weight"], before_linear)


def test_experiment_save_resume(tmp_path):
    s = make_strategy(RandomSampler, ckpt_path=str(tmp_path), exp_hash="abc")
    s.update(np.array([1, 2, 3]), 3)
    s.round = 4

    import argparse
    args = argparse.Namespace(ckpt_path=str(tmp_path), exp_name="t", exp_hash="abc",
                              log_dir=str(tmp_path), strategy="RandomSampler",
                              resume_training=False)
    import logging
    save_experiment(s, args, logging.getLogger("ActiveLearning"))
    assert os.path.exists(tmp_path / "t_abc" / "strategy.pick")

    s2, next_round, _exp = load_experiment(args, check_args_match=False)
    assert next_round == 5
    assert s2.cumulative_cost == 3
    assert (s2.idxs_lb == s.idxs_lb).all()
    assert s2.comet_experiment is not None
