"""CLI-surface parity + end-to-end debug-mode rounds (the reference's de
facto integration harness)."""

import numpy as np
import pytest

from active_learning_amd.cli import get_args
from active_learning_amd.main_al import main

REFERENCE_FLAGS = [
    "project_name", "exp_name", "log_dir", "enable_comet", "dataset", "dataset_dir",
    "arg_pool", "imbalance_type", "imbalance_factor", "imbalance_seed", "strategy",
    "rounds", "round_budget", "freeze_feature", "init_pool_size", "init_pool_type",
    "model", "resume_training", "exp_hash", "ckpt_path", "n_epoch",
    "early_stop_patience", "debug_mode", "subset_labeled", "subset_unlabeled",
    "partitions", "vae_latent_dim", "vaal_adversary_param", "lr_vae",
    "lr_discriminator",
]


def test_cli_flag_surface():
    args = get_args([])
    for flag in REFERENCE_FLAGS:
        assert hasattr(args, flag), f"missing reference flag --{flag}"
    # reference defaults (src/utils/parser.py)
    assert args.strategy == "RandomSampler"
    assert args.rounds == 5
    assert args.round_budget == 5000
    assert args.n_epoch == 60
    assert args.early_stop_patience == 30
    assert args.init_pool_size == -1
    assert args.partitions == 1
    assert args.vaal_adversary_param == 10.0


def _debug_args(tmp_path, extra=()):
    return get_args([
        "--dataset", "synthetic_cifar10", "--rounds", "2", "--round_budget", "10",
        "--n_epoch", "2", "--early_stop_patience", "2", "--debug_mode",
        "--ckpt_path", str(tmp_path / "ckpt"), "--log_dir", str(tmp_path / "logs"),
        "--model", "SSLResNet18", *extra])


@pytest.mark.parametrize("strategy", ["RandomSampler", "MarginSampler", "MASESampler"])
def test_debug_round_loop(tmp_path, strategy):
    args = _debug_args(tmp_path, ["--strategy", strategy])
    s = main(args)
    assert s.cumulative_cost == 15  # 5 init + 10 queried
    assert s.idxs_lb.sum() == 15


def test_resume_from_checkpoint(tmp_path):
    args = _debug_args(tmp_path, ["--strategy", "RandomSampler", "--exp_hash", "rz"])
    s1 = main(args)
    # resume continues at the next round with state intact
    args2 = _debug_args(tmp_path, ["--strategy", "RandomSampler", "--exp_hash", "rz"])
    args2.resume_training = True
    args2.rounds = 3
    s2 = main(args2)
    assert s2.cumulative_cost == s1.cumulative_cost + 10
    assert s2.round == 2


def test_freeze_feature_linear_eval(tmp_path):
    args = _debug_args(tmp_path, ["--strategy", "RandomSampler", "--freeze_feature"])
    s = main(args)
    assert s.idxs_lb.sum() == 15


def test_gen_jobs_commands_parse(capsys):
    """Every command line gen_jobs emits must be accepted by our CLI parser
    and name a registered strategy (reference gen_jobs.py parity)."""
    import shlex
    from active_learning_amd import gen_jobs
    from active_learning_amd.cli import build_parser
    from active_learning_amd.strategies import get_strategy

    gen_jobs.linear_evaluation_imagenet_experiments(dataset_dir="/data")
    gen_jobs.end_to_end_imagenet_experiments_pretrained(dataset_dir="/data")
    gen_jobs.cifar10_experiments(dataset_dir="/data")
    out = capsys.readouterr().out.strip().splitlines()
    assert len(out) == 9 + 9 + 11
    parser = build_parser()
    for line in out:
        args = parser.parse_args(shlex.split(line)[2:])  # drop "python main_al.py"
        assert get_strategy(args.strategy) is not None
        assert args.rounds > 0 and args.round_budget > 0


def test_compute_dtype_fp32_round(tmp_path):
    """--compute_dtype fp32 disables the bf16 cast (CPU runs full fp32;
    PARITY.md documents the GPU path as bf16-with-fp32-master)."""
    args = _debug_args(tmp_path, ["--strategy", "RandomSampler",
                                  "--compute_dtype", "fp32"])
    s = main(args)
    assert s.net.encoder.compute_dtype is None
    assert s.round == 1
