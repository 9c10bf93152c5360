"""CLI flag surface — parity with the reference parser.

Reference: src/utils/parser.py:7-92 (flag names, defaults, choices kept
identical so reference job strings from gen_jobs.py run unchanged).
"""

import argparse

DEFAULT_CKPT_PATH = "../checkpoint"
DEFAULT_LOG_DIR = "./logs"


def build_parser() -> argparse.ArgumentParser:
    parser = argparse.ArgumentParser(description="MI355X-native active learning")

    # Experiment naming / logging (parser.py:15-21)
    parser.add_argument("--project_name", dest="project_name", default="active-learning",
                        type=str, help="project name of the experiment")
    parser.add_argument("--exp_name", dest="exp_name", default="active_learning", type=str,
                        help="exp_name for specification")
    parser.add_argument("--log_dir", dest="log_dir", default=DEFAULT_LOG_DIR,
                        help="logs are saved here")
    parser.add_argument("--enable_comet", dest="enable_comet", action="store_true",
                        help="Enable Comet ML logging (falls back to the local JSONL "
                             "tracker when comet_ml is not installed).")

    # Dataset (parser.py:24-29)
    parser.add_argument("--dataset", dest="dataset", default="cifar10", type=str,
                        help="name of the dataset")
    parser.add_argument("--dataset_dir", dest="dataset_dir",
                        help="path to the root dir of datasets")
    parser.add_argument("--arg_pool", dest="arg_pool", default="default",
                        help="Dataset specific args to use for this AL experiment")

    # Imbalanced datasets (parser.py:32-39)
    parser.add_argument("--imbalance_type", dest="imbalance_type", default=None,
                        choices=["exp", "step"],
                        help="Imbalance type: step = c/2 majority + c/2 minority classes; "
                             "exp = exponentially decaying class sizes")
    parser.add_argument("--imbalance_factor", dest="imbalance_factor", default=0.1, type=float,
                        help="Imbalance factor.")
    parser.add_argument("--imbalance_seed", dest="imbalance_seed", default=0, type=int,
                        help="Seed for generating the imbalanced dataset.")

    # Global active-learning parameters (parser.py:42-54)
    parser.add_argument("--strategy", dest="strategy", default="RandomSampler",
                        help="strategy for active learning")
    parser.add_argument("--rounds", dest="rounds", type=int, default=5,
                        help="# of rounds of active learning")
    parser.add_argument("--round_budget", dest="round_budget", type=float, default=5000,
                        help="Budget to exhaust per round.")
    parser.add_argument("--freeze_feature", dest="freeze_feature", default=False,
                        action="store_true",
                        help="Train only the final linear layer; backbone frozen")
    parser.add_argument("--init_pool_size", dest="init_pool_size", type=int, default=-1)
    parser.add_argument("--init_pool_type", dest="init_pool_type", type=str, default="random",
                        choices=["random", "random_balance"])

    # Global training args (parser.py:57-67)
    parser.add_argument("--model", dest="model", default="SSLResNet18", type=str)
    parser.add_argument("--resume_training", dest="resume_training", action="store_true")
    parser.add_argument("--exp_hash", dest="exp_hash", default=None, type=str)
    parser.add_argument("--ckpt_path", dest="ckpt_path", type=str, default=DEFAULT_CKPT_PATH)
    parser.add_argument("--n_epoch", dest="n_epoch", type=int, default=60,
                        help="The number of training epochs.")
    parser.add_argument("--early_stop_patience", dest="early_stop_patience", type=int,
                        default=30,
                        help="Early stopping patience; 0 disables early stopping.")

    # Debug (parser.py:70-71)
    parser.add_argument("--debug_mode", dest="debug_mode", default=False, action="store_true",
                        help="Use debug mode (tiny datasets, fixed pool indices)")

    # Partitioned coreset / BADGE (parser.py:74-79)
    parser.add_argument("--subset_labeled", dest="subset_labeled", type=int,
                        help="Number of labeled samples subsampled for coreset.")
    parser.add_argument("--subset_unlabeled", dest="subset_unlabeled", type=int,
                        help="Number of unlabeled samples subsampled for coreset.")
    parser.add_argument("--partitions", dest="partitions", type=int, default=1,
                        help="Number of random partitions for partitioned coreset/BADGE.")

    # VAAL (parser.py:82-90)
    parser.add_argument("--vae_latent_dim", dest="vae_latent_dim", type=int, default=64,
                        help="ImageNet 64, CIFAR-10 32")
    parser.add_argument("--vaal_adversary_param", dest="vaal_adversary_param", type=float,
                        default=10.0,
                        help="lambda2 in the VAAL paper: 10 for ImageNet, 1 for CIFAR-10")
    parser.add_argument("--lr_vae", dest="lr_vae", type=float, default=5e-5,
                        help="ImageNet 5e-5, CIFAR 5e-4")
    parser.add_argument("--lr_discriminator", dest="lr_discriminator", type=float, default=1e-3,
                        help="ImageNet 1e-3, CIFAR 5e-4")

    # MI355X-native extras (not in the reference; all optional with safe defaults)
    parser.add_argument("--compute_dtype", dest="compute_dtype", default="bf16",
                        choices=["bf16", "fp32"],
                        help="Device compute dtype for the HIP kernel path (bf16 default).")
    parser.add_argument("--world_size", dest="world_size", type=int, default=None,
                        help="Override number of GPUs (default: torch.cuda.device_count()).")
    return parser


def get_args(argv=None):
    return build_parser().parse_args(argv)
