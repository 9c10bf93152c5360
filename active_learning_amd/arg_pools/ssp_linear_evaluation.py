"""SSL linear-evaluation pool (frozen backbone, lr 15 head).

Reference: src/arg_pools/ssp_linear_evaluation.py. The headline ImageNet config
(README.md:53, gen_jobs.py:8-13) uses this pool with --freeze_feature.
"""

args_pool = {
    "imagenet": {
        "eval_split": 0.01,
        "loader_tr_args": {"batch_size": 128, "num_workers": 8, "prefetch_factor": 2},
        "loader_te_args": {"batch_size": 128, "num_workers": 8, "prefetch_factor": 2},
        "optimizer": "SGD",
        "optimizer_args": {"lr": 15, "weight_decay": 1e-4, "momentum": 0.9},
        "lr_scheduler": "StepLR",
        "lr_scheduler_args": {"step_size": 20, "gamma": 0.1},
        "init_pretrained_ckpt_path": "../pretrained_ckpt/imagenet/moco_v2_800ep_pretrain.pth.tar",
        "required_key": ["encoder_q"],
        "skip_key": ["fc"],
        "replace_key": {"encoder_q": "encoder"},
    },
    # Synthetic ImageNet variant for benchmarking the linear-eval path without
    # dataset files or a downloadable SSL checkpoint (random-init backbone).
    "synthetic_imagenet": {
        "eval_split": 0.01,
        "loader_tr_args": {"batch_size": 128, "num_workers": 4},
        "loader_te_args": {"batch_size": 128, "num_workers": 4},
        "optimizer": "SGD",
        "optimizer_args": {"lr": 15, "weight_decay": 1e-4, "momentum": 0.9},
        "lr_scheduler": "StepLR",
        "lr_scheduler_args": {"step_size": 20, "gamma": 0.1},
        "rd0_pretrained_ckpt_path": None,
    },
}
