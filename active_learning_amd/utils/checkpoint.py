"""Checkpoint / resume / pretrained-weight loading.

Format parity with the reference (BASELINE.json: "main_al.py CLI and
checkpoint format"):
  * model weights: torch.save(state_dict) at
    <ckpt_path>/<exp_name>_<exp_hash>/{best_rd_<r>.pth, rd_<r>.pth}
    — possibly with the DDP "module." prefix (strategy.py:165-173,430,440);
  * experiment state: pickles of the whole Strategy object + args Namespace +
    status dict to {strategy,args,status}.pick (resume_training.py:38-53),
    resume returns round+1 (:27-35) and re-attaches a tracker.

load_pretrained_weights reproduces the key surgery of
src/utils/load_pretrained_weights.py:5-66 ('state_dict' unwrap, module.
prefix both ways, replace_key renames, skip/required filters, merge into the
existing dict so the freshly initialized linear head survives) and adds one
MI355X-specific step: external checkpoints store conv weights as (K,C,R,S)
(torchvision OIHW); our native layout is (K,R,S,C), so 4-D conv weights from
checkpoints without the native layout marker are permuted on load.
"""

import logging
import os
import pickle
from collections import OrderedDict

import torch

LAYOUT_MARKER = "__al_amd_layout__"  # present in our own saved state dicts


def state_dict_with_marker(net):
    sd = OrderedDict(net.state_dict())
    sd[LAYOUT_MARKER] = torch.tensor(1)
    return sd


def _maybe_permute_conv(value, target_shape):
    """(K,C,R,S) -> (K,R,S,C) when that reproduces the target shape."""
    if value.dim() == 4 and tuple(value.permute(0, 2, 3, 1).shape) == tuple(target_shape):
        return value.permute(0, 2, 3, 1).contiguous()
    return value


def load_pretrained_weights(net, path, replace_key=None, skip_key=None, required_key=None):
    if path is None:
        return net
    replace_key = replace_key or {}
    init_dict = net.state_dict()
    ckpt = torch.load(path, map_location="cpu", weights_only=False)
    if isinstance(ckpt, dict) and "state_dict" in ckpt:
        ckpt = ckpt["state_dict"]

    native_layout = LAYOUT_MARKER in ckpt
    ckpt = {k: v for k, v in ckpt.items() if k != LAYOUT_MARKER}

    net_dp = next(iter(init_dict)).lower().startswith("module")
    ckpt_dp = next(iter(ckpt)).lower().startswith("module")

    def rename(name):
        for k, v in replace_key.items():
            if k in name:
                return name.replace(k, v)
        return name

    # accept bare strings for the key filters (a string would otherwise be
    # iterated per-character and match almost every key)
    if isinstance(skip_key, str):
        skip_key = [skip_key]
    if isinstance(required_key, str):
        required_key = [required_key]

    def skip(name):
        if skip_key and any(s in name for s in skip_key):
            return True
        if required_key:
            return not any(s in name for s in required_key)
        return False

    new_state = OrderedDict()
    for k, v in ckpt.items():
        if skip(k):
            continue
        if net_dp and not ckpt_dp:
            nk = "module." + rename(k)
        elif not net_dp and ckpt_dp:
            nk = rename(k[len("module."):])
        else:
            nk = rename(k)
        if (not native_layout and torch.is_tensor(v) and v.dim() == 4
                and nk in init_dict):
            bare = nk[len("module."):] if nk.startswith("module.") else nk
            if _is_conv_key(net.module if hasattr(net, "module") else net, bare):
                v = v.permute(0, 2, 3, 1).contiguous()  # OIHW -> OHWI (KRSC)
            else:
                v = _maybe_permute_conv(v, init_dict[nk].shape)
        new_state[nk] = v

    matched = {k: v for k, v in new_state.items()
               if k in init_dict and torch.is_tensor(v) and v.shape == init_dict[k].shape}
    dropped = [k for k in new_state if k not in matched]
    if dropped:
        logging.getLogger("ActiveLearning").warning(
            f"load_pretrained_weights: dropped {len(dropped)} unmatched keys "
            f"(first few: {dropped[:5]})")
    init_dict.update(matched)
    net.load_state_dict(init_dict)
    return net


def _is_conv_key(net, key):
    """True if `key` addresses a native NHWC conv weight (layout KRSC)."""
    from ..models.layers import Conv2dNHWC, ConvTranspose2dNHWC
    mod = net
    parts = key.split(".")[:-1]
    for p in parts:
        if not hasattr(mod, p):
            return False
        mod = getattr(mod, p)
    return isinstance(mod, (Conv2dNHWC, ConvTranspose2dNHWC))


# --------------------------------------------------------------------------- #
# experiment-level save / resume
# --------------------------------------------------------------------------- #

def save_experiment(strategy, args, logger):
    prefix = f"{args.ckpt_path}/{args.exp_name}_{args.exp_hash}"
    os.makedirs(prefix, exist_ok=True)
    status = {"round": strategy.round, "comet_exp_key": strategy.comet_experiment.get_key()}
    tmp = strategy.comet_experiment
    strategy.comet_experiment = None
    try:
        with open(f"{prefix}/status.pick", "wb") as fh:
            pickle.dump(status, fh)
        with open(f"{prefix}/strategy.pick", "wb") as fh:
            pickle.dump(strategy, fh)
        with open(f"{prefix}/args.pick", "wb") as fh:
            pickle.dump(args, fh)
    finally:
        strategy.comet_experiment = tmp
    logger.info(f"Save experiment {args.exp_name} at round {strategy.round}")


def load_experiment(args, check_args_match=True):
    from .tracking import ExistingExperiment
    assert args.exp_hash
    prefix = f"{args.ckpt_path}/{args.exp_name}_{args.exp_hash}"
    if not os.path.exists(prefix):
        raise ValueError("checkpoint does not exist")
    with open(f"{prefix}/args.pick", "rb") as fh:
        prev_args = pickle.load(fh)
    ignore = {"resume_training", "exp_name", "world_size"}
    cur = {k: v for k, v in vars(args).items() if k not in ignore}
    prev = {k: v for k, v in vars(prev_args).items() if k not in ignore}
    if check_args_match and cur != prev:
        logging.warning("Loaded experiment however args are not the same!")
        logging.warning(f"Initial args: {prev_args}")
        logging.warning(f"Current args: {args}")
    with open(f"{prefix}/status.pick", "rb") as fh:
        status = pickle.load(fh)
    with open(f"{prefix}/strategy.pick", "rb") as fh:
        strategy = pickle.load(fh)
    experiment = ExistingExperiment(previous_experiment=status["comet_exp_key"],
                                    log_dir=args.log_dir,
                                    mirror_comet=getattr(args, "enable_comet", False))
    experiment.add_tag(args.exp_name)
    experiment.add_tag(args.strategy)
    strategy.comet_experiment = experiment
    return strategy, status["round"] + 1, experiment
