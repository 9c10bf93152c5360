"""Fused inference ops: conv + frozen-stats BN (+ residual + ReLU) in one
kernel pass.

Used on the query/eval paths (always under torch.no_grad with eval-mode BN):
folds scale = gamma/sqrt(var+eps), shift = beta - mean*scale into the conv
epilogue, removing one full read+write of every conv output (53 BN passes per
ResNet-50 forward). Training-mode or grad-enabled calls fall back to the
separate autograd ops.
"""

import os

import torch

from .extension import extension_available, load_extension

_DISABLED = os.environ.get("AL_AMD_DISABLE_FUSED_EVAL") == "1"
_STATS_DISABLED = os.environ.get("AL_AMD_DISABLE_CONV_STATS") == "1"
from .functional import _igemm_eligible, _stem_pads, _wpad_cached, cast_cached


def _tick(t):
    return getattr(t, "_al_tick", 0)


def _bn_fold_cached(bn):
    # key includes _al_tick: the fused optimizers and the BN finalize kernel
    # update these tensors in-kernel, which does NOT bump torch's _version
    # (a stale fold here showed up as one-epoch-old validation accuracy)
    key = (bn.weight._version, bn.bias._version, bn.running_mean._version,
           bn.running_var._version, _tick(bn.weight), _tick(bn.bias),
           _tick(bn.running_mean), _tick(bn.running_var))
    cache = getattr(bn, "_al_fold", None)
    if cache is not None and cache[0] == key:
        return cache[1], cache[2]
    invstd = (bn.running_var.float() + bn.eps).rsqrt()
    scale = (bn.weight.float() * invstd).contiguous()
    shift = (bn.bias.float() - bn.running_mean.float() * scale).contiguous()
    try:
        bn._al_fold = (key, scale, shift)
    except Exception:
        pass
    return scale, shift


def conv_bn_act(conv, bn, x, residual=None):
    """bn(conv(x), residual) with the fused single-kernel path when eligible."""
    if (not _STATS_DISABLED and x.is_cuda and bn.training
            and extension_available()):
        # training: conv computes the BN batch statistics in its epilogue
        from .functional import batch_norm_act, conv2d_with_stats
        y, s, ss = conv2d_with_stats(x, conv.weight, conv.stride, conv.padding)
        return batch_norm_act(y, bn.weight, bn.bias, bn.running_mean,
                              bn.running_var, True, bn.momentum, bn.eps, bn.relu,
                              residual, bn._pg(), pre_sums=(s, ss),
                              fuse_backward=True)
    if (not _DISABLED and x.is_cuda and x.dtype == torch.bfloat16
            and not bn.training
            and not torch.is_grad_enabled() and extension_available()):
        ext = load_extension()
        w_c = cast_cached(conv.weight, x.dtype)
        K, R, S, C = w_c.shape
        scale, shift = _bn_fold_cached(bn)
        res = residual if residual is not None else x.new_empty(0)
        if _igemm_eligible(C, R * S * C):
            return ext.conv2d_fwd_fused(x, w_c, conv.stride, conv.padding,
                                        scale, shift, bn.relu, res)
        rowpad, kdpad = _stem_pads(R, S, C)
        apack = ext.im2col_pack(x, R, S, conv.stride, conv.padding, kdpad, rowpad)
        return ext.conv2d_fwd_fused(apack, _wpad_cached(w_c, kdpad, rowpad), 1, 0,
                                    scale, shift, bn.relu, res)
    return bn(conv(x), residual=residual)
