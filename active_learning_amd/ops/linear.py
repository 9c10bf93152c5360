"""First-party fp32 linear head (classifier) on exact-f32 MFMA.

Replaces the last library GEMM on the hot path (reference:
resnet_simclr.py:22,33,38 reaches cuBLAS through nn.Linear). NativeLinear is
state_dict-compatible with nn.Linear (same `weight`/`bias` keys and shapes);
CUDA fp32 runs the linear.hip kernels (mfma_f32_16x16x4f32 — EXACT fp32
numerics, guide §3), everything else falls back to F.linear.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from .extension import extension_available, load_extension


def _native(x, weight):
    return (x.is_cuda and x.dtype == torch.float32
            and weight.dtype == torch.float32 and extension_available())


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        if _native(x, weight):
            ext = load_extension()
            return ext.linear_fwd(x.contiguous(), weight,
                                  bias if bias is not None
                                  else weight.new_empty(0))
        return F.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy = dy.contiguous()
        if _native(x, weight) and dy.dtype == torch.float32:
            ext = load_extension()
            dx, dw, db = ext.linear_bwd(dy, x.contiguous(), weight,
                                        ctx.needs_input_grad[0],
                                        ctx.needs_input_grad[1],
                                        ctx.has_bias and ctx.needs_input_grad[2])
            return (dx if ctx.needs_input_grad[0] else None,
                    dw if ctx.needs_input_grad[1] else None,
                    db if (ctx.has_bias and ctx.needs_input_grad[2]) else None)
        dx = dy @ weight if ctx.needs_input_grad[0] else None
        dw = dy.t() @ x if ctx.needs_input_grad[1] else None
        db = dy.sum(0) if (ctx.has_bias and ctx.needs_input_grad[2]) else None
        return dx, dw, db


def linear(x, weight, bias=None):
    return _LinearFn.apply(x, weight, bias)


class NativeLinear(nn.Linear):
    """Drop-in nn.Linear whose CUDA fp32 path runs the first-party MFMA
    kernels. Identical parameters/state_dict keys."""

    def forward(self, x):
        return linear(x, self.weight, self.bias)
