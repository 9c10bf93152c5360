"""Fused softmax + cross-entropy (optionally class-weighted).

Reference: nn.CrossEntropyLoss with optional imbalance weights
(strategy.py:352-356; weights from generate_imbalanced_training_weights,
strategy.py:444-457). Semantics match torch: mean reduction, weighted mean
divides by the sum of selected class weights.

GPU: one fused forward kernel (online softmax + nll + saved probs) and one
fused backward kernel; CPU: explicit torch math used as the test reference.
"""

import torch
import torch.nn.functional as F
from torch.autograd import Function

from .extension import require_extension


class CrossEntropyLogits(Function):
    @staticmethod
    def forward(ctx, logits, targets, class_weights):
        ctx.in_dtype = logits.dtype
        logits = logits.float()
        if logits.is_cuda:
            ext = require_extension()
            losses, probs = ext.ce_fwd(logits, targets,
                                       class_weights if class_weights is not None
                                       else logits.new_empty(0))
        else:
            logp = F.log_softmax(logits, dim=1)
            probs = logp.exp()
            losses = -logp.gather(1, targets[:, None])[:, 0]
        if class_weights is not None:
            w = class_weights.to(logits.device).float()[targets]
            denom = w.sum()
            loss = (losses * w).sum() / denom
            row_scale = w / denom
        else:
            loss = losses.mean()
            row_scale = torch.full((logits.shape[0],), 1.0 / logits.shape[0],
                                   device=logits.device)
        ctx.save_for_backward(probs, targets, row_scale)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        probs, targets, row_scale = ctx.saved_tensors
        scale = row_scale * dloss
        if probs.is_cuda:
            ext = require_extension()
            dlogits = ext.ce_bwd(probs, targets, scale)
        else:
            dlogits = probs * scale[:, None]
            dlogits[torch.arange(len(targets)), targets] -= scale
        return dlogits.to(ctx.in_dtype), None, None


def cross_entropy(logits, targets, class_weights=None):
    return CrossEntropyLogits.apply(logits, targets, class_weights)


class CrossEntropyLoss(torch.nn.Module):
    """Module form mirroring nn.CrossEntropyLoss(weight=..., reduction='mean')."""

    def __init__(self, weight=None):
        super().__init__()
        self.register_buffer("weight", weight if weight is not None else None)

    def forward(self, logits, targets):
        return cross_entropy(logits, targets, self.weight)
