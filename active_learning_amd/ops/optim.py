"""Fused optimizers.

Reference reaches SGD+momentum+weight-decay through torch.optim (built from
config strings by eval at strategy.py:345-350; hyperparameters in
arg_pools/default.py:39-40). Here: an SGD with a single fused HIP update
kernel per tensor on GPU (update + momentum + weight decay in one pass over
fp32 master params), torch math on CPU. Adam (VAAL's VAE/discriminator,
vaal_sampler.py:139-140) likewise.

Constructor signatures match torch.optim so arg-pool optimizer_args apply
unchanged; the trainer maps names through OPTIMIZERS.
"""

import math

import torch

from .extension import extension_available, load_extension
from .functional import bump_tick


def _invalidate(p):
    """Native updates mutate storage without bumping torch's version counter;
    drop the per-version bf16 weight cache (ops/functional.cast_cached)."""
    if getattr(p, "_al_cast", None) is not None:
        p._al_cast = None


class FusedSGD(torch.optim.Optimizer):
    def __init__(self, params, lr, momentum=0.0, weight_decay=0.0, dampening=0.0,
                 nesterov=False):
        if nesterov:
            raise ValueError("nesterov not supported")
        if dampening != 0:
            raise ValueError("dampening not supported")
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay,
                        dampening=dampening)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        ext = load_extension() if extension_available() else None
        for group in self.param_groups:
            lr = group["lr"]
            momentum = group["momentum"]
            wd = group["weight_decay"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad
                state = self.state[p]
                if momentum != 0 and "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(p, dtype=torch.float32)
                buf = state.get("momentum_buffer")
                if p.is_cuda and ext is not None:
                    cache = getattr(p, "_al_cast", None)
                    shadow = (cache[1] if cache is not None
                              and cache[1].numel() == p.numel()
                              and cache[1].dtype == torch.bfloat16
                              else p.new_empty(0, dtype=torch.bfloat16))
                    ext.sgd_step(p, g.to(torch.float32),
                                 buf if buf is not None else p.new_empty(0),
                                 lr, momentum, wd, shadow)
                    if shadow.numel():
                        # the kernel refreshed the bf16 copy in-pass; derived
                        # caches (bwd-data permute) must be recomputed
                        if getattr(cache[1], "_al_wt", None) is not None:
                            cache[1]._al_wt = None
                        p._al_cast = (p._version, cache[1])
                    else:
                        _invalidate(p)
                    bump_tick(p)  # in-kernel update: no _version bump
                else:
                    gf = g.float()
                    if wd != 0:
                        gf = gf.add(p, alpha=wd)
                    if momentum != 0:
                        buf.mul_(momentum).add_(gf)
                        gf = buf
                    p.add_(gf, alpha=-lr)
        return loss


class FusedAdam(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        ext = load_extension() if extension_available() else None
        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps = group["eps"]
            wd = group["weight_decay"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if not state:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                state["step"] += 1
                t = state["step"]
                bc1 = 1 - beta1 ** t
                bc2 = 1 - beta2 ** t
                if p.is_cuda and ext is not None:
                    ext.adam_step(p, p.grad.to(torch.float32), state["exp_avg"],
                                  state["exp_avg_sq"], lr, beta1, beta2, eps, wd, bc1, bc2)
                    _invalidate(p)
                    bump_tick(p)  # in-kernel update: no _version bump
                else:
                    gf = p.grad.float()
                    if wd != 0:
                        gf = gf.add(p, alpha=wd)
                    state["exp_avg"].mul_(beta1).add_(gf, alpha=1 - beta1)
                    state["exp_avg_sq"].mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
                    denom = (state["exp_avg_sq"] / bc2).sqrt_().add_(eps)
                    p.addcdiv_(state["exp_avg"] / bc1, denom, value=-lr)
        return loss


OPTIMIZERS = {
    "SGD": FusedSGD,
    "Adam": FusedAdam,
}


def build_optimizer(name: str, params, **kwargs):
    """Explicit registry replacing the reference's eval-based dispatch
    (strategy.py:345-346)."""
    if name not in OPTIMIZERS:
        raise ValueError(f"Unknown optimizer {name!r}")
    return OPTIMIZERS[name](params, **kwargs)


SCHEDULERS = {
    "CosineAnnealingLR": torch.optim.lr_scheduler.CosineAnnealingLR,
    "StepLR": torch.optim.lr_scheduler.StepLR,
    "MultiStepLR": torch.optim.lr_scheduler.MultiStepLR,
}


def build_scheduler(name: str, optimizer, **kwargs):
    """Explicit registry replacing eval-based dispatch (strategy.py:348-350).
    LR schedules are host-side scalar math."""
    if name not in SCHEDULERS:
        raise ValueError(f"Unknown lr scheduler {name!r}")
    return SCHEDULERS[name](optimizer, **kwargs)
