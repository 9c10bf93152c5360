"""Autograd-integrated NHWC ops: conv / BN(+residual)+ReLU / pooling.

Layout contract (MI355X-native, chosen for MFMA/LDS tiling and coalescing):
  activations: (N, H, W, C) contiguous, bf16 on GPU / fp32 on CPU
  conv weight: (K, R, S, C) fp32 master parameter; cast to the activation
               dtype inside the Function (so weight gradients are produced in
               fp32 straight from the accumulator — no bf16 round-trip).

Reference counterparts: these are the kernels the reference reaches through
cuDNN/ATen: conv fwd/bwd-data/bwd-weight (strategy.py:268,270),
fused BN+ReLU incl. the frozen-stats eval variant (strategy.py:366-367),
SyncBN stats exchange (strategy.py:292), maxpool/avgpool. The BN op
additionally fuses the residual add + ReLU that ends every ResNet block, so a
bottleneck tail is one kernel instead of three memory passes.

GPU dispatch goes to active_learning_amd._C (HIP/gfx950); CPU fallback uses
plain PyTorch and defines the semantics the kernels are tested against.
"""

import os

import numpy as np
import torch
import torch.distributed as dist
import torch.nn.functional as F
from torch.autograd import Function

from .extension import require_extension


def _native_ok(x):
    """The hand-written gfx950 kernels take bf16 NHWC; fp32-on-GPU
    (--compute_dtype fp32) runs the same ops through the torch math below —
    executed on-device via MIOpen/rocBLAS — as a numerics escape hatch
    (reference trains fp32 throughout)."""
    return x.is_cuda and x.dtype == torch.bfloat16


def _cpu_conv_args(x_nhwc, w_krsc):
    return x_nhwc.permute(0, 3, 1, 2), w_krsc.permute(0, 3, 1, 2)


def bump_tick(t):
    """In-kernel updates mutate storage without bumping torch's _version;
    derived caches (the folded-BN eval scale/shift, ops/fused.py) key on
    this counter as well."""
    if t is not None:
        try:
            t._al_tick = getattr(t, "_al_tick", 0) + 1
        except Exception:
            pass


def cast_cached(weight, dtype):
    """Per-version cached bf16 copy of an fp32 master weight (saves a cast +
    copy per layer per step; invalidated by optimizers via
    invalidate_param_cache)."""
    if weight.dtype == dtype:
        return weight
    cache = getattr(weight, "_al_cast", None)
    ver = weight._version
    if cache is not None and cache[0] == ver and cache[1].dtype == dtype:
        return cache[1]
    w = weight.detach().to(dtype)
    try:
        weight._al_cast = (ver, w)
    except Exception:
        pass
    return w


def invalidate_param_cache(p):
    if getattr(p, "_al_cast", None) is not None:
        p._al_cast = None


def wt_cached(w_c):
    """(K,R,S,C) -> (C,R,S,K) permutation for the bwd-data contraction,
    cached on the bf16 weight copy (which is itself version-cached)."""
    wt = getattr(w_c, "_al_wt", None)
    if wt is None:  # also reset to None by FusedSGD after in-place refresh
        wt = w_c.permute(3, 1, 2, 0).contiguous()
        try:
            w_c._al_wt = wt
        except Exception:
            pass
    return wt


def _igemm_eligible(C, KD):
    """Shapes the MFMA igemm gathers directly (C%8 for 16B chunks, KD%64 for
    exact contraction tiles). Others — the C=3 stems — go through the packed
    im2col path so they still run on MFMA instead of the slow direct kernel."""
    return C % 8 == 0 and KD % 64 == 0


def _stem_pads(R, S, C):
    """Row-padded im2col geometry: each filter row (S*C values) padded to an
    8-aligned rowpad so every 16B chunk of the packed matrix lies within one
    row and the pack kernel's reads are lane-contiguous."""
    rowpad = ((S * C + 7) // 8) * 8
    kdpad = ((R * rowpad + 63) // 64) * 64
    return rowpad, kdpad


def _wpad_cached(w_c, kdpad, rowpad):
    """Zero-padded (K,1,1,kdpad) copy of the stem weight in the row-padded
    im2col layout, cached on the bf16 weight (rebuilt only when the bf16
    copy is recreated)."""
    wpad = getattr(w_c, "_al_wpad", None)
    K, R, S, C = w_c.shape
    if wpad is None:
        wpad = torch.zeros(K, 1, 1, kdpad, dtype=w_c.dtype, device=w_c.device)
        try:
            w_c._al_wpad = wpad
        except Exception:
            pass
    wpad.view(K, kdpad)[:, :R * rowpad].view(K, R, rowpad)[:, :, :S * C] = \
        w_c.reshape(K, R, S * C)
    return wpad


def _gpu_conv_fwd_packed(ext, x, w_c, stride, padding):
    """Stem path: A = im2col(x) in the row-padded layout, conv as 1x1 igemm.
    Returns (y, apack); apack is reused by the backward wgrad."""
    K, R, S, C = w_c.shape
    rowpad, kdpad = _stem_pads(R, S, C)
    apack = ext.im2col_pack(x, R, S, stride, padding, kdpad, rowpad)
    return ext.conv2d_fwd(apack, _wpad_cached(w_c, kdpad, rowpad), 1, 0), apack


def _gpu_conv_wgrad_packed(ext, dy, apack, w_shape):
    K, R, S, C = w_shape
    rowpad, kdpad = _stem_pads(R, S, C)
    assert kdpad == apack.shape[-1]
    dwpad = ext.conv2d_bwd_weight(dy, apack, 1, 1, 1, 0)  # (K,1,1,kdpad) fp32
    return (dwpad.view(K, kdpad)[:, :R * rowpad].view(K, R, rowpad)
            [:, :, :S * C].reshape(K, R, S, C).contiguous())


class _GradArena:
    """One flat fp32 buffer for every conv weight gradient.

    conv2d_bwd_weight's split-K accumulation needs a zeroed output, which
    cost ~50 small fill launches per training step. The arena hands out
    views of a single flat buffer instead, zeroed ONCE at the start of each
    backward (the first take() after any forward marked the step dirty).
    Keys are (weight data_ptr, shape); unknown keys fall back to the plain
    self-allocating path for that call and join the arena at the next
    rebuild. NOTE: assumes the standard step discipline (one backward per
    forward, optimizer consumes grads before the next forward) — gradient
    accumulation across multiple backwards must disable it
    (AL_GRAD_ARENA=0), see PARITY.md.
    """

    def __init__(self):
        self.enabled = os.environ.get("AL_GRAD_ARENA", "1") == "1"
        self.views = {}
        self.shapes = {}
        self.pending = []
        self.flat = None
        self.dirty = False
        self.touched = set()  # keys taken since the last rebuild/zero

    def mark_step(self):
        self.dirty = True

    def take(self, key, shape, device):
        if not self.enabled:
            return None
        if self.dirty:
            if self.pending:
                self._rebuild(device)
            elif self.flat is not None:
                self.flat.zero_()
                self.touched = set()
            self.dirty = False
        self.touched.add(key)
        v = self.views.get(key)
        if v is not None:
            return v
        if (key, tuple(shape)) not in [(k, s) for k, s in self.pending]:
            self.pending.append((key, tuple(shape)))
        return None

    def _rebuild(self, device):
        # evict keys not used since the previous rebuild/zero: per-round
        # weight re-init allocates fresh bf16 weight copies (new data_ptr
        # keys), so without eviction the flat buffer grows every AL round
        live = self.touched | {k for k, _ in self.pending}
        self.shapes = {k: s for k, s in self.shapes.items() if k in live}
        for key, shape in self.pending:
            self.shapes[key] = shape
        self.pending = []
        self.touched = set()
        total = sum(int(np.prod(s)) for s in self.shapes.values())
        self.flat = torch.zeros(total, dtype=torch.float32, device=device)
        off = 0
        self.views = {}
        for key, shape in self.shapes.items():
            n = int(np.prod(shape))
            self.views[key] = self.flat.narrow(0, off, n).view(shape)
            off += n


_grad_arena = _GradArena()


class _StatsArena:
    """Flat fp32 slab for the conv-epilogue BN-stat accumulators (2K floats
    per conv): ONE zero fill per forward pass instead of one per conv (the
    per-conv torch.zeros showed as ~53 FillFunctor launches/step). A pass
    boundary is detected when a key repeats (the same conv runs again)."""

    def __init__(self):
        self.enabled = os.environ.get("AL_STATS_ARENA", "1") == "1"
        self.views = {}
        self.shapes = {}
        self.pending = []
        self.flat = None
        self.seen = set()

    def take(self, key, n, device):
        if not self.enabled:
            return None
        if key in self.seen:
            # new forward pass: previous stats were consumed by bn_finalize
            if self.pending:
                self._rebuild(device)
            elif self.flat is not None:
                self.flat.zero_()
            self.seen = set()
        self.seen.add(key)
        v = self.views.get(key)
        if v is not None:
            return v
        if key not in [k for k, _ in self.pending]:
            self.pending.append((key, n))
        return None

    def _rebuild(self, device):
        live = self.seen | {k for k, _ in self.pending}
        self.shapes = {k: s for k, s in self.shapes.items() if k in live}
        for key, n in self.pending:
            self.shapes[key] = n
        self.pending = []
        total = sum(self.shapes.values())
        self.flat = torch.zeros(total, dtype=torch.float32, device=device)
        off = 0
        self.views = {}
        for key, n in self.shapes.items():
            self.views[key] = self.flat.narrow(0, off, n)
            off += n


_stats_arena = _StatsArena()

# Measured NEGATIVE at B=256 (profiles/fused_breakdown.md): the epilogue's
# dependent global mask/x reads stall the bwd-data kernels (igemm<1,4,1>
# +57%, stride-2 parity +160%) by more than the removed bn_bwd_reduce pass
# saves. Kept opt-in for shapes/hardware where the tradeoff differs.
_BNBACK_ON = os.environ.get("AL_BNBACK_FUSE", "0") == "1"

# Residual-gradient fan-in fusion: a residual join gives the block input TWO
# gradient contributions (the BN's dres and the block's conv1 dx), which
# autograd sums in a separate elementwise pass (~1.3 ms/step of
# CUDAFunctor_add at B=256). Instead the BN backward attaches dres to the
# shared tensor and returns None for the residual grad; the consuming conv's
# bwd-data accumulates it in its epilogue (one extra coalesced read vs a
# whole 3-pass add). Only engages when the residual tensor IS a conv input
# (marked at forward) — transition blocks (residual = downsample output)
# keep the normal path.
_RESBACK_ON = os.environ.get("AL_RESBACK_FUSE", "1") == "1"


def _claim_bnback(x):
    """Claim the upstream BatchNorm's backward side-channel (set by
    BatchNormAct.forward on its output when it has a single consumer): this
    conv's bwd-data will then mask dx in its epilogue and pre-reduce the
    BN-backward channel sums, so bn_bwd_reduce never runs for that BN."""
    if not _BNBACK_ON:
        return None
    bnb = getattr(x, "_al_bnback", None)
    if bnb is None or getattr(x, "_al_bnback_claimed", False):
        return None
    try:
        x._al_bnback_claimed = True
    except Exception:
        return None
    return bnb


class Conv2dNHWC(Function):
    """y[N,P,Q,K] = conv(x[N,H,W,C], w[K,R,S,C]; stride, pad), no bias
    (ResNet convs carry no bias; BN follows)."""

    @staticmethod
    def forward(ctx, x, weight, stride, padding):
        w_c = cast_cached(weight, x.dtype)  # fp32 master -> compute dtype
        ctx.save_for_backward(x)
        ctx.w_c = w_c
        ctx.stride, ctx.padding = stride, padding
        ctx.weight_dtype = weight.dtype
        ctx.bnb = _claim_bnback(x)
        if _RESBACK_ON and _native_ok(x) and x.requires_grad:
            try:
                x._al_res_consumer = True
            except Exception:
                pass
        if _native_ok(x):
            ext = require_extension()
            _grad_arena.mark_step()
            K, R, S, C = w_c.shape
            if _igemm_eligible(C, R * S * C):
                ctx.apack = None
                return ext.conv2d_fwd(x, w_c, stride, padding)
            y, apack = _gpu_conv_fwd_packed(ext, x, w_c, stride, padding)
            ctx.apack = apack
            return y
        xc, wc = _cpu_conv_args(x, w_c)
        y = F.conv2d(xc.float(), wc.float(), stride=stride, padding=padding)
        return y.to(x.dtype).permute(0, 2, 3, 1).contiguous()

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        w_c = ctx.w_c
        dy = dy.contiguous()
        dx = dw = None
        # residual-grad side channel: the upstream BN's backward already ran
        # (topological order) and may have parked dres on our input tensor
        dres_s = None
        if ctx.needs_input_grad[0]:
            dres_s = getattr(x, "_al_dres", None)
            if dres_s is not None:
                try:
                    x._al_dres = None  # consume once
                except Exception:
                    pass
        if _native_ok(x):
            ext = require_extension()
            K, R, S, C = w_c.shape
            if ctx.needs_input_grad[0]:
                if R == 1 and S == 1 and ctx.stride > 1 and ctx.padding == 0:
                    # 1x1 strided conv (ResNet downsample): dx is nonzero only
                    # at stride-aligned pixels — dense GEMM on dY + a one-pass
                    # scatter kernel (zeros + copy fused) instead of a
                    # 4x-redundant gather igemm
                    tmp = ext.conv2d_fwd(dy, wt_cached(w_c), 1, 0)  # (N,P,Q,C)
                    dx = ext.scatter_s2(tmp, x.shape[1], x.shape[2], ctx.stride)
                elif (getattr(ctx, "bnb", None) is not None
                      and (R * S * K) % 64 == 0 and K % 8 == 0):
                    # fuse the upstream BN's backward reduction into this
                    # bwd-data epilogue: dx comes back PRE-MASKED with the
                    # channel sums attached for BatchNormAct.backward
                    mask, xbn, mean, invstd = ctx.bnb
                    dx, s, sx = ext.conv2d_bwd_data_bn(
                        dy, wt_cached(w_c), ctx.stride, ctx.padding,
                        x.shape[1], x.shape[2], mask, xbn, mean, invstd)
                    try:
                        # stamp the tensor version: if autograd later
                        # accumulates another grad INTO this tensor in place
                        # (fan-in reuses the buffer), the version bumps and
                        # the BN backward falls back to its own reduce
                        dx._al_bnsums = (mask, s, sx, dx._version)
                    except Exception:
                        pass
                elif (dres_s is not None and dres_s.dtype == dy.dtype
                      and dres_s.is_contiguous()):
                    dx = ext.conv2d_bwd_data_res(dy, wt_cached(w_c), ctx.stride,
                                                 ctx.padding, x.shape[1],
                                                 x.shape[2], dres_s)
                    dres_s = None
                else:
                    dx = ext.conv2d_bwd_data(dy, wt_cached(w_c), ctx.stride,
                                             ctx.padding, x.shape[1], x.shape[2])
            if ctx.needs_input_grad[1]:
                if _igemm_eligible(C, R * S * C):
                    buf = _grad_arena.take((w_c.data_ptr(), w_c.shape), w_c.shape,
                                           dy.device)
                    if buf is not None:
                        dw = ext.conv2d_bwd_weight_into(dy, x, R, S, ctx.stride,
                                                        ctx.padding, buf)
                    else:
                        dw = ext.conv2d_bwd_weight(dy, x, R, S, ctx.stride,
                                                   ctx.padding)
                else:
                    dw = _gpu_conv_wgrad_packed(ext, dy, ctx.apack, w_c.shape)
        else:
            xc, wc = _cpu_conv_args(x, w_c)
            dyc = dy.permute(0, 3, 1, 2).float()
            if ctx.needs_input_grad[0]:
                dx = torch.nn.grad.conv2d_input(list(xc.shape), wc.float(), dyc,
                                                stride=ctx.stride, padding=ctx.padding)
                dx = dx.to(x.dtype).permute(0, 2, 3, 1).contiguous()
            if ctx.needs_input_grad[1]:
                dw = torch.nn.grad.conv2d_weight(xc.float(), list(wc.shape), dyc,
                                                 stride=ctx.stride, padding=ctx.padding)
                dw = dw.permute(0, 2, 3, 1).contiguous()
        if dres_s is not None and dx is not None:
            dx = dx + dres_s  # routes that could not fuse the accumulation
        if dw is not None:
            dw = dw.to(ctx.weight_dtype)
        return dx, dw, None, None


def conv2d(x, weight, stride=1, padding=0):
    return Conv2dNHWC.apply(x, weight, stride, padding)


class Conv2dNHWCStats(Function):
    """Conv that also returns per-column (sum, sumsq) of its output, computed
    in the conv epilogue — feeds the following BatchNorm's batch statistics
    without a separate full read of the activation (training path)."""

    @staticmethod
    def forward(ctx, x, weight, stride, padding):
        # without this, autograd MATERIALIZES a zero [K] tensor for each of
        # the two non-differentiable outputs (s, ss) on every backward:
        # 2 x 53 FillFunctor launches/step at B=256 (tools/fill_audit.py)
        ctx.set_materialize_grads(False)
        w_c = cast_cached(weight, x.dtype)
        ctx.save_for_backward(x)
        ctx.w_c = w_c
        ctx.stride, ctx.padding = stride, padding
        ctx.weight_dtype = weight.dtype
        ctx.bnb = _claim_bnback(x)
        if _RESBACK_ON and _native_ok(x) and x.requires_grad:
            try:
                x._al_res_consumer = True
            except Exception:
                pass
        if _native_ok(x):
            ext = require_extension()
            _grad_arena.mark_step()
            K, R, S, C = w_c.shape
            buf = _stats_arena.take((w_c.data_ptr(), K), 2 * K, x.device)
            empty = x.new_empty(0, dtype=torch.float32)
            if _igemm_eligible(C, R * S * C):
                ctx.apack = None
                y, s, ss = ext.conv2d_fwd_stats(x, w_c, stride, padding,
                                                buf if buf is not None else empty)
            else:
                K, R, S, C = w_c.shape
                rowpad, kdpad = _stem_pads(R, S, C)
                apack = ext.im2col_pack(x, R, S, stride, padding, kdpad, rowpad)
                ctx.apack = apack
                y, s, ss = ext.conv2d_fwd_stats(apack,
                                                _wpad_cached(w_c, kdpad, rowpad),
                                                1, 0,
                                                buf if buf is not None else empty)
        else:
            xc, wc = _cpu_conv_args(x, w_c)
            yt = F.conv2d(xc.float(), wc.float(), stride=stride, padding=padding)
            y = yt.to(x.dtype).permute(0, 2, 3, 1).contiguous()
            yf = y.float()
            s = yf.sum(dim=(0, 1, 2))
            ss = (yf * yf).sum(dim=(0, 1, 2))
        ctx.mark_non_differentiable(s, ss)
        return y, s, ss

    @staticmethod
    def backward(ctx, dy, _ds, _dss):
        if dy is None:  # only the (never-used) stats outputs were consumed
            return None, None, None, None
        return Conv2dNHWC.backward(ctx, dy)


def conv2d_with_stats(x, weight, stride=1, padding=0):
    return Conv2dNHWCStats.apply(x, weight, stride, padding)


class BatchNormAct(Function):
    """Fused BatchNorm (+ optional residual add + optional ReLU) over NHWC.

    Three stat modes:
    * batch stats (training)          — per-channel mean/var over N*H*W
    * synced batch stats (SyncBN)     — partial sums all-reduced over `pg`
      (reference: SyncBatchNorm conversion at strategy.py:292)
    * frozen running stats            — the reference's net.eval()-while-
      training semantics (strategy.py:366-367): grads flow, stats frozen.

    y = act(bn(x) + residual); residual may be None.
    """

    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, use_batch_stats,
                momentum, eps, relu, residual, pg, pre_sums=None,
                fuse_backward=False):
        n_local = x.numel() // x.shape[-1]
        sync = (pg is not None and dist.is_initialized()
                and dist.get_world_size(pg) > 1)
        if use_batch_stats:
            if pre_sums is not None:
                # sums produced in the preceding conv's epilogue
                s, ss = pre_sums
                if sync:
                    count = torch.tensor([float(n_local)], device=x.device)
                    packed = torch.cat([s, ss, count])
                    dist.all_reduce(packed, group=pg)
                    s, ss = packed[:len(s)], packed[len(s):2 * len(s)]
                    n = float(packed[-1].item())
                    mean = s / n
                    var = (ss / n - mean * mean).clamp_min_(0)
                    invstd = (var + eps).rsqrt()
                    with torch.no_grad():
                        if running_mean is not None:
                            unbiased = var * (n / max(n - 1, 1))
                            running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                            running_var.mul_(1 - momentum).add_(unbiased,
                                                                alpha=momentum)
                elif _native_ok(x) and running_mean is not None:
                    from .extension import require_extension as _re
                    with torch.no_grad():
                        mean, invstd = _re().bn_finalize(
                            s, ss, running_mean, running_var, float(n_local),
                            momentum, eps, True)
                        bump_tick(running_mean)
                        bump_tick(running_var)
                    n = float(n_local)
                else:
                    n = float(n_local)
                    mean = s / n
                    var = (ss / n - mean * mean).clamp_min_(0)
                    invstd = (var + eps).rsqrt()
                    with torch.no_grad():
                        if running_mean is not None:
                            unbiased = var * (n / max(n - 1, 1))
                            running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                            running_var.mul_(1 - momentum).add_(unbiased,
                                                                alpha=momentum)
            elif _native_ok(x) and not sync:
                # fused path: partial sums -> mean/invstd + running update in
                # two kernels, no small ATen ops, no host sync
                ext = require_extension()
                with torch.no_grad():
                    mean, invstd = ext.bn_stats_finalize(
                        x, running_mean, running_var, momentum, eps,
                        running_mean is not None)
                    bump_tick(running_mean)
                    bump_tick(running_var)
                n = float(n_local)
            else:
                if _native_ok(x):
                    ext = require_extension()
                    s, ss = ext.bn_stats(x)  # fp32 per-channel sums
                else:
                    xf = x.float()
                    s = xf.sum(dim=(0, 1, 2))
                    ss = (xf * xf).sum(dim=(0, 1, 2))
                if sync:
                    count = torch.tensor([float(n_local)], device=x.device)
                    packed = torch.cat([s, ss, count])
                    dist.all_reduce(packed, group=pg)
                    s, ss = packed[:len(s)], packed[len(s):2 * len(s)]
                    n = float(packed[-1].item())
                else:
                    n = float(n_local)
                mean = s / n
                var = (ss / n - mean * mean).clamp_min_(0)
                invstd = (var + eps).rsqrt()
                with torch.no_grad():
                    if running_mean is not None:
                        unbiased = var * (n / max(n - 1, 1))
                        running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                        running_var.mul_(1 - momentum).add_(unbiased, alpha=momentum)
        else:
            mean = running_mean.to(torch.float32)
            invstd = (running_var.to(torch.float32) + eps).rsqrt()
            n = float(n_local)

        if _native_ok(x):
            ext = require_extension()
            # want_mask: the backward re-streams a 1-bit relu mask instead of
            # y. (grad mode is always off inside Function.forward, so gate on
            # the inputs' requires_grad flags only.)
            need_grad = (x.requires_grad or weight.requires_grad or
                         (residual is not None and residual.requires_grad))
            y, relu_mask = ext.bn_norm_fwd(
                x, mean, invstd, weight, bias, relu,
                residual if residual is not None else x.new_empty(0), need_grad)
        else:
            xf = x.float()
            y = (xf - mean) * invstd * weight + bias
            if residual is not None:
                y = y + residual.float()
            if relu:
                y = F.relu(y)
            y = y.to(x.dtype)
            relu_mask = None
        # GPU path saves the bit mask (backward never re-reads y); CPU path
        # keeps y for its fp32 reference masks
        ctx.save_for_backward(x, relu_mask if _native_ok(x) else y, weight, mean,
                              invstd)
        ctx.use_batch_stats = use_batch_stats
        ctx.relu = relu
        ctx.has_residual = residual is not None
        ctx.res_ref = residual if (_RESBACK_ON and residual is not None
                                   and _native_ok(x)) else None
        ctx.pg = pg
        ctx.n = n
        if (fuse_backward and relu and residual is None and relu_mask is not None
                and relu_mask.numel() > 0 and _native_ok(x)):
            # single-consumer side-channel: the downstream conv's bwd-data
            # epilogue masks dy and pre-reduces (sum dy~, sum dy~*xhat), so
            # the backward below can skip bn_bwd_reduce entirely
            try:
                y._al_bnback = (relu_mask, x, mean, invstd)
            except Exception:
                pass
        return y

    @staticmethod
    def backward(ctx, dy):
        x, mask_or_y, weight, mean, invstd = ctx.saved_tensors
        y = None if _native_ok(x) else mask_or_y
        dy = dy.contiguous()
        premasked = False
        if _native_ok(x):
            ext = require_extension()
            pre = getattr(dy, "_al_bnsums", None)
            if (pre is not None and not ctx.has_residual and ctx.relu
                    and pre[0].data_ptr() == mask_or_y.data_ptr()
                    and dy._version == pre[3]):
                # sums were computed in the producing conv's bwd-data
                # epilogue, and dy arrived PRE-MASKED — skip the reduce pass
                sum_dy, sum_dy_xhat = pre[1], pre[2]
                premasked = True
                if os.environ.get("AL_BNBACK_DEBUG") == "1":
                    rs, rsx = ext.bn_bwd_reduce(dy, x, mask_or_y, mean,
                                                invstd, False)
                    e1 = ((pre[1] - rs).norm() / rs.norm().clamp_min(1e-6)).item()
                    e2 = ((pre[2] - rsx).norm() / rsx.norm().clamp_min(1e-6)).item()
                    print(f"[bnback] C={x.shape[-1]} rows={x.numel()//x.shape[-1]}"
                          f" sum_err={e1:.2e} sumx_err={e2:.2e}", flush=True)
            else:
                # reduce pass: per-channel sums of dy~, dy~*xhat (dy~ = mask*dy)
                sum_dy, sum_dy_xhat = ext.bn_bwd_reduce(dy, x, mask_or_y, mean,
                                                        invstd, ctx.relu)
        else:
            dyf = dy.float()
            if ctx.relu:
                dyf = dyf * (y > 0).float()
            xhat = (x.float() - mean) * invstd
            sum_dy = dyf.sum(dim=(0, 1, 2))
            sum_dy_xhat = (dyf * xhat).sum(dim=(0, 1, 2))

        if ctx.use_batch_stats and ctx.pg is not None and dist.is_initialized() \
                and dist.get_world_size(ctx.pg) > 1:
            dgamma = sum_dy_xhat.clone()
            dbeta = sum_dy.clone()
            packed = torch.cat([sum_dy, sum_dy_xhat])
            dist.all_reduce(packed, group=ctx.pg)
            sum_dy, sum_dy_xhat = packed[:len(sum_dy)], packed[len(sum_dy):]
        else:
            # no aliasing hazard without the all-reduce: the sums ARE the grads
            dgamma = sum_dy_xhat
            dbeta = sum_dy
        # ctx.n is already the GLOBAL element count when stats were synced
        # (the forward all-reduced the counts), else the local count.
        n_global = ctx.n

        dres = None
        if _native_ok(x):
            ext = require_extension()
            dx, dres_t = ext.bn_bwd(dy, x, mask_or_y, mean, invstd, weight, sum_dy,
                                    sum_dy_xhat, n_global, ctx.use_batch_stats,
                                    ctx.relu and not premasked, ctx.has_residual)
            if ctx.has_residual:
                dres = dres_t
        else:
            dyf = dy.float()
            if ctx.relu:
                dyf = dyf * (y > 0).float()
            if ctx.has_residual:
                dres = dyf.to(x.dtype)
            g = weight * invstd
            if ctx.use_batch_stats:
                xhat = (x.float() - mean) * invstd
                dx = g * (dyf - sum_dy / n_global - xhat * (sum_dy_xhat / n_global))
            else:
                dx = g * dyf
            dx = dx.to(x.dtype)
        if (dres is not None and getattr(ctx, "res_ref", None) is not None
                and getattr(ctx.res_ref, "_al_res_consumer", False)):
            # hand dres to the conv that shares this tensor: its bwd-data
            # epilogue accumulates it, and autograd skips the fan-in add
            try:
                ctx.res_ref._al_dres = dres
                dres = None
            except Exception:
                pass
        return (dx, dgamma, dbeta, None, None, None, None, None, None, dres,
                None, None, None)


def batch_norm_act(x, weight, bias, running_mean, running_var, use_batch_stats,
                   momentum=0.1, eps=1e-5, relu=True, residual=None, pg=None,
                   pre_sums=None, fuse_backward=False):
    return BatchNormAct.apply(x, weight, bias, running_mean, running_var,
                              use_batch_stats, momentum, eps, relu, residual, pg,
                              pre_sums, fuse_backward)


class MaxPool2dNHWC(Function):
    @staticmethod
    def forward(ctx, x, kernel, stride, padding):
        ctx.native = _native_ok(x)
        if ctx.native:
            ext = require_extension()
            y, idx = ext.maxpool2d_fwd(x, kernel, stride, padding)
        else:
            xc = x.permute(0, 3, 1, 2).float()
            y, idx = F.max_pool2d(xc, kernel, stride, padding, return_indices=True)
            y = y.to(x.dtype).permute(0, 2, 3, 1).contiguous()
        ctx.save_for_backward(idx)
        ctx.x_shape = x.shape
        ctx.params = (kernel, stride, padding)
        ctx.x_dtype = x.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        kernel, stride, padding = ctx.params
        dy = dy.contiguous()
        if ctx.native:
            ext = require_extension()
            dx = ext.maxpool2d_bwd(dy, idx, ctx.x_shape[1], ctx.x_shape[2],
                                   kernel, stride, padding)
        else:
            dyc = dy.permute(0, 3, 1, 2).float()
            n, h, w, c = ctx.x_shape
            # scatter-ADD (max_unpool2d overwrites duplicate indices; gradients
            # must accumulate when windows share an argmax)
            dx_flat = torch.zeros(n, c, h * w, dtype=dyc.dtype)
            dx_flat.scatter_add_(2, idx.reshape(n, c, -1), dyc.reshape(n, c, -1))
            dx = dx_flat.reshape(n, c, h, w).to(ctx.x_dtype)
            dx = dx.permute(0, 2, 3, 1).contiguous()
        return dx, None, None, None


def max_pool2d(x, kernel=3, stride=2, padding=1):
    return MaxPool2dNHWC.apply(x, kernel, stride, padding)


class GlobalAvgPoolNHWC(Function):
    """(N,H,W,C) -> (N,C) mean over H,W (ResNet head)."""

    @staticmethod
    def forward(ctx, x):
        ctx.x_shape = x.shape
        ctx.x_dtype = x.dtype
        if _native_ok(x):
            ext = require_extension()
            return ext.global_avg_pool(x)
        return x.float().mean(dim=(1, 2)).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        n, h, w, c = ctx.x_shape
        scale = 1.0 / (h * w)
        dx = (dy.float() * scale)[:, None, None, :].expand(n, h, w, c)
        return dx.to(ctx.x_dtype).contiguous()


def global_avg_pool(x):
    return GlobalAvgPoolNHWC.apply(x)
