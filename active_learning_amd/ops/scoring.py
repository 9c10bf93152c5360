"""Query-scoring math: fused softmax scores, MASE margins, BADGE Gram
factorization, pairwise distances and the greedy k-center loop.

Reference counterparts:
  confidence/margin: softmax + top-k over logits (confidence_sampler.py:27-36,
    margin_sampler.py:29-39) — here a single fused kernel pass on GPU.
  MASE (mase_sampler.py:59-79): the reference materializes weight_delta
    (B,C,M) and epsilon (B,C,M). Closed form used here: the L2 distance from
    embedding e to the decision boundary between predicted class p and class c
    is |logit_p - logit_c| / ||w_p - w_c|| — needs only the logits and the
    C x C weight Gram matrix; no (B,C,M) intermediate exists.
  BADGE (badge_sampler.py:36-48): gradient embedding g_i = a_i (x) e_i with
    a_i = softmax(l_i) - onehot(argmax l_i). Since <a (x) e, a' (x) e'> =
    (a.a')(e.e'), the N x N Gram of gradient embeddings factorizes into the
    elementwise product of two small Grams; the (B, C*M) embedding is never
    materialized. The adaptive-avg-pool variant (badge_sampler.py:41-44)
    pools the outer product over a (16, 32) grid, which equals the outer
    product of the pooled vectors, so the factorization survives pooling.
  Coreset greedy k-center (coreset_sampler.py:66-105): the reference keeps the
    N x N matrix on GPU but runs the selection loop on the host with numpy.
    Here the distance matrix, the running min-distance vector and the
    argmax/sampling all stay on device (HBM-resident per BASELINE.json).
"""

import os

import torch
import torch.nn.functional as F

from .extension import require_extension


# --------------------------------------------------------------------------- #
# softmax-based uncertainty scores
# --------------------------------------------------------------------------- #

def softmax_scores(logits: torch.Tensor):
    """Return (top1_prob, margin = p1 - p2, entropy) per row, one fused pass
    on GPU."""
    if logits.is_cuda:
        ext = require_extension()
        out = ext.softmax_scores(logits)
        return out[0], out[1], out[2]
    p = F.softmax(logits.float(), dim=1)
    top2 = torch.topk(p, k=min(2, p.shape[1]), dim=1).values
    ent = -(p * torch.log(p.clamp_min(1e-12))).sum(dim=1)
    margin = top2[:, 0] - (top2[:, 1] if top2.shape[1] > 1 else torch.zeros_like(top2[:, 0]))
    return top2[:, 0], margin, ent


def confidence_scores(logits):
    return softmax_scores(logits)[0]


def margin_scores(logits):
    return softmax_scores(logits)[1]


def entropy_scores(logits):
    return softmax_scores(logits)[2]


# --------------------------------------------------------------------------- #
# MASE margins
# --------------------------------------------------------------------------- #

def mase_margins(logits: torch.Tensor, weight: torch.Tensor):
    """Distances to pairwise decision boundaries in feature space.

    radius[i, c] = |logit_i[pred_i] - logit_i[c]| / ||w_pred_i - w_c||,
    radius[i, pred_i] = +inf (reference gets inf from 0/0 -> nan -> inf,
    mase_sampler.py:76-78). Returns (min_margins (B,), radius (B,C), pred (B,)).

    Bias contributes through the logits (logit = e.w + b), identical to the
    reference's lam formula with bias_delta folded in (mase_sampler.py:66-79).
    """
    logits = logits.float()
    w = weight.float()
    pred = logits.argmax(dim=1)
    gram = w @ w.t()  # (C, C)
    sq_norms = gram.diagonal()
    # ||w_p - w_c||^2 for every (p, c)
    d2 = (sq_norms[:, None] + sq_norms[None, :] - 2 * gram).clamp_min_(0)
    denom = d2[pred].sqrt()  # (B, C)
    num = (logits.gather(1, pred[:, None]) - logits).abs()
    radius = num / denom
    radius[torch.arange(len(pred), device=pred.device), pred] = float("inf")
    radius = torch.nan_to_num(radius, nan=float("inf"))
    min_margins = radius.min(dim=1).values
    return min_margins, radius, pred


# --------------------------------------------------------------------------- #
# pairwise distances / BADGE Gram factorization
# --------------------------------------------------------------------------- #

def pairwise_sqdist(features: torch.Tensor, chunk=8192) -> torch.Tensor:
    """||x_i - x_j||^2 as norms + dot products (coreset_sampler.py:59-64);
    the N x N fp32 output is materialized (N=130k -> 68 GB, resident in the
    288 GB HBM).

    GPU default: the bf16-MFMA pairwise kernel (linear.hip) — dots on
    mfma_f32_16x16x32_bf16 with fp32 accumulation and fp32 norms/output.
    The embeddings are bf16-computed to begin with (PARITY.md);
    AL_PAIRWISE_BF16=0 falls back to the exact-fp32 rocBLAS composition."""
    f = features.float()
    n = f.shape[0]
    if (f.is_cuda and f.shape[1] % 64 == 0
            and os.environ.get("AL_PAIRWISE_BF16", "1") == "1"):
        from .extension import require_extension
        fb = f.to(torch.bfloat16).contiguous()
        # norms from the SAME rounded values the dots use: the result is a
        # true distance matrix of the bf16-rounded points (near-zero diag)
        sqb = (fb.float() ** 2).sum(dim=1).contiguous()
        return require_extension().pairwise_sqdist_dev(fb, sqb)
    sq = (f * f).sum(dim=1)
    out = torch.empty((n, n), dtype=torch.float32, device=f.device)
    for i0 in range(0, n, chunk):
        i1 = min(i0 + chunk, n)
        dp = f[i0:i1] @ f.t()
        out[i0:i1] = sq[i0:i1, None] + sq[None, :] - 2 * dp
    return out


def badge_pairwise_sqdist(a_vec: torch.Tensor, e_vec: torch.Tensor,
                          chunk=8192) -> torch.Tensor:
    """N x N squared distances between gradient embeddings g_i = a_i (x) e_i
    without materializing them: <g_i, g_j> = (a_i.a_j)(e_i.e_j).

    Row-chunked so only ONE N x N fp32 matrix is resident (68 GB at N=130k);
    the two Gram factors exist chunk-at-a-time."""
    a = a_vec.float().contiguous()
    e = e_vec.float().contiguous()
    n = a.shape[0]
    d = ((a * a).sum(dim=1) * (e * e).sum(dim=1)).contiguous()  # <g_i, g_i>
    if (a.is_cuda and a.shape[1] + e.shape[1] <= 128
            and os.environ.get("AL_BADGE_GRAM_DEV", "1") == "1"):
        # fused kernel for the POOLED factors (16 x 32, the reference's
        # ImageNet-scale BADGE): both rank-K grams on f32 MFMA + the distance
        # combine, ONE write of the N x N output. Wide unpooled factors
        # (plain CIFAR BADGE: K = C + M) stay on rocBLAS, which wins at
        # large contraction depth.
        from .extension import require_extension
        return require_extension().badge_gram(a, e, d)
    out = torch.empty((n, n), dtype=torch.float32, device=a.device)
    for i0 in range(0, n, chunk):
        i1 = min(i0 + chunk, n)
        g = (a[i0:i1] @ a.t()) * (e[i0:i1] @ e.t())
        g.mul_(-2).add_(d[i0:i1, None]).add_(d[None, :])
        out[i0:i1] = g
    return out


def badge_vectors(logits: torch.Tensor, embedding: torch.Tensor, pool=None):
    """(a_i, e_i) pair defining the BADGE gradient embedding; with pool=(Ph,Pw)
    both vectors are adaptive-avg-pooled (outer product of pooled vectors ==
    pooled outer product, badge_sampler.py:41-44)."""
    p = F.softmax(logits.float(), dim=1)
    yhat = logits.argmax(dim=1)
    a = p.clone()
    a[torch.arange(len(yhat), device=logits.device), yhat] -= 1.0
    e = embedding.float()
    if pool is not None:
        ph, pw = pool
        a = F.adaptive_avg_pool1d(a[:, None, :], ph)[:, 0, :]
        e = F.adaptive_avg_pool1d(e[:, None, :], pw)[:, 0, :]
    return a, e


# --------------------------------------------------------------------------- #
# greedy k-center (device-resident)
# --------------------------------------------------------------------------- #

def kcenter_greedy(dist: torch.Tensor, labeled: torch.Tensor, budget: int,
                   randomize=False, generator=None):
    """Greedy farthest-point selection over a pairwise-distance matrix.

    dist: (N, N) on device; labeled: (N,) bool on device; returns list of N-space
    indices. Semantics mirror coreset_sampler.py:66-105, including:
      * no labeled points yet: deterministic -> argmin of row-max
        (coreset_sampler.py:100), randomized -> uniform choice (:97);
      * randomize=True: probability proportional to clamp(min_dist, 0) with
        already-selected masked to 0 (k-means++ seeding, :81-92).
    Everything stays on device; per-iteration work is one column min-update
    plus one argmax/multinomial.
    """
    n = dist.shape[0]
    dev = dist.device
    labeled = labeled.clone()
    # fully device-side selection: the chosen index never round-trips to the
    # host inside the loop (one sync per QUERY, not per iteration — the
    # reference syncs every iteration, coreset_sampler.py:82-104)
    sel_buf = torch.empty(budget, dtype=torch.int64, device=dev)

    if labeled.any():
        min_dist = _masked_col_min(dist, labeled)
        start = 0
    else:
        if randomize:
            j = torch.randint(n, (1,), device=dev)[0]
        else:
            j = dist.max(dim=1).values.argmin()
        sel_buf[0] = j
        labeled[j] = True
        min_dist = dist[j].clone()  # symmetric: row == column
        start = 1

    if (dist.is_cuda and budget > start
            and os.environ.get("AL_KCENTER_DEV", "1") == "1"):
        # persistent cooperative kernel: the ENTIRE greedy loop is one launch
        # (fused min-update + argmax / inverse-CDF sample per iteration,
        # grid.sync between) instead of 3+ launches per iteration
        from .extension import require_extension
        ext = require_extension()
        iters = budget - start
        lab_u8 = labeled.to(torch.uint8)
        randu = (torch.rand(iters, device=dev) if randomize
                 else torch.empty(0, device=dev))
        md = min_dist.contiguous()
        sel_tail = sel_buf[start:].contiguous()
        # j_init < 0: min_dist is already up to date (labeled-init path);
        # j_init >= 0 would re-apply a row — the caller already applied it,
        # so always pass -1 and keep min_dist authoritative
        rc = ext.kcenter_greedy_dev(dist, md, lab_u8, sel_tail, randu, -1,
                                    bool(randomize))
        if rc == 0:
            sel_buf[start:] = sel_tail
            return sel_buf.cpu().tolist()
        # cooperative launch unsupported: fall through to the torch loop

    for it in range(start, budget):
        if randomize:
            probs = min_dist.clamp_min(0.0)
            probs = torch.where(labeled, torch.zeros_like(probs), probs)
            total = probs.sum()
            # reference jitters on degenerate weights (coreset_sampler.py:85-92);
            # equivalent: uniform over unlabeled in that case, chosen
            # device-side so the loop stays sync-free
            uniform = (~labeled).float()
            bad = ~torch.isfinite(total) | (total <= 0)
            probs = torch.where(bad, uniform, probs)
            j = torch.multinomial(probs, 1)[0]
        else:
            scores = min_dist.masked_fill(labeled, float("-inf"))
            j = scores.argmax()
        sel_buf[it] = j
        labeled[j] = True
        torch.minimum(min_dist, dist[j], out=min_dist)
    return sel_buf.cpu().tolist()


def _masked_col_min(dist, labeled_mask, chunk=16384):
    """min over labeled columns, chunked (avoids a N x N_l gather copy)."""
    cols = labeled_mask.nonzero(as_tuple=True)[0]
    out = None
    for c0 in range(0, len(cols), chunk):
        part = dist[:, cols[c0:c0 + chunk]].min(dim=1).values
        out = part if out is None else torch.minimum(out, part)
    return out


# --------------------------------------------------------------------------- #
# class centroids (BalancingSampler)
# --------------------------------------------------------------------------- #

def class_centroids(embeddings: torch.Tensor, labels: torch.Tensor, num_classes: int):
    """Per-class mean embedding via scatter-add; classes with no samples get a
    zero centroid (balancing_sampler.py:87-96)."""
    e = embeddings.float()
    out = torch.zeros(num_classes, e.shape[1], device=e.device)
    out.index_add_(0, labels, e)
    counts = torch.bincount(labels, minlength=num_classes).float()
    return out / (counts[:, None] + 1e-5)


def sqdist_to_centers(x: torch.Tensor, centers: torch.Tensor):
    x = x.float()
    c = centers.float()
    return ((x * x).sum(1)[:, None] + (c * c).sum(1)[None, :] - 2 * x @ c.t())
