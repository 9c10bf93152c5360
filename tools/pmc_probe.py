#!/usr/bin/env python3
"""Minimal kernel set for PMC counter collection (keep dispatch count tiny)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from active_learning_amd.ops.extension import require_extension
ext = require_extension()
x = torch.randn(64, 28, 28, 128, device="cuda").to(torch.bfloat16)
w = (torch.randn(128, 3, 3, 128, device="cuda") * 0.05).to(torch.bfloat16)
dy = torch.randn(64, 28, 28, 128, device="cuda").to(torch.bfloat16)
# 256^2 8-phase shapes: pure 1x1 and gathered 3x3
x1 = torch.randn(64, 14, 14, 256, device="cuda").to(torch.bfloat16)
w1 = (torch.randn(1024, 1, 1, 256, device="cuda") * 0.05).to(torch.bfloat16)
x3 = torch.randn(64, 28, 28, 128, device="cuda").to(torch.bfloat16)
w3 = (torch.randn(512, 3, 3, 128, device="cuda") * 0.05).to(torch.bfloat16)
for _ in range(2):
    y = ext.conv2d_fwd(x, w, 1, 1)
    dw = ext.conv2d_bwd_weight(dy, x, 3, 3, 1, 1)   # wgrad2 (tr_b16)
    s, ss = ext.bn_stats(x)
    y1 = ext.conv2d_fwd(x1, w1, 1, 0)               # gemm256 MODE_PURE
    y3 = ext.conv2d_fwd(x3, w3, 1, 1)               # gemm256 MODE_FWD gather

# round-2 kernels -----------------------------------------------------------
# fp32 MFMA linear head (fwd/dx/dw)
xf = torch.randn(128, 2048, device="cuda")
wf = torch.randn(1000, 2048, device="cuda") * 0.05
bf = torch.randn(1000, device="cuda")
yl = ext.linear_fwd(xf, wf, bf)
dxl, dwl, dbl = ext.linear_bwd(torch.randn_like(yl), xf, wf, True, True, True)
# fused BADGE gram (pooled widths)
av = torch.randn(8192, 16, device="cuda")
ev = torch.randn(8192, 32, device="cuda")
dv = ((av * av).sum(1) * (ev * ev).sum(1)).contiguous()
g = ext.badge_gram(av, ev, dv)
# persistent k-center (one cooperative launch, 64 iterations)
n = 8192
dist = torch.rand(n, n, device="cuda")
dist = dist + dist.t()
md = dist[0].clone()
lab = torch.zeros(n, dtype=torch.uint8, device="cuda"); lab[0] = 1
sel = torch.empty(64, dtype=torch.int64, device="cuda")
ext.kcenter_greedy_dev(dist, md, lab, sel, torch.empty(0, device="cuda"), -1, False)
# multi-tensor SGD (one launch over a mixed param set)
ps = [torch.randn(s, device="cuda") for s in (100000, 32768, 1003)]
rows = []
states = [torch.zeros_like(p) for p in ps]
for p, m in zip(ps, states):
    off = 0
    while off < p.numel():
        rows.append((p.data_ptr(), p.data_ptr(), m.data_ptr(), 0, off,
                     min(32768, p.numel() - off)))
        off += 32768
table = torch.tensor(rows, dtype=torch.int64).cuda()
ext.sgd_step_multi(table, len(rows), 0.1, 0.9, 1e-4, False)
torch.cuda.synchronize()
print("pmc probe done")
