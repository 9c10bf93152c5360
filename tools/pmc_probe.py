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
torch.cuda.synchronize()
print("pmc probe done")
