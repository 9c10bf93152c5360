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
for _ in range(2):
    y = ext.conv2d_fwd(x, w, 1, 1)
    dw = ext.conv2d_bwd_weight(dy, x, 3, 3, 1, 1)
    s, ss = ext.bn_stats(x)
torch.cuda.synchronize()
print("pmc probe done")
