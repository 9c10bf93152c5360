#!/usr/bin/env python3
"""A/B the 8-phase 256^2 NT GEMM (1x1 s1 conv route) against the 128^2 igemm.
Run twice on a GPU box (the route switch is read once per process):
  python tools/ab_gemm256.py            # gemm256 enabled
  AL_DISABLE_GEMM256=1 python tools/ab_gemm256.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from active_learning_amd.ops.extension import require_extension


def timeit(fn, iters=30, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


# R50 @ B=256 stride-1 shapes: (name, N, HW, C_in, K_out, R)
SHAPES = [
    ("l2.conv3", 256, 28, 128, 512, 1),
    ("l3.conv1", 256, 14, 1024, 256, 1),
    ("l3.conv3", 256, 14, 256, 1024, 1),
    ("l4.conv3", 256, 7, 512, 2048, 1),
    ("l3.conv2", 256, 14, 256, 256, 3),   # gathered 256^2 (grid 196)
    ("l4.conv2", 256, 7, 512, 512, 3),    # grid 98 < 192: stays 128^2
    ("l2.conv1", 256, 28, 256, 128, 1),   # 256x128 tile (pure)
    ("l2.conv2", 256, 28, 128, 128, 3),   # 512x128 tall tile (gathered)
]


def main():
    ext = require_extension()
    tag = "OFF(128^2)" if os.environ.get("AL_DISABLE_GEMM256") == "1" else "ON(256^2)"
    print(f"== gemm256 {tag} ==")
    tot_f = tot_b = 0.0
    for name, n, hw, c, k, r in SHAPES:
        pad = r // 2
        x = torch.randn(n, hw, hw, c, device="cuda").to(torch.bfloat16)
        w = (torch.randn(k, r, r, c, device="cuda") * 0.05).to(torch.bfloat16)
        flops = 2.0 * n * hw * hw * c * k * r * r
        tf = timeit(lambda: ext.conv2d_fwd(x, w, 1, pad))
        dy = torch.randn(n, hw, hw, k, device="cuda").to(torch.bfloat16)
        wt = w.permute(3, 1, 2, 0).contiguous()
        tb = timeit(lambda: ext.conv2d_bwd_data(dy, wt, 1, pad, hw, hw))
        tot_f += tf
        tot_b += tb
        print(f"{name:10s} fwd {tf*1e3:7.3f} ms {flops/tf/1e12:7.1f} TF   "
              f"bwd {tb*1e3:7.3f} ms {flops/tb/1e12:7.1f} TF")
    print(f"TOTAL fwd {tot_f*1e3:.3f} ms  bwd {tot_b*1e3:.3f} ms")


if __name__ == "__main__":
    main()
