#!/usr/bin/env python3
"""A/B wgrad v2 (row-major staging + ds_read_b64_tr_b16 fragments) against v1
(register-transpose staging). Route switch is read once per process:
  python tools/ab_wgrad.py            # v2 (default)
  AL_WGRAD_V2=0 python tools/ab_wgrad.py
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


# R50 @ B=256 wgrad shapes: (name, N, Hin, C, K, R, stride)
SHAPES = [
    ("l1.conv2", 256, 56, 64, 64, 3, 1),
    ("l2.conv2", 256, 28, 128, 128, 3, 1),
    ("l2.conv3", 256, 28, 128, 512, 1, 1),
    ("l3.conv1", 256, 14, 1024, 256, 1, 1),
    ("l3.conv2", 256, 14, 256, 256, 3, 1),
    ("l4.conv2", 256, 7, 512, 512, 3, 1),
    ("l4.conv3", 256, 7, 512, 2048, 1, 1),
]


def main():
    ext = require_extension()
    tag = "v1(xpose)" if os.environ.get("AL_WGRAD_V2") == "0" else "v2(tr_b16)"
    print(f"== wgrad {tag} ==")
    tot = 0.0
    for name, n, hin, c, k, r, stride in SHAPES:
        pad = r // 2
        hout = (hin + 2 * pad - r) // stride + 1
        x = torch.randn(n, hin, hin, c, device="cuda").to(torch.bfloat16)
        dy = torch.randn(n, hout, hout, k, device="cuda").to(torch.bfloat16)
        flops = 2.0 * n * hout * hout * c * k * r * r
        t = timeit(lambda: ext.conv2d_bwd_weight(dy, x, r, r, stride, pad))
        tot += t
        print(f"{name:10s} {t*1e3:7.3f} ms {flops/t/1e12:7.1f} TF")
    print(f"TOTAL {tot*1e3:.3f} ms")


if __name__ == "__main__":
    main()
