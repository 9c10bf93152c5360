#!/usr/bin/env python3
"""Per-kernel micro-benchmarks on MI355X: achieved GB/s / TFLOP/s for each
HIP kernel at representative ResNet-50 shapes. Run on a GPU box:
  python tools/bench_kernels.py [--iters 50]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from active_learning_amd.ops.extension import require_extension


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    ext = require_extension()
    dev = "cuda"
    results = []

    def report(name, sec, byte=0, flops=0):
        gbs = byte / sec / 1e9 if byte else 0
        tf = flops / sec / 1e12 if flops else 0
        results.append((name, sec * 1e3, gbs, tf))
        print(f"{name:42s} {sec*1e3:8.3f} ms  {gbs:8.1f} GB/s  {tf:8.1f} TF")

    # ---- BN stack at the two dominant R50 shapes -------------------------
    for (rows, C) in [(128 * 56 * 56, 256), (128 * 112 * 112, 64),
                      (128 * 14 * 14, 1024)]:
        x = torch.randn(rows, C, device=dev).to(torch.bfloat16).view(rows, 1, 1, C)
        dy = torch.randn_like(x)
        y = torch.randn_like(x)
        mk = torch.randint(0, 256, (rows, C // 8), device=dev,
                           dtype=torch.uint8)
        mean = torch.zeros(C, device=dev)
        invstd = torch.ones(C, device=dev)
        g = torch.ones(C, device=dev)
        b = torch.zeros(C, device=dev)
        e = x.new_empty(0)
        byte = rows * C * 2
        report(f"bn_stats     r={rows} C={C}", timeit(lambda: ext.bn_stats(x), args.iters),
               byte=byte)
        report(f"bn_norm_fwd  r={rows} C={C}",
               timeit(lambda: ext.bn_norm_fwd(x, mean, invstd, g, b, True, e, True),
                      args.iters), byte=2 * byte)
        report(f"bn_bwd_reduce r={rows} C={C}",
               timeit(lambda: ext.bn_bwd_reduce(dy, x, mk, mean, invstd, True),
                      args.iters), byte=3 * byte)
        report(f"bn_bwd       r={rows} C={C}",
               timeit(lambda: ext.bn_bwd(dy, x, mk, mean, invstd, g, mean, mean,
                                         float(rows), True, True, False),
                      args.iters), byte=3 * byte)

    # ---- conv shapes (R50 hot layers) ------------------------------------
    CONVS = [
        ("l1 3x3 C64->64 56sq", 128, 56, 64, 64, 3, 1),
        ("l1 1x1 C64->256 56sq", 128, 56, 64, 256, 1, 1),
        ("l2 3x3 C128 28sq", 128, 28, 128, 128, 3, 1),
        ("l3 3x3 C256 14sq", 128, 14, 256, 256, 3, 1),
        ("l4 3x3 C512 7sq", 128, 7, 512, 512, 3, 1),
        ("l4 1x1 C2048->512 7sq", 128, 7, 2048, 512, 1, 1),
        ("l2 3x3 s2 C128 56->28", 128, 56, 128, 128, 3, 2),
    ]
    for name, n, hw, c, k, r, stride in CONVS:
        pad = r // 2
        p = (hw + 2 * pad - r) // stride + 1
        x = torch.randn(n, hw, hw, c, device=dev).to(torch.bfloat16)
        w = (torch.randn(k, r, r, c, device=dev) * 0.05).to(torch.bfloat16)
        dy = torch.randn(n, p, p, k, device=dev).to(torch.bfloat16)
        wt = w.permute(3, 1, 2, 0).contiguous()
        flops = 2.0 * n * p * p * k * r * r * c
        report(f"conv_fwd  {name}", timeit(lambda: ext.conv2d_fwd(x, w, stride, pad),
                                           args.iters), flops=flops)
        report(f"conv_bwdd {name}",
               timeit(lambda: ext.conv2d_bwd_data(dy, wt, stride, pad, hw, hw),
                      args.iters), flops=flops)
        report(f"conv_wgrad {name}",
               timeit(lambda: ext.conv2d_bwd_weight(dy, x, r, r, stride, pad),
                      args.iters), flops=flops)

    # ---- big-GEMM ceiling probe via 1x1 conv ------------------------------
    for m, k_dim, n_out in [(8192, 2048, 2048), (16384, 4096, 4096)]:
        x = torch.randn(1, m, 1, k_dim, device=dev).to(torch.bfloat16)
        w = (torch.randn(n_out, 1, 1, k_dim, device=dev) * 0.02).to(torch.bfloat16)
        flops = 2.0 * m * k_dim * n_out
        report(f"gemm {m}x{n_out}x{k_dim}",
               timeit(lambda: ext.conv2d_fwd(x, w, 1, 0), args.iters), flops=flops)
        # rocBLAS comparison point
        a = torch.randn(m, k_dim, device=dev, dtype=torch.bfloat16)
        bm = torch.randn(k_dim, n_out, device=dev, dtype=torch.bfloat16)
        report(f"rocblas {m}x{n_out}x{k_dim}", timeit(lambda: a @ bm, args.iters),
               flops=flops)

    print("\nSUMMARY")
    for name, ms, gbs, tf in results:
        print(f"{name:42s} {ms:8.3f} ms {gbs:8.1f} GB/s {tf:8.1f} TF")


if __name__ == "__main__":
    main()
