#!/usr/bin/env python3
"""BN element-pass bandwidth A/B: structural variants of the bwd triad
(read dy + x + mask, write dx) vs a pure-streaming roofline, at the B=256
ResNet-50 shapes. Run on a GPU box:
  python tools/bn_ab.py [--iters 100] [--blocks 4096]
"""

import argparse
import ctypes
import os
import subprocess
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

HERE = os.path.dirname(os.path.abspath(__file__))
SO = os.path.join(HERE, "bn_ab.so")
SRC = os.path.join(HERE, "bn_ab.hip")


def build():
    if (os.path.exists(SO)
            and os.path.getmtime(SO) >= os.path.getmtime(SRC)):
        return
    subprocess.run(["hipcc", "--offload-arch=gfx950", "-O3", "-shared", "-fPIC",
                    "-o", SO, SRC], check=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=100)
    ap.add_argument("--blocks", type=int, default=4096,
                    help="row-block cap (production uses 4096)")
    args = ap.parse_args()
    build()
    lib = ctypes.CDLL(SO)
    lib.run_bwd.argtypes = [ctypes.c_int] + [ctypes.c_void_p] * 3 + [
        ctypes.c_void_p, ctypes.c_float, ctypes.c_void_p, ctypes.c_long,
        ctypes.c_int, ctypes.c_int]
    lib.run_triad.argtypes = [ctypes.c_void_p] * 3 + [ctypes.c_long, ctypes.c_int]
    assert torch.cuda.is_available()
    dev = "cuda"

    B = 256
    shapes = [(B * 112 * 112, 64), (B * 56 * 56, 64), (B * 56 * 56, 256),
              (B * 28 * 28, 512), (B * 14 * 14, 1024), (B * 7 * 7, 2048)]

    def timeit(fn, iters):
        for _ in range(10):
            fn()
        lib.dev_sync()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        lib.dev_sync()
        return (time.perf_counter() - t0) / iters

    # roofline first: 3-stream triad at ~400 MB/stream
    n4 = 100 * 1024 * 1024 // 16
    a = torch.randn(n4 * 4, device=dev)
    b = torch.randn(n4 * 4, device=dev)
    c = torch.empty(n4 * 4, device=dev)
    for blocks in (2048, 4096, 8192):
        t = timeit(lambda: lib.run_triad(a.data_ptr(), b.data_ptr(), c.data_ptr(),
                                         n4, blocks), args.iters)
        bw = 3 * n4 * 16 / t / 1e12
        print(f"triad fp32x4 blocks={blocks:5d}  {t*1e3:7.3f} ms  {bw:6.2f} TB/s")

    for rows, C in shapes:
        dy = torch.randn(rows, C, device=dev).to(torch.bfloat16)
        x = torch.randn_like(dy)
        mask = torch.randint(0, 256, (rows, C // 8), device=dev, dtype=torch.uint8)
        par = torch.randn(3 * C, device=dev)
        dx = torch.empty_like(dy)
        nbytes = rows * C * 2 * 3 + rows * (C // 8)
        ref = None
        for v in range(5):
            dx.zero_()
            t = timeit(lambda: lib.run_bwd(v, dy.data_ptr(), x.data_ptr(),
                                           mask.data_ptr(), par.data_ptr(),
                                           1.0 / rows, dx.data_ptr(), rows, C,
                                           args.blocks), args.iters)
            if ref is None:
                ref = dx.clone().float()
                ok = "ref"
            else:
                ok = "OK" if torch.equal(dx.float(), ref) else "MISMATCH"
            bw = nbytes / t / 1e12
            print(f"bwd v{v} rows={rows:9d} C={C:4d}  {t*1e3:7.3f} ms  "
                  f"{bw:6.2f} TB/s  {ok}")
        print()


if __name__ == "__main__":
    main()
