#!/usr/bin/env python3
"""Audit the remaining per-step fill/zero launches in the eager training step
(the B=256 trace shows ~166 FillFunctor calls/step after the arenas): profile
one steady-state step with stacks and group every aten::zero_/fill_/zeros by
call site. Run on a GPU box: python tools/fill_audit.py
"""

import collections
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("AL_TRAIN_GRAPH", "0")

import torch

from active_learning_amd.models.ssl_resnet import ResNetSimCLR
from active_learning_amd.ops.loss import cross_entropy
from active_learning_amd.ops.optim import FusedSGD


def main():
    assert torch.cuda.is_available()
    dev = "cuda"
    torch.manual_seed(1234)
    net = ResNetSimCLR("resnet50", num_classes=1000).to(dev)
    opt = FusedSGD(net.parameters(), lr=0.1, momentum=0.9, weight_decay=1e-4)
    x = torch.randn(256, 3, 224, 224, device=dev)
    y = torch.randint(0, 1000, (256,), device=dev)

    def step():
        opt.zero_grad(set_to_none=True)
        loss = cross_entropy(net(x), y)
        loss.backward()
        opt.step()

    for _ in range(4):
        step()
    torch.cuda.synchronize()

    from torch.profiler import ProfilerActivity, profile
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=True) as prof:
        step()
        torch.cuda.synchronize()
    shapes = collections.Counter()
    for e in prof.events():
        if e.name in ("aten::zero_", "aten::zeros", "aten::fill_"):
            shapes[f"{e.name} {e.input_shapes}"] += 1
    print(f"fill-ish ops in 1 step: {sum(shapes.values())}")
    for s, n in shapes.most_common(25):
        print(f"{n:5d}  {s}")

    # copies and any high-count ATen ops (launch-bound suspects)
    copies = collections.Counter()
    counts = collections.Counter()
    for e in prof.events():
        if e.name.startswith("aten::"):
            counts[e.name] += 1
            if e.name in ("aten::copy_", "aten::cat", "aten::clone",
                          "aten::contiguous", "aten::to", "aten::_to_copy"):
                copies[f"{e.name} {e.input_shapes}"] += 1
    print("\nhigh-count aten ops (>=40/step):")
    for nm, n in counts.most_common(25):
        if n >= 40:
            print(f"{n:5d}  {nm}")
    print("\ncopy-ish ops by shape:")
    for s, n in copies.most_common(20):
        print(f"{n:5d}  {s}")

    # where the copy/cast GPU time actually goes
    ka = prof.key_averages()
    print("\naten op CUDA totals (copy/cast/clone/transpose suspects):")
    for row in ka:
        if row.key in ("aten::copy_", "aten::to", "aten::_to_copy",
                       "aten::clone", "aten::contiguous", "aten::permute",
                       "aten::cat") or "copyBuffer" in row.key \
                or "elementwise" in row.key or "Fill" in row.key:
            cuda_us = getattr(row, "self_device_time_total",
                              getattr(row, "self_cuda_time_total", 0))
            print(f"{row.count:6d}  {cuda_us/1e3:8.3f} ms  {row.key[:90]}")


if __name__ == "__main__":
    main()
