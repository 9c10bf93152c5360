#!/usr/bin/env python3
"""Minimal repro for the R18/CIFAR hipGraph capture abort. Builds the same
net as tests/test_graph_gpu.py and steps it through GraphedTrainStep with
verbose stage prints so the aborting operation is identifiable."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from active_learning_amd.models import get_networks
from active_learning_amd.ops.graph import GraphedTrainStep
from active_learning_amd.ops.loss import CrossEntropyLoss
from active_learning_amd.ops.optim import FusedSGD


def main():
    dev = torch.device("cuda", 0)
    torch.manual_seed(3)
    net = get_networks("synthetic_cifar10", "SSLResNet18").cuda()
    opt = FusedSGD(net.parameters(), lr=0.05, momentum=0.9, weight_decay=1e-4)
    gs = GraphedTrainStep(net, opt, CrossEntropyLoss().to(dev), dev, warmup=2)
    net.train()
    torch.manual_seed(0)
    for i in range(6):
        x = torch.randn(16, 3, 16, 16)
        y = torch.randint(0, 10, (16,))
        print(f"step {i} (graph={'yes' if gs._graph is not None else 'no'})",
              flush=True)
        loss = gs.step(x, y)
        torch.cuda.synchronize()
        print(f"  loss={loss.item():.4f}", flush=True)
    print("OK", flush=True)


if __name__ == "__main__":
    main()
