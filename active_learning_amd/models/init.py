"""Per-round weight re-initialization.

Reference: src/models/utils.py:5-18 — Conv kaiming fan_out, BN (1, 0),
Linear normal sigma=1e-3 bias 0; applied every round via net.apply
(strategy.py:184). Adapted to the native layer types/layouts.
"""

import math

import torch
import torch.nn as nn

from .layers import BatchNormAct2d, Conv2dNHWC, ConvTranspose2dNHWC


@torch.no_grad()
def init_params(net):
    for m in net.modules():
        if isinstance(m, (Conv2dNHWC, ConvTranspose2dNHWC)):
            k = m.kernel_size
            fan_out = m.out_channels * k * k
            m.weight.normal_(0, math.sqrt(2.0 / fan_out))
            if getattr(m, "bias", None) is not None:
                m.bias.zero_()
        elif isinstance(m, nn.Conv2d):
            nn.init.kaiming_normal_(m.weight, mode="fan_out")
            if m.bias is not None:
                m.bias.zero_()
        elif isinstance(m, (BatchNormAct2d, nn.BatchNorm2d, nn.BatchNorm1d)):
            # reference init_params touches only BN weight/bias, NOT the
            # running stats (src/models/utils.py:10-13) — keep that behavior
            m.weight.fill_(1.0)
            m.bias.zero_()
        elif isinstance(m, nn.Linear):
            m.weight.normal_(0, 1e-3)
            if m.bias is not None:
                m.bias.zero_()
