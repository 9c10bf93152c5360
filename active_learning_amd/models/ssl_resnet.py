"""SSLResNet wrapper: encoder + separate linear head.

Reference: src/models/resnet_simclr.py — fc replaced by Identity + separate
self.linear (:20-22); CIFAR stem surgery when num_classes == 10 (:17-18);
freeze_feature detaches the embedding (:36-37); specify_input_layer=
'finalembed' runs only the head on a given embedding (:31-33, used by the
MASE boundary sanity check, mase_sampler.py:85-90).
"""

import torch
import torch.nn as nn

from ..ops.linear import NativeLinear
from .resnet import resnet18_encoder, resnet50_encoder

_ENCODERS = {"resnet18": resnet18_encoder, "resnet50": resnet50_encoder}


class ResNetSimCLR(nn.Module):
    def __init__(self, base_model, num_classes=10):
        super().__init__()
        if base_model not in _ENCODERS:
            raise ValueError(f"Unknown base model {base_model!r}")
        cifar_stem = num_classes == 10  # parity: resnet_simclr.py:17-18
        self.encoder = _ENCODERS[base_model](cifar_stem=cifar_stem)
        self.dim_mlp = self.encoder.embed_dim
        self.num_classes = num_classes
        # state_dict-compatible nn.Linear running first-party MFMA GEMMs on
        # CUDA fp32 (ops/linear.py; reference: resnet_simclr.py:22)
        self.linear = NativeLinear(self.dim_mlp, num_classes)
        self.freeze_feature = False

    def forward(self, x, return_features=False, specify_input_layer=None):
        if specify_input_layer:
            assert specify_input_layer == "finalembed"
            return self.linear(x.float())
        intermediate = self.encoder(x)
        if self.freeze_feature:
            intermediate = intermediate.detach()
        out = self.linear(intermediate)
        if return_features:
            return out, intermediate
        return out
