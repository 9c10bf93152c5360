"""Native NHWC ResNet-18/50 encoder on hand-written CDNA4 kernels.

Architecture parity with torchvision's resnet18/resnet50 v1.5 (what the
reference wraps, src/models/resnet_simclr.py:10-11) including the CIFAR stem
surgery: 3x3 s1 conv1 + no maxpool when num_classes == 10
(src/models/resnet_hacks.py:31-35, applied at resnet_simclr.py:17-18).

Differences by design (MI355X-first):
  * NHWC activations end to end, bf16 on GPU with fp32 master weights;
  * BN+ReLU (and the block-final BN+add+ReLU) are single fused ops;
  * the final fc is NOT part of the encoder (reference replaces it with
    Identity, resnet_simclr.py:20-21) — the encoder returns the embedding.

Module names (conv1/bn1/layer{1..4}.{i}.conv{j}/bn{j}/downsample) mirror
torchvision so SSL-checkpoint key surgery (required_key/replace_key filters,
ssp_finetuning.py:34-37) maps onto the same names.
"""

import torch
import torch.nn as nn

from ..ops import functional as AF
from ..ops.fused import conv_bn_act
from .layers import BatchNormAct2d, Conv2dNHWC, nchw_to_nhwc


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_planes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = Conv2dNHWC(in_planes, planes, 3, stride, 1)
        self.bn1 = BatchNormAct2d(planes, relu=True)
        self.conv2 = Conv2dNHWC(planes, planes, 3, 1, 1)
        self.bn2 = BatchNormAct2d(planes, relu=True)  # fused add+relu via residual arg
        self.downsample = downsample

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = conv_bn_act(self.conv1, self.bn1, x)
        out = conv_bn_act(self.conv2, self.bn2, out, residual=identity)
        return out


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_planes, planes, stride=1, downsample=None, v1=False):
        super().__init__()
        # v1.5 (torchvision default): stride on the 3x3; v1 (resnet_hacks.py:36-46
        # stride swap): stride on the first 1x1 instead
        s1, s2 = (stride, 1) if v1 else (1, stride)
        self.conv1 = Conv2dNHWC(in_planes, planes, 1, s1, 0)
        self.bn1 = BatchNormAct2d(planes, relu=True)
        self.conv2 = Conv2dNHWC(planes, planes, 3, s2, 1)
        self.bn2 = BatchNormAct2d(planes, relu=True)
        self.conv3 = Conv2dNHWC(planes, planes * self.expansion, 1, 1, 0)
        self.bn3 = BatchNormAct2d(planes * self.expansion, relu=True)
        self.downsample = downsample

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = conv_bn_act(self.conv1, self.bn1, x)
        out = conv_bn_act(self.conv2, self.bn2, out)
        out = conv_bn_act(self.conv3, self.bn3, out, residual=identity)
        return out


class Downsample(nn.Module):
    """1x1 stride-s conv + BN (no relu) on the shortcut path."""

    def __init__(self, in_planes, out_planes, stride):
        super().__init__()
        self.conv = Conv2dNHWC(in_planes, out_planes, 1, stride, 0)
        self.bn = BatchNormAct2d(out_planes, relu=False)

    def forward(self, x):
        return conv_bn_act(self.conv, self.bn, x)


class ResNetEncoder(nn.Module):
    def __init__(self, block, layers, cifar_stem=False, compute_dtype=torch.bfloat16,
                 v1=False):
        super().__init__()
        self.compute_dtype = compute_dtype
        self.cifar_stem = cifar_stem
        self.v1 = v1
        self.in_planes = 64
        if cifar_stem:
            self.conv1 = Conv2dNHWC(3, 64, 3, 1, 1)
        else:
            self.conv1 = Conv2dNHWC(3, 64, 7, 2, 3)
        self.bn1 = BatchNormAct2d(64, relu=True)
        self.layer1 = self._make_layer(block, 64, layers[0], 1)
        self.layer2 = self._make_layer(block, 128, layers[1], 2)
        self.layer3 = self._make_layer(block, 256, layers[2], 2)
        self.layer4 = self._make_layer(block, 512, layers[3], 2)
        self.embed_dim = 512 * block.expansion

    def _make_layer(self, block, planes, num_blocks, stride):
        downsample = None
        if stride != 1 or self.in_planes != planes * block.expansion:
            downsample = Downsample(self.in_planes, planes * block.expansion, stride)
        kw = {"v1": self.v1} if block is Bottleneck else {}
        blocks = [block(self.in_planes, planes, stride, downsample, **kw)]
        self.in_planes = planes * block.expansion
        for _ in range(1, num_blocks):
            blocks.append(block(self.in_planes, planes))
        return nn.Sequential(*blocks)

    def forward(self, x):
        # boundary: accept NCHW fp32 (reference dataloader contract), move to
        # NHWC + compute dtype for the kernel path
        if x.dim() != 4:
            raise ValueError("expected NCHW input")
        x = nchw_to_nhwc(x)
        if x.is_cuda and self.compute_dtype is not None:
            x = x.to(self.compute_dtype)
        x = conv_bn_act(self.conv1, self.bn1, x)
        if not self.cifar_stem:
            x = AF.max_pool2d(x, 3, 2, 1)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        emb = AF.global_avg_pool(x)
        return emb.float()


def resnet18_encoder(cifar_stem=False, **kw):
    return ResNetEncoder(BasicBlock, [2, 2, 2, 2], cifar_stem=cifar_stem, **kw)


def resnet50_encoder(cifar_stem=False, **kw):
    return ResNetEncoder(Bottleneck, [3, 4, 6, 3], cifar_stem=cifar_stem, **kw)


class NormedLinear(nn.Module):
    """Cosine-normalized linear head.

    Parity with src/models/resnet_hacks.py:50-59 (defined there but unused
    anywhere in the reference; provided for capability completeness).
    """

    def __init__(self, in_features, out_features):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(in_features, out_features))
        with torch.no_grad():
            self.weight.uniform_(-1, 1).renorm_(2, 1, 1e-5).mul_(1e5)

    def forward(self, x):
        import torch.nn.functional as F
        return F.normalize(x.float(), dim=1) @ F.normalize(self.weight, dim=0)
