"""VAE + discriminator for VAAL, on the native NHWC conv / conv-transpose ops.

Reference: src/query_strategies/vae.py (4x strided-conv encoder 128->1024ch
:26-35, fc_mu/fc_logvar :37-38, ConvTranspose decoder :39-51, seeded 64x64
crop :62-82, reparameterize :90-96) and src/query_strategies/
vaal_discriminator.py (z->512->512->1 MLP + Sigmoid).
"""

import math

import numpy as np
import torch
import torch.nn as nn

from .layers import (BatchNormAct2d, Conv2dNHWC, ConvTranspose2dNHWC, nchw_to_nhwc,
                     nhwc_to_nchw)

CROP_H = 64
CROP_W = 64


class VAE(nn.Module):
    def __init__(self, z_dim=32, nc=3, latent_scale=1.0):
        super().__init__()
        self.z_dim = z_dim
        self.nc = nc
        ls = int(latent_scale)
        self.ls = ls
        self.crop_seed = 0

        self.enc_conv1 = Conv2dNHWC(nc, 128, 4, 2, 1)
        self.enc_bn1 = BatchNormAct2d(128, relu=True)
        self.enc_conv2 = Conv2dNHWC(128, 256, 4, 2, 1)
        self.enc_bn2 = BatchNormAct2d(256, relu=True)
        self.enc_conv3 = Conv2dNHWC(256, 512, 4, 2, 1)
        self.enc_bn3 = BatchNormAct2d(512, relu=True)
        self.enc_conv4 = Conv2dNHWC(512, 1024, 4, 2, 1)
        self.enc_bn4 = BatchNormAct2d(1024, relu=True)

        feat = 1024 * 2 * 2 * ls * ls
        self.fc_mu = nn.Linear(feat, z_dim)
        self.fc_logvar = nn.Linear(feat, z_dim)
        self.dec_fc = nn.Linear(z_dim, 1024 * 4 * 4 * ls * ls)
        self.dec_conv1 = ConvTranspose2dNHWC(1024, 512, 4, 2, 1)
        self.dec_bn1 = BatchNormAct2d(512, relu=True)
        self.dec_conv2 = ConvTranspose2dNHWC(512, 256, 4, 2, 1)
        self.dec_bn2 = BatchNormAct2d(256, relu=True)
        self.dec_conv3 = ConvTranspose2dNHWC(256, 128, 4, 2, 1)
        self.dec_bn3 = BatchNormAct2d(128, relu=True)
        self.dec_conv4 = ConvTranspose2dNHWC(128, nc, 1, 1, 0, bias=True)
        self.weight_init()

    @torch.no_grad()
    def weight_init(self):
        for m in self.modules():
            if isinstance(m, (Conv2dNHWC, ConvTranspose2dNHWC)):
                fan_out = m.out_channels * m.kernel_size * m.kernel_size
                m.weight.normal_(0, math.sqrt(2.0 / fan_out))
                if getattr(m, "bias", None) is not None:
                    m.bias.zero_()
            elif isinstance(m, nn.Linear):
                nn.init.kaiming_normal_(m.weight)
                if m.bias is not None:
                    m.bias.zero_()
            elif isinstance(m, BatchNormAct2d):
                m.weight.fill_(1.0)
                m.bias.zero_()

    def set_crop_seed(self, seed):
        self.crop_seed = seed

    def _gen_random_crop_index(self, h, w):
        # parity with vae.py:62-82 (np seeded by crop_seed so labeled and
        # unlabeled batches crop identically within a step)
        rng = np.random.RandomState(self.crop_seed)
        if w < CROP_W and h < CROP_H:
            return 0, CROP_H, 0, CROP_W
        if w >= CROP_W and h >= CROP_H:
            w0 = rng.randint(w - CROP_W + 1)
            h0 = rng.randint(h - CROP_H + 1)
            return h0, h0 + CROP_H, w0, w0 + CROP_W
        raise ValueError("Unimplemented architecture for current image size.")

    def forward(self, x):
        # x: NCHW fp32
        h0, h1, w0, w1 = self._gen_random_crop_index(x.size(2), x.size(3))
        x_crop = x[:, :, h0:h1, w0:w1]
        z_feat = self._encode(x_crop)
        mu, logvar = self.fc_mu(z_feat), self.fc_logvar(z_feat)
        z = self.reparameterize(mu, logvar)
        recon = self._decode(z, batch=x.size(0))
        return x_crop, recon, z, mu, logvar

    def reparameterize(self, mu, logvar):
        stds = (0.5 * logvar).exp()
        epsilon = torch.randn_like(mu)
        return epsilon * stds + mu

    def _encode(self, x_nchw):
        x = nchw_to_nhwc(x_nchw)
        if x.is_cuda:
            x = x.to(torch.bfloat16)
        x = self.enc_bn1(self.enc_conv1(x))
        x = self.enc_bn2(self.enc_conv2(x))
        x = self.enc_bn3(self.enc_conv3(x))
        x = self.enc_bn4(self.enc_conv4(x))
        # (N, 2ls, 2ls, 1024) -> flatten; channel-last flatten order differs
        # from the reference's NCHW flatten, which only permutes fc_mu input
        # features — an internal representation choice, not a behavior change.
        return x.reshape(x.shape[0], -1).float()

    def _decode(self, z, batch):
        h = self.dec_fc(z)
        ls = self.ls
        h = h.reshape(batch, 1024, 4 * ls, 4 * ls)  # NCHW logical
        x = nchw_to_nhwc(h)
        if x.is_cuda:
            x = x.to(torch.bfloat16)
        x = self.dec_bn1(self.dec_conv1(x))
        x = self.dec_bn2(self.dec_conv2(x))
        x = self.dec_bn3(self.dec_conv3(x))
        x = self.dec_conv4(x)
        return nhwc_to_nchw(x.float())


class Discriminator(nn.Module):
    """z -> 512 -> 512 -> 1 MLP + Sigmoid (vaal_discriminator.py:5-22)."""

    def __init__(self, z_dim=10):
        super().__init__()
        self.z_dim = z_dim
        self.net = nn.Sequential(nn.Linear(z_dim, 512), nn.ReLU(True),
                                 nn.Linear(512, 512), nn.ReLU(True),
                                 nn.Linear(512, 1), nn.Sigmoid())
        self.weight_init()

    @torch.no_grad()
    def weight_init(self):
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.kaiming_normal_(m.weight)
                if m.bias is not None:
                    m.bias.zero_()

    def forward(self, z):
        return self.net(z)
