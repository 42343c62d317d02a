# Copyright (c) Flashy-AMD authors.
"""DCGAN-style 64x64 generator/discriminator on the native NHWC gfx950
kernels (BASELINE.json config 4: adversarial G/D on 64x64 synthetic images).

Transposed convs run through the conv-kernel adjoints
(flashy_amd.nn.ConvTranspose2d); BatchNorm fuses ReLU/LeakyReLU; RGB edge
layers use the direct small-C kernels.  nz is 128 (the trunk requires
channel multiples of 64; the reference-style 100-d latent pads to 128).
GPU-only.
"""
from __future__ import annotations

import torch
from torch import nn

from .. import nn as fnn


class NativeDCGANGenerator(nn.Module):
    """latent z [N, nz] -> image [N, 3, 64, 64] (NCHW out for parity with
    the torch twin; internals are logical NHWC)."""

    def __init__(self, nz: int = 128, ngf: int = 64):
        super().__init__()
        assert nz % 64 == 0, "native trunk needs nz % 64 == 0 (e.g. 128)"
        self.nz = nz
        self.ct1 = fnn.ConvTranspose2d(nz, ngf * 8, 4, 1, 0, input_grad=False)
        self.bn1 = fnn.BatchNorm2d(ngf * 8)
        self.ct2 = fnn.ConvTranspose2d(ngf * 8, ngf * 4, 4, 2, 1)
        self.bn2 = fnn.BatchNorm2d(ngf * 4)
        self.ct3 = fnn.ConvTranspose2d(ngf * 4, ngf * 2, 4, 2, 1)
        self.bn3 = fnn.BatchNorm2d(ngf * 2)
        self.ct4 = fnn.ConvTranspose2d(ngf * 2, ngf, 4, 2, 1)
        self.bn4 = fnn.BatchNorm2d(ngf)
        self.head = fnn.ConvTranspose2d(ngf, 3, 4, 2, 1)  # RGB edge kernel

    def forward(self, z: torch.Tensor) -> torch.Tensor:
        if z.dim() == 4:  # accept [N, nz, 1, 1] like the torch twin
            z = z.reshape(z.shape[0], -1)
        x = z.reshape(z.shape[0], 1, 1, self.nz).to(torch.bfloat16)
        x = self.bn1(self.ct1(x), relu=True)
        x = self.bn2(self.ct2(x), relu=True)
        x = self.bn3(self.ct3(x), relu=True)
        x = self.bn4(self.ct4(x), relu=True)
        x = torch.tanh(self.head(x).float())
        return x.permute(0, 3, 1, 2)  # NCHW out


class NativeDCGANDiscriminator(nn.Module):
    """image [N, 3, 64, 64] -> logits [N]."""

    def __init__(self, ndf: int = 64):
        super().__init__()
        self.conv1 = fnn.Conv2d(3, ndf, 4, 2, 1)            # RGB edge kernel
        self.conv2 = fnn.Conv2d(ndf, ndf * 2, 4, 2, 1, feeds_bn=True)
        self.bn2 = fnn.BatchNorm2d(ndf * 2)
        self.conv3 = fnn.Conv2d(ndf * 2, ndf * 4, 4, 2, 1, feeds_bn=True)
        self.bn3 = fnn.BatchNorm2d(ndf * 4)
        self.conv4 = fnn.Conv2d(ndf * 4, ndf * 8, 4, 2, 1, feeds_bn=True)
        self.bn4 = fnn.BatchNorm2d(ndf * 8)
        self.head = fnn.Linear(ndf * 8 * 4 * 4, 1)  # = the final 4x4 conv

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.shape[1] == 3:  # NCHW -> logical NHWC
            x = x.permute(0, 2, 3, 1).contiguous()
        x = x.to(torch.bfloat16)
        x = torch.nn.functional.leaky_relu(self.conv1(x), 0.2)
        x = self.bn2(self.conv2(x), relu=True, slope=0.2)
        x = self.bn3(self.conv3(x), relu=True, slope=0.2)
        x = self.bn4(self.conv4(x), relu=True, slope=0.2)
        return self.head(x.float().reshape(x.shape[0], -1)).squeeze(1)
