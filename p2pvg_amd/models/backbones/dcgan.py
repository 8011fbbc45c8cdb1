"""DCGAN-style strided-conv backbones for 64x64 and 128x128 frames.

Capability parity with reference models/dcgan_64.py:28-88 and
models/dcgan_128.py:28-94 (shape tables in SURVEY §2.3): 5/6-stage stride-2
encoder to a g_dim vector + U-Net skip list, mirrored ConvTranspose decoder with
channel-concat skips. state_dict keys match the reference modules.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops.conv import Conv2d, ConvTranspose2d
from ...ops.norm import BatchNorm2d
from ...ops.fused_norm import FusedSequential

from .blocks import dcgan_conv, dcgan_upconv


def _interior(t):
    """Logical (interior) view of a padded zero-ring map; identity on dense
    tensors. The DCGAN decoder runs dense, so padded encoder skips are
    sliced before the concat (the slice copy happens inside the consuming
    conv's contiguous() — the same cost the cat always paid)."""
    p = getattr(t, "_pvg_pad", 0)
    return t[:, :, p:-p, p:-p] if p else t


class Encoder64(nn.Module):
    def __init__(self, dim: int, nc: int = 1):
        super().__init__()
        self.dim = dim
        nf = 64
        self.c1 = dcgan_conv(nc, nf, pad_out=True)       # nc x64x64 -> nf x32x32
        self.c2 = dcgan_conv(nf, nf * 2, pad_out=True)    # -> 128x16x16
        self.c3 = dcgan_conv(nf * 2, nf * 4, pad_out=True)  # -> 256x8x8
        self.c4 = dcgan_conv(nf * 4, nf * 8)              # -> 512x4x4 (dense: c5 GEMM)
        self.c5 = FusedSequential(                  # -> dim x1x1
            Conv2d(nf * 8, dim, 4, 1, 0),
            BatchNorm2d(dim),
            nn.Tanh(),
        )

    def forward(self, x):
        h1 = self.c1(x)
        h2 = self.c2(h1)
        h3 = self.c3(h2)
        h4 = self.c4(h3)
        h5 = self.c5(h4)
        return h5.view(-1, self.dim), [h1, h2, h3, h4]


class Decoder64(nn.Module):
    def __init__(self, dim: int, nc: int = 1):
        super().__init__()
        self.dim = dim
        nf = 64
        self.upc1 = FusedSequential(
            ConvTranspose2d(dim, nf * 8, 4, 1, 0),
            BatchNorm2d(nf * 8),
            nn.LeakyReLU(0.2, inplace=True),
        )
        self.upc2 = dcgan_upconv(nf * 8 * 2, nf * 4)
        self.upc3 = dcgan_upconv(nf * 4 * 2, nf * 2)
        self.upc4 = dcgan_upconv(nf * 2 * 2, nf)
        self.upc5 = FusedSequential(
            ConvTranspose2d(nf * 2, nc, 4, 2, 1),
            nn.Sigmoid(),
        )

    def forward(self, inp):
        vec, skip = inp
        d1 = self.upc1(vec.view(-1, self.dim, 1, 1))
        d2 = self.upc2(torch.cat([d1, _interior(skip[3])], 1))
        d3 = self.upc3(torch.cat([d2, _interior(skip[2])], 1))
        d4 = self.upc4(torch.cat([d3, _interior(skip[1])], 1))
        return self.upc5(torch.cat([d4, _interior(skip[0])], 1))


class Encoder128(nn.Module):
    """6-stage variant for 128x128 (reference models/dcgan_128.py:28-57)."""

    def __init__(self, dim: int, nc: int = 1):
        super().__init__()
        self.dim = dim
        nf = 64
        self.c1 = dcgan_conv(nc, nf, pad_out=True)        # nc x128 -> 64x64
        self.c2 = dcgan_conv(nf, nf * 2, pad_out=True)     # -> 128x32
        self.c3 = dcgan_conv(nf * 2, nf * 4, pad_out=True)  # -> 256x16
        self.c4 = dcgan_conv(nf * 4, nf * 8, pad_out=True)  # -> 512x8
        self.c5 = dcgan_conv(nf * 8, nf * 8)               # -> 512x4 (dense: c6 GEMM)
        self.c6 = FusedSequential(
            Conv2d(nf * 8, dim, 4, 1, 0),
            BatchNorm2d(dim),
            nn.Tanh(),
        )

    def forward(self, x):
        h1 = self.c1(x)
        h2 = self.c2(h1)
        h3 = self.c3(h2)
        h4 = self.c4(h3)
        h5 = self.c5(h4)
        h6 = self.c6(h5)
        return h6.view(-1, self.dim), [h1, h2, h3, h4, h5]


class Decoder128(nn.Module):
    def __init__(self, dim: int, nc: int = 1):
        super().__init__()
        self.dim = dim
        nf = 64
        self.upc1 = FusedSequential(
            ConvTranspose2d(dim, nf * 8, 4, 1, 0),
            BatchNorm2d(nf * 8),
            nn.LeakyReLU(0.2, inplace=True),
        )
        self.upc2 = dcgan_upconv(nf * 8 * 2, nf * 8)
        self.upc3 = dcgan_upconv(nf * 8 * 2, nf * 4)
        self.upc4 = dcgan_upconv(nf * 4 * 2, nf * 2)
        self.upc5 = dcgan_upconv(nf * 2 * 2, nf)
        self.upc6 = FusedSequential(
            ConvTranspose2d(nf * 2, nc, 4, 2, 1),
            nn.Sigmoid(),
        )

    def forward(self, inp):
        vec, skip = inp
        d1 = self.upc1(vec.view(-1, self.dim, 1, 1))
        d2 = self.upc2(torch.cat([d1, _interior(skip[4])], 1))
        d3 = self.upc3(torch.cat([d2, _interior(skip[3])], 1))
        d4 = self.upc4(torch.cat([d3, _interior(skip[2])], 1))
        d5 = self.upc5(torch.cat([d4, _interior(skip[1])], 1))
        return self.upc6(torch.cat([d5, _interior(skip[0])], 1))
