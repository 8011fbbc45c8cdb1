"""Shared conv/upconv/linear blocks for the backbones.

Parameter layout (module names / Sequential indices) intentionally matches the
reference blocks (reference models/dcgan_64.py:4-26, models/vgg_64.py:4-14,
models/h36m_mlp.py:28-46) so that state_dicts are interchangeable with
reference-trained checkpoints. The forward path is ours: on gfx950 the
Conv+BN+LeakyReLU group dispatches to the fused HIP kernels via p2pvg_amd.ops;
on CPU it runs the stock modules.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops.conv import Conv2d, ConvTranspose2d
from ...ops.norm import BatchNorm2d
from ...ops.fused_norm import FusedSequential


class dcgan_conv(nn.Module):
    """Conv2d(k4,s2,p1) + BatchNorm2d + LeakyReLU(0.2).

    pad_out: emit the activation as a padded zero-ring map so the NEXT
    stride-2 conv's gathers are in-bounds (glds staging pipeline)."""

    def __init__(self, nin: int, nout: int, pad_out: bool = False):
        super().__init__()
        self.main = FusedSequential(
            Conv2d(nin, nout, 4, 2, 1),
            BatchNorm2d(nout),
            nn.LeakyReLU(0.2, inplace=True),
            pad_out=pad_out,
        )

    def forward(self, x):
        return self.main(x)


class dcgan_upconv(nn.Module):
    """ConvTranspose2d(k4,s2,p1) + BatchNorm2d + LeakyReLU(0.2)."""

    def __init__(self, nin: int, nout: int):
        super().__init__()
        self.main = FusedSequential(
            ConvTranspose2d(nin, nout, 4, 2, 1),
            BatchNorm2d(nout),
            nn.LeakyReLU(0.2, inplace=True),
        )

    def forward(self, x):
        return self.main(x)


class vgg_layer(nn.Module):
    """Conv2d(k3,s1,p1) + BatchNorm2d + LeakyReLU(0.2).

    pad_out: emit the activation as a padded zero-ring map for the consumer
    OUTSIDE this block (pool, upsample, or the next stage's conv); inside a
    FusedSequential chain the ring decision is automatic."""

    def __init__(self, nin: int, nout: int, pad_out: bool = False):
        super().__init__()
        self.main = FusedSequential(
            Conv2d(nin, nout, 3, 1, 1),
            BatchNorm2d(nout),
            nn.LeakyReLU(0.2, inplace=True),
            pad_out=pad_out,
        )

    def forward(self, x):
        return self.main(x)


class residual_linear(nn.Module):
    """Linear shortcut + 3-Linear long path + LayerNorm
    (reference models/h36m_mlp.py:28-46)."""

    def __init__(self, nin: int, nout: int):
        super().__init__()
        self.shortcut = FusedSequential(nn.Linear(nin, nout), nn.ReLU(inplace=True))
        self.long_path = FusedSequential(
            nn.Linear(nin, nin // 2),
            nn.ReLU(inplace=True),
            nn.Linear(nin // 2, nin // 2),
            nn.ReLU(inplace=True),
            nn.Linear(nin // 2, nout),
            nn.ReLU(inplace=True),
        )
        self.norm = nn.LayerNorm(nout)

    def forward(self, x):
        return self.norm(self.shortcut(x) + self.long_path(x))
