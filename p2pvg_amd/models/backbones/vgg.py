"""VGG-style backbones for 64x64 and 128x128 frames.

Capability parity with reference models/vgg_64.py:16-105 and
models/vgg_128.py:16-120 (shape tables in SURVEY §2.3): 3x3-conv blocks with
2x2 maxpool between encoder stages; decoder is nearest-upsample + conv blocks
with channel-concat skips. state_dict keys match the reference modules.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops.conv import Conv2d, ConvTranspose2d
from ...ops.norm import BatchNorm2d
from ...ops.fused_norm import FusedSequential
from ...ops.pool import MaxPool2d, UpsamplingNearest2d

from .blocks import vgg_layer


def _cat_skip(a, b):
    """Channel-concat whose padded-map marker survives (torch.cat drops
    python attrs); both operands share the same zero-ring width by
    construction (upsample out and the encoder skip)."""
    r = torch.cat([a, b], 1)
    pa = getattr(a, "_pvg_pad", 0)
    if pa and pa == getattr(b, "_pvg_pad", 0):
        r._pvg_pad = pa
    return r


def _skip_join(a, b):
    """Hand (up_out, skip) to the consuming conv as a PAIR when the
    dual-pointer glds path can gather both directly (SURVEY K8 concat
    elimination — no cat kernel, no cat backward splits); falls back to the
    marked concat otherwise (CPU, torch backend, odd channel splits). The
    first fused conv of the receiving FusedSequential consumes the pair."""
    from ...ops.conv import _use_hip_path

    if (getattr(a, "_pvg_pad", 0) and _use_hip_path(a)
            and a.shape[1] % 64 == 0 and b.shape[1] % 64 == 0
            and getattr(a, "_pvg_pad", 0) == getattr(b, "_pvg_pad", 0)):
        return (a, b)
    return _cat_skip(a, b)


class Encoder64(nn.Module):
    def __init__(self, dim: int, nc: int = 1):
        super().__init__()
        self.dim = dim
        P = dict(pad_out=True)
        self.c1 = FusedSequential(vgg_layer(nc, 64, **P), vgg_layer(64, 64, **P))
        self.c2 = FusedSequential(vgg_layer(64, 128, **P), vgg_layer(128, 128, **P))
        self.c3 = FusedSequential(
            vgg_layer(128, 256, **P), vgg_layer(256, 256, **P), vgg_layer(256, 256, **P)
        )
        self.c4 = FusedSequential(
            vgg_layer(256, 512, **P), vgg_layer(512, 512, **P), vgg_layer(512, 512, **P)
        )
        self.c5 = FusedSequential(
            Conv2d(512, dim, 4, 1, 0), BatchNorm2d(dim), nn.Tanh()
        )
        self.mp = MaxPool2d(kernel_size=2, stride=2, padding=0)

    def forward(self, x):
        h1 = self.c1(x)
        h2 = self.c2(self.mp(h1, pad_out=True))
        h3 = self.c3(self.mp(h2, pad_out=True))
        h4 = self.c4(self.mp(h3, pad_out=True))
        h5 = self.c5(self.mp(h4))
        return h5.view(-1, self.dim), [h1, h2, h3, h4]


class Decoder64(nn.Module):
    def __init__(self, dim: int, nc: int = 1):
        super().__init__()
        self.dim = dim
        self.upc1 = FusedSequential(
            ConvTranspose2d(dim, 512, 4, 1, 0),
            BatchNorm2d(512),
            nn.LeakyReLU(0.2, inplace=True),
        )
        P = dict(pad_out=True)
        self.upc2 = FusedSequential(
            vgg_layer(512 * 2, 512, **P), vgg_layer(512, 512, **P), vgg_layer(512, 256, **P)
        )
        self.upc3 = FusedSequential(
            vgg_layer(256 * 2, 256, **P), vgg_layer(256, 256, **P), vgg_layer(256, 128, **P)
        )
        self.upc4 = FusedSequential(vgg_layer(128 * 2, 128, **P), vgg_layer(128, 64, **P))
        self.upc5 = FusedSequential(
            vgg_layer(64 * 2, 64, **P),
            ConvTranspose2d(64, nc, 3, 1, 1),
            nn.Sigmoid(),
        )
        self.up = UpsamplingNearest2d(scale_factor=2)

    def forward(self, inp):
        vec, skip = inp
        d1 = self.upc1(vec.view(-1, self.dim, 1, 1))
        d2 = self.upc2(_skip_join(self.up(d1, pad_out=True), skip[3]))
        d3 = self.upc3(_skip_join(self.up(d2, pad_out=True), skip[2]))
        d4 = self.upc4(_skip_join(self.up(d3, pad_out=True), skip[1]))
        return self.upc5(_skip_join(self.up(d4, pad_out=True), skip[0]))


class Encoder128(nn.Module):
    def __init__(self, dim: int, nc: int = 1):
        super().__init__()
        self.dim = dim
        P = dict(pad_out=True)
        self.c1 = FusedSequential(vgg_layer(nc, 64, **P), vgg_layer(64, 64, **P))
        self.c2 = FusedSequential(vgg_layer(64, 128, **P), vgg_layer(128, 128, **P))
        self.c3 = FusedSequential(
            vgg_layer(128, 256, **P), vgg_layer(256, 256, **P), vgg_layer(256, 256, **P)
        )
        self.c4 = FusedSequential(
            vgg_layer(256, 512, **P), vgg_layer(512, 512, **P), vgg_layer(512, 512, **P)
        )
        self.c5 = FusedSequential(
            vgg_layer(512, 512, **P), vgg_layer(512, 512, **P), vgg_layer(512, 512, **P)
        )
        self.c6 = FusedSequential(
            Conv2d(512, dim, 4, 1, 0), BatchNorm2d(dim), nn.Tanh()
        )
        self.mp = MaxPool2d(kernel_size=2, stride=2, padding=0)

    def forward(self, x):
        h1 = self.c1(x)
        h2 = self.c2(self.mp(h1, pad_out=True))
        h3 = self.c3(self.mp(h2, pad_out=True))
        h4 = self.c4(self.mp(h3, pad_out=True))
        h5 = self.c5(self.mp(h4, pad_out=True))
        h6 = self.c6(self.mp(h5))
        return h6.view(-1, self.dim), [h1, h2, h3, h4, h5]


class Decoder128(nn.Module):
    def __init__(self, dim: int, nc: int = 1):
        super().__init__()
        self.dim = dim
        self.upc1 = FusedSequential(
            ConvTranspose2d(dim, 512, 4, 1, 0),
            BatchNorm2d(512),
            nn.LeakyReLU(0.2, inplace=True),
        )
        P = dict(pad_out=True)
        self.upc2 = FusedSequential(
            vgg_layer(512 * 2, 512, **P), vgg_layer(512, 512, **P), vgg_layer(512, 512, **P)
        )
        self.upc3 = FusedSequential(
            vgg_layer(512 * 2, 512, **P), vgg_layer(512, 512, **P), vgg_layer(512, 256, **P)
        )
        self.upc4 = FusedSequential(
            vgg_layer(256 * 2, 256, **P), vgg_layer(256, 256, **P), vgg_layer(256, 128, **P)
        )
        self.upc5 = FusedSequential(vgg_layer(128 * 2, 128, **P), vgg_layer(128, 64, **P))
        self.upc6 = FusedSequential(
            vgg_layer(64 * 2, 64, **P),
            ConvTranspose2d(64, nc, 3, 1, 1),
            nn.Sigmoid(),
        )
        self.up = UpsamplingNearest2d(scale_factor=2)

    def forward(self, inp):
        vec, skip = inp
        d1 = self.upc1(vec.view(-1, self.dim, 1, 1))
        d2 = self.upc2(_skip_join(self.up(d1, pad_out=True), skip[4]))
        d3 = self.upc3(_skip_join(self.up(d2, pad_out=True), skip[3]))
        d4 = self.upc4(_skip_join(self.up(d3, pad_out=True), skip[2]))
        d5 = self.upc5(_skip_join(self.up(d4, pad_out=True), skip[1]))
        return self.upc6(_skip_join(self.up(d5, pad_out=True), skip[0]))
