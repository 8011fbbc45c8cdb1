"""Human3.6M skeleton MLP backbone (non-image modality).

Capability parity with reference models/h36m_mlp.py:49-95: residual-linear
encoder 17*3 -> g_dim with two MLP skips, decoder g_dim(+skips) -> 17x3 with no
output squashing. state_dict keys match the reference modules.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from .blocks import residual_linear


class Encoder(nn.Module):
    def __init__(self, in_dim: int = 17 * 3, out_dim: int = 128, h_dim: int = 128):
        super().__init__()
        self.in_dim = in_dim
        self.out_dim = out_dim
        self.h_dim = h_dim
        self.fc1 = residual_linear(in_dim, h_dim)
        self.fc2 = residual_linear(h_dim, h_dim)
        self.fc3 = nn.Linear(h_dim, out_dim)
        self.tanh = nn.Tanh()

    def forward(self, x):
        bs = x.shape[0]
        x = x.reshape(bs, -1)
        h1 = self.fc1(x)
        h2 = self.fc2(h1)
        out = self.tanh(self.fc3(h2))
        return out, [h1, h2]


class Decoder(nn.Module):
    def __init__(self, in_dim: int = 128, out_dim: int = 17 * 3, h_dim: int = 128):
        super().__init__()
        self.in_dim = in_dim
        self.h_dim = h_dim
        self.out_dim = out_dim
        self.fc1 = residual_linear(in_dim, h_dim)
        self.fc2 = residual_linear(h_dim * 2, h_dim)
        self.fc3 = nn.Linear(h_dim * 2, out_dim)

    def forward(self, inp):
        x, skip = inp
        bs = x.shape[0]
        d1 = self.fc1(x)
        d2 = self.fc2(torch.cat([d1, skip[1]], 1))
        out = self.fc3(torch.cat([d2, skip[0]], 1))
        return out.view(bs, 17, 3)
