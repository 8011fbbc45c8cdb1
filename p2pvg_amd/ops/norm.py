"""BatchNorm2d with a composite path for degenerate shapes.

MIOpen's NHWC bf16 batch-norm segfaults on (N, C, 1, 1) activations (the
encoder tail applies BN to the 1x1 latent map — reference models/dcgan_64.py:44,
SURVEY §7 'hard parts'). For that shape BN is just a per-channel batch
standardization; this subclass computes it with plain fp32 tensor ops
(autograd-friendly, identical running-stat semantics) and defers to the stock
implementation everywhere else. state_dict-compatible with nn.BatchNorm2d.
"""
from __future__ import annotations

import torch
import torch.nn as nn


class BatchNorm2d(nn.BatchNorm2d):
    def forward(self, x):
        if x.is_cuda and x.dim() == 4 and x.shape[2] == 1 and x.shape[3] == 1:
            xf = x.float()
            if self.training or not self.track_running_stats:
                mean = xf.mean(dim=(0, 2, 3))
                var = xf.var(dim=(0, 2, 3), unbiased=False)
                if self.track_running_stats:
                    with torch.no_grad():
                        n = x.numel() / x.shape[1]
                        mom = self.momentum if self.momentum is not None else 0.1
                        self.running_mean.mul_(1 - mom).add_(mean.detach(), alpha=mom)
                        ub = var.detach() * (n / max(n - 1, 1))
                        self.running_var.mul_(1 - mom).add_(ub, alpha=mom)
                        self.num_batches_tracked += 1
            else:
                mean = self.running_mean
                var = self.running_var
            inv = torch.rsqrt(var + self.eps)
            y = (xf - mean.view(1, -1, 1, 1)) * inv.view(1, -1, 1, 1)
            if self.affine:
                y = y * self.weight.float().view(1, -1, 1, 1) + self.bias.float().view(1, -1, 1, 1)
            return y.to(x.dtype)
        return super().forward(x)
