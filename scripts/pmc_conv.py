#!/usr/bin/env python
"""Minimal conv-kernel-only workload for rocprofv3 PMC capture."""
import sys

import torch

sys.path.insert(0, "/root/repo")
from p2pvg_amd.ops import _hip_ext_loader

ext = _hip_ext_loader.load()
CL = torch.channels_last

N, C, H, W, K = 128, 256, 16, 16, 256
x = (torch.randn(N, C, H, W, device="cuda") * 0.5).bfloat16().contiguous(memory_format=CL)
w = (torch.randn(K, C, 3, 3, device="cuda") * 0.02).bfloat16().contiguous(memory_format=CL)
b = torch.randn(K, device="cuda").float()
for _ in range(10):
    out = ext.conv2d_nhwc_fwd(x, w, b, 1, 1, 0, False)[0]
torch.cuda.synchronize()
print("ok", out.shape)
