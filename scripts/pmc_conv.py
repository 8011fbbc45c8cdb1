#!/usr/bin/env python
"""Minimal conv-kernel-only workload for rocprofv3 PMC capture."""
import sys

import torch

sys.path.insert(0, "/root/repo")
from p2pvg_amd.ops import _hip_ext_loader

ext = _hip_ext_loader.load()
CL = torch.channels_last

import torch.nn.functional as F

mode = sys.argv[1] if len(sys.argv) > 1 else "glds"
N, C, H, W, K = 448, 256, 16, 16, 256
x = (torch.randn(N, C, H, W, device="cuda") * 0.5).bfloat16().contiguous(memory_format=CL)
w = (torch.randn(K, C, 3, 3, device="cuda") * 0.02).bfloat16().contiguous(memory_format=CL)
b = torch.randn(K, device="cuda").float()
if mode == "glds":
    xp = F.pad(x.float(), (1, 1, 1, 1)).bfloat16().contiguous(memory_format=CL)
    for _ in range(10):
        out = ext.conv2d_glds_fwd(xp, w, b, 1, 0, False)[0]
else:
    import os

    os.environ["P2PVG_GLDS"] = "0"
    for _ in range(10):
        out = ext.conv2d_nhwc_fwd(x, w, b, 1, 1, 0, False)[0]
torch.cuda.synchronize()
print("ok", mode, out.shape)
