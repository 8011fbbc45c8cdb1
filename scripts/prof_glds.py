#!/usr/bin/env python
"""A/B the glds 3-buffer conv against the register-pipeline kernel and MIOpen.

Run on a GPU box: python scripts/prof_glds.py [batch]
For each dense conv shape: checks numerics (glds on a pre-padded input ==
reference conv with pad) then times old kernel / glds / MIOpen forward.
"""
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, ".")
from p2pvg_amd.ops import _hip_ext_loader  # noqa: E402

CL = torch.channels_last
ext = _hip_ext_loader.load()
B = int(sys.argv[1]) if len(sys.argv) > 1 else 128

SHAPES = [
    ("vgg c1.1", B, 64, 64, 64, 64, 3, 1, 1),
    ("vgg c2.0", B, 64, 32, 32, 128, 3, 1, 1),
    ("vgg c2.1", B, 128, 32, 32, 128, 3, 1, 1),
    ("vgg c3.0", B, 128, 16, 16, 256, 3, 1, 1),
    ("vgg c3.x", B, 256, 16, 16, 256, 3, 1, 1),
    ("vgg c4.0", B, 256, 8, 8, 512, 3, 1, 1),
    ("vgg c4.x", B, 512, 8, 8, 512, 3, 1, 1),
    ("vgg d2cat", B, 1024, 8, 8, 512, 3, 1, 1),
    ("vgg d3cat", B, 512, 16, 16, 256, 3, 1, 1),
    ("vgg d4cat", B, 256, 32, 32, 128, 3, 1, 1),
    ("vgg d5cat", B, 128, 64, 64, 64, 3, 1, 1),
    ("dcgan c2", B, 64, 32, 32, 128, 4, 2, 1),
    ("dcgan c3", B, 128, 16, 16, 256, 4, 2, 1),
    ("dcgan c4", B, 256, 8, 8, 512, 4, 2, 1),
]


def timeit(f, n=20):
    for _ in range(3):
        f()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(n):
        f()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / n * 1000


print(f"{'shape':<12} {'old us':>9} {'glds us':>9} {'miopen us':>10} "
      f"{'glds/old':>9} {'glds/mi':>8}  TF  err")
for tag, n, c, h, w_, k, ks, st, pad in SHAPES:
    x = torch.randn(n, c, h, w_, device="cuda").bfloat16().contiguous(memory_format=CL)
    wt = (torch.randn(k, c, ks, ks, device="cuda") * 0.05).bfloat16().contiguous(memory_format=CL)
    xp = F.pad(x.float(), (pad,) * 4).bfloat16().contiguous(memory_format=CL)

    old = ext.conv2d_nhwc_fwd(x, wt, None, st, pad, 0, False)[0]
    glds = ext.conv2d_glds_fwd(xp, wt, None, st, 0, False)[0]
    ref = F.conv2d(x.float(), wt.float(), None, st, pad)
    err_old = (old.float() - ref).abs().max().item()
    err_glds = (glds.float() - ref).abs().max().item()
    scale = ref.abs().max().item()

    t_old = timeit(lambda: ext.conv2d_nhwc_fwd(x, wt, None, st, pad, 0, False))
    t_glds = timeit(lambda: ext.conv2d_glds_fwd(xp, wt, None, st, 0, False))
    t_mi = timeit(lambda: F.conv2d(x, wt, None, st, pad))

    ho = (h + 2 * pad - ks) // st + 1
    flops = 2.0 * n * ho * ho * k * c * ks * ks
    tf = flops / (t_glds * 1e-6) / 1e12
    ok = "OK" if err_glds < 0.02 * scale + 0.02 else f"FAIL({err_glds:.4f}/{scale:.2f})"
    print(f"{tag:<12} {t_old:>9.1f} {t_glds:>9.1f} {t_mi:>10.1f} "
          f"{t_old / t_glds:>9.2f} {t_mi / t_glds:>8.2f} {tf:>5.0f} {ok}"
          f"{'' if err_old < 0.02 * scale + 0.02 else ' OLD-FAIL'}")
