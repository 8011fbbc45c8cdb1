#!/usr/bin/env python
"""Per-shape A/B of the gfx950 conv kernels vs MIOpen (fwd / dgrad / wgrad).

Run on a GPU box:  python scripts/prof_convs.py [batch]
Prints one line per (shape, op): hip us, miopen us, ratio.
"""
import sys
import time

import torch

sys.path.insert(0, ".")
from p2pvg_amd.ops import _hip_ext_loader  # noqa: E402
from p2pvg_amd.ops.conv import Conv2dNHWCFn, ConvT2dNHWCFn  # noqa: E402

CL = torch.channels_last
ext = _hip_ext_loader.load()

B = int(sys.argv[1]) if len(sys.argv) > 1 else 128

# the vgg_64 + dcgan_64 training shapes
CONVS = [
    ("vgg c1.0", B, 3, 64, 64, 64, 3, 1, 1),
    ("vgg c1.1", B, 64, 64, 64, 64, 3, 1, 1),
    ("vgg c2.0", B, 64, 32, 32, 128, 3, 1, 1),
    ("vgg c2.1", B, 128, 32, 32, 128, 3, 1, 1),
    ("vgg c3.0", B, 128, 16, 16, 256, 3, 1, 1),
    ("vgg c3.x", B, 256, 16, 16, 256, 3, 1, 1),
    ("vgg c4.0", B, 256, 8, 8, 512, 3, 1, 1),
    ("vgg c4.x", B, 512, 8, 8, 512, 3, 1, 1),
    ("vgg c5", B, 512, 4, 4, 128, 4, 1, 0),
    ("vgg d2cat", B, 1024, 8, 8, 512, 3, 1, 1),
    ("vgg d3cat", B, 512, 16, 16, 256, 3, 1, 1),
    ("vgg d4cat", B, 256, 32, 32, 128, 3, 1, 1),
    ("vgg d5cat", B, 128, 64, 64, 64, 3, 1, 1),
    ("dcgan c1", B, 1, 64, 64, 64, 4, 2, 1),
    ("dcgan c2", B, 64, 32, 32, 128, 4, 2, 1),
    ("dcgan c3", B, 128, 16, 16, 256, 4, 2, 1),
    ("dcgan c4", B, 256, 8, 8, 512, 4, 2, 1),
]
CONVTS = [
    ("dc upc1", B, 128, 1, 1, 512, 4, 1, 0),
    ("dc upc2", B, 1024, 4, 4, 256, 4, 2, 1),
    ("dc upc3", B, 512, 8, 8, 128, 4, 2, 1),
    ("dc upc4", B, 256, 16, 16, 64, 4, 2, 1),
    ("dc upc5", B, 128, 32, 32, 1, 4, 2, 1),
    ("vgg out", B, 64, 64, 64, 3, 3, 1, 1),
]


def timeit(f, n=10):
    for _ in range(3):
        f()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        f()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6


def bench_conv(name, N, C, H, W, K, ks, st, pad, transposed=False):
    torch.manual_seed(0)
    if transposed:
        x = torch.randn(N, C, H, W, device="cuda").bfloat16()
        w = (torch.randn(C, K, ks, ks, device="cuda") * 0.05).bfloat16()
        ref_f = lambda: torch.nn.functional.conv_transpose2d(xl, wl, None, st, pad)  # noqa: E731
        Fn = ConvT2dNHWCFn
    else:
        x = torch.randn(N, C, H, W, device="cuda").bfloat16()
        w = (torch.randn(K, C, ks, ks, device="cuda") * 0.05).bfloat16()
        ref_f = lambda: torch.nn.functional.conv2d(xl, wl, None, st, pad)  # noqa: E731
        Fn = Conv2dNHWCFn
    xl = x.contiguous(memory_format=CL)
    wl = w.contiguous(memory_format=CL)

    # fwd
    t_ref = timeit(ref_f)
    t_hip = timeit(lambda: Fn.apply(xl, wl, None, st, pad, 0, False)[0])
    print(f"{name:10s} fwd  hip {t_hip:8.1f}us  miopen {t_ref:8.1f}us  x{t_ref/t_hip:5.2f}")

    # full fwd+bwd
    xg = xl.detach().requires_grad_()
    wg = wl.detach().requires_grad_()
    g = torch.randn_like(Fn.apply(xg, wg, None, st, pad, 0, False)[0])

    def hip_fb():
        out, _ = Fn.apply(xg, wg, None, st, pad, 0, False)
        out.backward(g)

    xr = xl.detach().requires_grad_()
    wr = wl.detach().requires_grad_()

    def ref_fb():
        if transposed:
            out = torch.nn.functional.conv_transpose2d(xr, wr, None, st, pad)
        else:
            out = torch.nn.functional.conv2d(xr, wr, None, st, pad)
        out.backward(g)

    t_ref = timeit(ref_fb)
    t_hip = timeit(hip_fb)
    # isolated wgrad kernel time
    if transposed:
        t_wg = timeit(lambda: ext.conv2d_nhwc_wgrad(xl, g.bfloat16().contiguous(memory_format=CL), ks, ks, st, pad, 0))
    else:
        t_wg = timeit(lambda: ext.conv2d_nhwc_wgrad(g.bfloat16().contiguous(memory_format=CL), xl, ks, ks, st, pad, 0))
    print(f"{name:10s} f+b  hip {t_hip:8.1f}us  miopen {t_ref:8.1f}us  x{t_ref/t_hip:5.2f}  (wgrad {t_wg:7.1f}us)")


for row in CONVS:
    bench_conv(*row)
for row in CONVTS:
    bench_conv(*row, transposed=True)
