#!/usr/bin/env python
"""Per-op microbench of the custom kernels at headline-bench shapes.

Run on a GPU box: python scripts/microbench_ops.py [batch]
Times each hot op standalone (CUDA events, 20 reps) to localize step-time
regressions without a full rocprof pass.
"""
import sys

import torch

sys.path.insert(0, ".")
from p2pvg_amd.ops import _hip_ext_loader  # noqa: E402

CL = torch.channels_last
ext = _hip_ext_loader.load()
B = int(sys.argv[1]) if len(sys.argv) > 1 else 448


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
    return s.elapsed_time(e) / n * 1000  # us


def mk(n, c, h, w):
    return torch.randn(n, c, h, w, device="cuda").bfloat16().contiguous(memory_format=CL)


def report(name, us):
    print(f"{name:<44} {us:>10.1f} us")


SHAPES = [  # (tag, C, H, K) vgg_64 encoder/decoder k3s1p1 convs
    ("c1.1 64ch 64sp", 64, 64, 64),
    ("c2.1 128ch 32sp", 128, 32, 128),
    ("c3.x 256ch 16sp", 256, 16, 256),
    ("c4.x 512ch 8sp", 512, 8, 512),
]

for tag, C, H, K in SHAPES:
    x = mk(B, C, H, H)
    w = mk(K, C, 3, 3)
    g = mk(B, K, H, H)
    gamma = torch.ones(K, device="cuda")
    beta = torch.zeros(K, device="cuda")
    rm = torch.zeros(K, device="cuda")
    rv = torch.ones(K, device="cuda")

    report(f"{tag} conv fwd (no stats)",
           timeit(lambda: ext.conv2d_nhwc_fwd(x, w, None, 1, 1, 0, False)))
    report(f"{tag} conv fwd + stats",
           timeit(lambda: ext.conv2d_nhwc_fwd(x, w, None, 1, 1, 0, True)))
    out, stats = ext.conv2d_nhwc_fwd(x, w, None, 1, 1, 0, True)
    report(f"{tag} bn_act_fwd_train (nb={stats.shape[0]})",
           timeit(lambda: ext.bn_act_fwd_train(out, stats, gamma, beta, rm, rv, 0.1, 1e-5, 1)))
    y, mean, invstd, scale = ext.bn_act_fwd_train(out, stats, gamma, beta, rm, rv, 0.1, 1e-5, 1)
    report(f"{tag} bn_act_bwd",
           timeit(lambda: ext.bn_act_bwd(out, g, mean, invstd, gamma, beta, scale, 1)))
    acc = torch.zeros(K, 3, 3, C, device="cuda")
    report(f"{tag} bn_act_bwd (acc)",
           timeit(lambda: ext.bn_act_bwd(out, g, mean, invstd, gamma, beta, scale, 1,
                                         gamma.clone(), beta.clone())))
    # padded (ring=1) variants: conv writes a padded map, bn walks rows
    outp, statsp = ext.conv2d_nhwc_fwd(x, w, None, 1, 1, 0, True,
                                       H + 2, H + 2, 1, 1)
    gp = mk(B, K, H + 2, H + 2)
    report(f"{tag} conv fwd padded-out",
           timeit(lambda: ext.conv2d_nhwc_fwd(x, w, None, 1, 1, 0, True,
                                              H + 2, H + 2, 1, 1)))
    report(f"{tag} bn_act_fwd ring=1",
           timeit(lambda: ext.bn_act_fwd_train(outp, statsp, gamma, beta, rm, rv,
                                               0.1, 1e-5, 1, 1)))
    yp, meanp, invstdp, scalep = ext.bn_act_fwd_train(outp, statsp, gamma, beta,
                                                      rm, rv, 0.1, 1e-5, 1, 1)
    report(f"{tag} bn_act_bwd ring=1",
           timeit(lambda: ext.bn_act_bwd(outp, gp, meanp, invstdp, gamma, beta,
                                         scalep, 1, None, None, 1)))
    xp = mk(B, C, H + 2, H + 2)
    report(f"{tag} glds fwd (padded in+out)",
           timeit(lambda: ext.conv2d_glds_fwd(xp, w, None, 1, 0, True,
                                              H + 2, H + 2, 1, 1)))
    report(f"{tag} wgrad yring=1",
           timeit(lambda: ext.conv2d_nhwc_wgrad(gp, xp, 3, 3, 1, 0, 0, None, 1)))
    report(f"{tag} wgrad",
           timeit(lambda: ext.conv2d_nhwc_wgrad(g, x, 3, 3, 1, 1, 0)))
    report(f"{tag} wgrad (acc)",
           timeit(lambda: ext.conv2d_nhwc_wgrad(g, x, 3, 3, 1, 1, 0, acc)))
    report(f"{tag} channel_sum",
           timeit(lambda: ext.channel_sum_nhwc(g)))
    print()

a = mk(B, 3, 64, 64)
b = mk(B, 3, 64, 64)
report("sqdiff_sum (B,3,64,64)", timeit(lambda: ext.sqdiff_sum(a, b)))
mu1, lv1, mu2, lv2 = (torch.randn(B, 10, device="cuda") for _ in range(4))
report("gaussian_kl_fwd (B,10)",
       timeit(lambda: ext.gaussian_kl_fwd(mu1, lv1, mu2, lv2, float(B))))
