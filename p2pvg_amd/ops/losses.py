"""Autograd wrapper for the fused gaussian KL kernel (csrc/gaussian_kl.hip)."""
from __future__ import annotations

import torch


def _ext():
    from . import _hip_ext_loader

    return _hip_ext_loader.load()


class GaussianKLFn(torch.autograd.Function):
    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.float32)
    def forward(ctx, mu1, lv1, mu2, lv2, denom: float):
        mu1, lv1 = mu1.contiguous(), lv1.contiguous()
        mu2, lv2 = mu2.contiguous(), lv2.contiguous()
        out = _ext().gaussian_kl_fwd(mu1, lv1, mu2, lv2, float(denom))
        ctx.save_for_backward(mu1, lv1, mu2, lv2)
        ctx.denom = float(denom)
        return out

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, dout):
        mu1, lv1, mu2, lv2 = ctx.saved_tensors
        dmu1, dlv1, dmu2, dlv2 = _ext().gaussian_kl_bwd(
            mu1, lv1, mu2, lv2, dout.contiguous(), ctx.denom
        )
        return dmu1, dlv1, dmu2, dlv2, None
