"""Autograd wrappers + module subclasses for the NHWC pool/upsample kernels
(SURVEY §2.6 K6/K7)."""
from __future__ import annotations

import torch
import torch.nn as nn

CL = torch.channels_last


def _ext():
    from . import _hip_ext_loader

    return _hip_ext_loader.load()


class MaxPool2x2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, ri: int = 0, ro: int = 0):
        # ri/ro: zero-ring widths of the physical input/output maps (the
        # padded-activation scheme: a padded output's ring is the consuming
        # conv's padding)
        out, idx = _ext().maxpool2x2_fwd(x, ri, ro)
        ctx.save_for_backward(idx)
        ctx.hw = (x.shape[2] - 2 * ri, x.shape[3] - 2 * ri)
        ctx.rings = (ri, ro)
        return out

    @staticmethod
    def backward(ctx, gout):
        (idx,) = ctx.saved_tensors
        ri, ro = ctx.rings
        gout = gout.contiguous(memory_format=CL)
        if gout.dtype != torch.bfloat16:
            gout = gout.to(torch.bfloat16)
        return _ext().maxpool2x2_bwd(gout, idx, *ctx.hw, ri, ro), None, None


class Upsample2xFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, ri: int = 0, ro: int = 0):
        ctx.rings = (ri, ro)
        return _ext().upsample2x_fwd(x, ri, ro)

    @staticmethod
    def backward(ctx, gout):
        ri, ro = ctx.rings
        gout = gout.contiguous(memory_format=CL)
        if gout.dtype != torch.bfloat16:
            gout = gout.to(torch.bfloat16)
        return _ext().upsample2x_bwd(gout, ri, ro), None, None


def _dispatchable(x, ri: int = 0) -> bool:
    from .conv import _use_hip_path

    return (
        _use_hip_path(x)
        and x.dim() == 4
        and x.shape[1] % 8 == 0
        and (x.shape[2] - 2 * ri) % 2 == 0
        and (x.shape[3] - 2 * ri) % 2 == 0
    )


class MaxPool2d(nn.MaxPool2d):
    def forward(self, x, pad_out: bool = False):
        ri = getattr(x, "_pvg_pad", 0)
        if (
            self.kernel_size in (2, (2, 2))
            and self.stride in (2, (2, 2))
            and self.padding in (0, (0, 0))
            and _dispatchable(x, ri)
        ):
            with torch.autocast("cuda", enabled=False):
                from .conv import _to_cl_bf16

                ro = 1 if pad_out else 0
                y = MaxPool2x2Fn.apply(_to_cl_bf16(x), ri, ro)
                if ro:
                    y._pvg_pad = ro
                return y
        assert ri == 0, "padded input reached the stock MaxPool2d path"
        return super().forward(x)


class UpsamplingNearest2d(nn.UpsamplingNearest2d):
    def forward(self, x, pad_out: bool = False):
        ri = getattr(x, "_pvg_pad", 0)
        if self.scale_factor in (2, 2.0, (2, 2), (2.0, 2.0)) and _dispatchable(x, ri):
            with torch.autocast("cuda", enabled=False):
                from .conv import _to_cl_bf16

                ro = 1 if pad_out else 0
                y = Upsample2xFn.apply(_to_cl_bf16(x), ri, ro)
                if ro:
                    y._pvg_pad = ro
                return y
        assert ri == 0, "padded input reached the stock UpsamplingNearest2d path"
        return super().forward(x)
