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
    def forward(ctx, x):
        out, idx = _ext().maxpool2x2_fwd(x)
        ctx.save_for_backward(idx)
        ctx.hw = (x.shape[2], x.shape[3])
        return out

    @staticmethod
    def backward(ctx, gout):
        (idx,) = ctx.saved_tensors
        gout = gout.contiguous(memory_format=CL)
        if gout.dtype != torch.bfloat16:
            gout = gout.to(torch.bfloat16)
        return _ext().maxpool2x2_bwd(gout, idx, *ctx.hw)


class Upsample2xFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        return _ext().upsample2x_fwd(x)

    @staticmethod
    def backward(ctx, gout):
        gout = gout.contiguous(memory_format=CL)
        if gout.dtype != torch.bfloat16:
            gout = gout.to(torch.bfloat16)
        return _ext().upsample2x_bwd(gout)


def _dispatchable(x) -> bool:
    from .conv import _use_hip_path

    return (
        _use_hip_path(x)
        and x.dim() == 4
        and x.shape[1] % 8 == 0
        and x.shape[2] % 2 == 0
        and x.shape[3] % 2 == 0
    )


class MaxPool2d(nn.MaxPool2d):
    def forward(self, x):
        if (
            self.kernel_size in (2, (2, 2))
            and self.stride in (2, (2, 2))
            and self.padding in (0, (0, 0))
            and _dispatchable(x)
        ):
            with torch.autocast("cuda", enabled=False):
                from .conv import _to_cl_bf16

                return MaxPool2x2Fn.apply(_to_cl_bf16(x))
        return super().forward(x)


class UpsamplingNearest2d(nn.UpsamplingNearest2d):
    def forward(self, x):
        if self.scale_factor in (2, 2.0, (2, 2), (2.0, 2.0)) and _dispatchable(x):
            with torch.autocast("cuda", enabled=False):
                from .conv import _to_cl_bf16

                return Upsample2xFn.apply(_to_cl_bf16(x))
        return super().forward(x)
