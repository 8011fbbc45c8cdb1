"""Autograd integration of the gfx950 NHWC conv kernels.

Forward, dgrad and wgrad all run on the in-tree MFMA kernels:
- fwd: conv2d_nhwc_fwd (implicit GEMM, LDS-staged im2col).
- dgrad stride 1: the same fwd kernel on grad_out with flipped/transposed
  weights (pad = k-1-p).
- dgrad stride 2 / ConvTranspose fwd: parity decomposition — four dense
  stride-1 sub-convolutions with per-parity weight slices, scatter-written
  into the interleaved output (conv2d_nhwc_fwd_scatter). No zero-dilated
  intermediate, no transposes.
- wgrad: conv2d_nhwc_wgrad (pixel-major TN GEMM, split-K over pixel slabs,
  fp32 workspace accumulation).

Modules Conv2d / ConvTranspose2d subclass the torch ones (state_dict keys
unchanged) and route to this path for CUDA bf16/autocast inputs.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn as nn

CL = torch.channels_last


def _ext():
    from . import _hip_ext_loader

    return _hip_ext_loader.load()


def _to_cl_bf16(t: torch.Tensor) -> torch.Tensor:
    if t.dtype != torch.bfloat16:
        t = t.to(torch.bfloat16)
    return t.contiguous(memory_format=CL)


def parity_taps(k: int, pad: int, stride: int, p: int) -> Tuple[List[int], int]:
    """Valid taps r (descending) and implicit pad for the parity-p compact
    stride-1 sub-convolution of a fractionally-strided conv."""
    taps = [(r, (p + pad - r) // stride) for r in range(k)
            if (p + pad - r) % stride == 0]
    taps.sort(key=lambda t: t[1])  # ascending offset
    offs = [t[1] for t in taps]
    ipad = -offs[0]
    assert all(o == i - ipad for i, o in enumerate(offs)), (k, pad, stride, p, offs)
    return [t[0] for t in taps], ipad


@torch.no_grad()
def _fracstride_apply(inp, weight_kcrs, bias, k, pad, stride, out):
    """Shared machinery for convT fwd / conv dgrad(s>1): weight_kcrs is the
    weight already arranged logically as (K_out, C_in, k, k) channels_last.
    Fills `out` (N, K_out, H*stride.., W*stride..) channels_last in place."""
    ext = _ext()
    for py in range(stride):
        rs_y, ipad_y = parity_taps(k, pad, stride, py)
        for px in range(stride):
            rs_x, ipad_x = parity_taps(k, pad, stride, px)
            wc = weight_kcrs[:, :, rs_y][:, :, :, rs_x].contiguous(memory_format=CL)
            ext.conv2d_nhwc_fwd_scatter(
                inp, wc, bias, out, ipad_y, ipad_x, stride, py, px, 0
            )
    return out


class _ConvFwdCtx:
    pass


class Conv2dNHWCFn(torch.autograd.Function):
    """y = conv2d(x, w, b, stride, pad) with x, w bf16 channels_last."""

    @staticmethod
    def forward(ctx, x, w, b, stride: int, pad: int):
        ext = _ext()
        b32 = b.float() if b is not None else None
        out = ext.conv2d_nhwc_fwd(x, w, b32, stride, pad, 0)
        ctx.save_for_backward(x, w)
        ctx.stride, ctx.pad, ctx.has_bias = stride, pad, b is not None
        return out

    @staticmethod
    def backward(ctx, gout):
        x, w = ctx.saved_tensors
        stride, pad = ctx.stride, ctx.pad
        k = w.shape[2]
        ext = _ext()
        gout = gout.contiguous(memory_format=CL)
        if gout.dtype != torch.bfloat16:
            gout = gout.to(torch.bfloat16)

        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            if stride == 1:
                wt = w.flip(2, 3).transpose(0, 1).contiguous(memory_format=CL)
                dx = ext.conv2d_nhwc_fwd(gout, wt, None, 1, k - 1 - pad, 0)
            else:
                wt = w.transpose(0, 1)  # (C, K, k, k) logical
                dx = torch.empty_like(x)
                _fracstride_apply(gout, wt, None, k, pad, stride, dx)
        if ctx.needs_input_grad[1]:
            ws = ext.conv2d_nhwc_wgrad(gout, x, k, k, stride, pad, 0)
            # ws is (K, R, S, C) fp32 == physical layout of the channels_last
            # (K, C, R, S) weight grad
            dw = ws.permute(0, 3, 1, 2).to(torch.bfloat16) \
                .contiguous(memory_format=CL)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = gout.float().sum(dim=(0, 2, 3))
        return dx, dw, db, None, None


class ConvT2dNHWCFn(torch.autograd.Function):
    """y = conv_transpose2d(x, w, b, stride, pad); w logical (Ci, Co, k, k)."""

    @staticmethod
    def forward(ctx, x, w, b, stride: int, pad: int):
        ext = _ext()
        k = w.shape[2]
        ci, co = w.shape[0], w.shape[1]
        n, _, h, wdt = x.shape
        b32 = b.float() if b is not None else None
        if stride == 1:
            # convT s1 p: out = conv(x, flip(w)^T, pad=k-1-p)
            wt = w.flip(2, 3).transpose(0, 1).contiguous(memory_format=CL)  # (Co,Ci,k,k)
            out = ext.conv2d_nhwc_fwd(x, wt, b32, 1, k - 1 - pad, 0)
        else:
            oh = (h - 1) * stride - 2 * pad + k
            ow = (wdt - 1) * stride - 2 * pad + k
            out = torch.empty((n, co, oh, ow), dtype=x.dtype, device=x.device
                              ).contiguous(memory_format=CL)
            wt = w.transpose(0, 1)  # (Co, Ci, k, k) logical
            _fracstride_apply(x, wt, b32, k, pad, stride, out)
        ctx.save_for_backward(x, w)
        ctx.stride, ctx.pad, ctx.has_bias = stride, pad, b is not None
        return out

    @staticmethod
    def backward(ctx, gout):
        x, w = ctx.saved_tensors
        stride, pad = ctx.stride, ctx.pad
        k = w.shape[2]
        ext = _ext()
        gout = gout.contiguous(memory_format=CL)
        if gout.dtype != torch.bfloat16:
            gout = gout.to(torch.bfloat16)

        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            # dgrad of convT = plain conv with the untransposed weight
            wl = w.contiguous(memory_format=CL)  # (Ci, Co, k, k): Co in, Ci out
            dx = ext.conv2d_nhwc_fwd(gout, wl, None, stride, pad, 0)
        if ctx.needs_input_grad[1]:
            # dW[ci, r, s, co] = sum in[...,ci] * gout[scatter...,co]
            ws = ext.conv2d_nhwc_wgrad(x, gout, k, k, stride, pad, 0)
            dw = ws.permute(0, 3, 1, 2).to(torch.bfloat16) \
                .contiguous(memory_format=CL)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = gout.float().sum(dim=(0, 2, 3))
        return dx, dw, db, None, None


def _use_hip_path(x: torch.Tensor) -> bool:
    from . import backend_mode, hip_available

    if not x.is_cuda or backend_mode() == "torch":
        return False
    if not (x.dtype == torch.bfloat16 or torch.is_autocast_enabled("cuda")):
        return False
    return hip_available()


class Conv2d(nn.Conv2d):
    """nn.Conv2d that routes CUDA bf16/autocast inputs through the gfx950
    implicit-GEMM kernels. state_dict-compatible with nn.Conv2d."""

    def forward(self, x):
        if (
            _use_hip_path(x)
            and self.kernel_size[0] == self.kernel_size[1]
            and self.stride[0] == self.stride[1]
            and self.padding[0] == self.padding[1]
            and self.kernel_size[0] in (1, 2, 3, 4)
            and self.stride[0] in (1, 2)
            and self.dilation == (1, 1)
            and self.groups == 1
        ):
            with torch.autocast("cuda", enabled=False):
                xl = _to_cl_bf16(x)
                wl = _to_cl_bf16(self.weight)
                return Conv2dNHWCFn.apply(
                    xl, wl, self.bias, self.stride[0], self.padding[0]
                )
        return super().forward(x)


class ConvTranspose2d(nn.ConvTranspose2d):
    """nn.ConvTranspose2d on the gfx950 parity-decomposed scatter kernels."""

    def forward(self, x, output_size=None):
        if (
            _use_hip_path(x)
            and output_size is None
            and self.kernel_size[0] == self.kernel_size[1]
            and self.stride[0] == self.stride[1]
            and self.padding[0] == self.padding[1]
            and self.kernel_size[0] in (1, 2, 3, 4)
            and self.stride[0] in (1, 2)
            and self.dilation == (1, 1)
            and self.groups == 1
            and self.output_padding == (0, 0)
        ):
            with torch.autocast("cuda", enabled=False):
                xl = _to_cl_bf16(x)
                wl = _to_cl_bf16(self.weight)
                return ConvT2dNHWCFn.apply(
                    xl, wl, self.bias, self.stride[0], self.padding[0]
                )
        return super().forward(x, output_size)
