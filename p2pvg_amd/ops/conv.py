"""Autograd integration of the gfx950 NHWC conv kernels.

Forward, dgrad and wgrad all run on the in-tree MFMA kernels:
- fwd: conv2d_nhwc_fwd (implicit GEMM, LDS-staged im2col; tile size and a
  small-C rsc-linear staging mode picked per shape).
- dgrad stride 1: the same fwd kernel on grad_out with flipped/transposed
  weights (pad = k-1-p).
- dgrad stride 2 / ConvTranspose fwd: conv2d_nhwc_fracstride — ONE launch,
  grid.z = parity, per-parity tap maps resolved in-kernel. No dilated
  intermediates, no transposes, no host-side weight slicing.
- wgrad: conv2d_nhwc_wgrad (pixel-major TN GEMM, split-K over pixel slabs,
  fp32 workspace accumulation).
- degenerate 1x1-output geometries (encoder tail k4s1p0 on 4x4 input,
  decoder head ConvTranspose 1x1 -> 4x4) are plain GEMMs in NHWC layout and
  dispatch to hipBLASLt matmul (library GEMMs are allowed exactly here).

Modules Conv2d / ConvTranspose2d subclass the torch ones (state_dict keys
unchanged) and route to this path for CUDA bf16/autocast inputs.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

CL = torch.channels_last

# Parameter-scoped weight-grad policy for the two-phase backward. The custom
# backward Functions accumulate weight grads by SIDE EFFECT whenever their
# node runs — but autograd's pruned traversal
# (`torch.autograd.backward(inputs=...)`) still visits nodes of the OTHER
# optimizer group when they sit on the path (phase 1 reaches the prior's
# embed on the way to the encoder; phase 2 reaches the decoder on the cpc
# path), and those groups' weight grads must NOT receive that phase's
# contributions (the reference discards them: models/p2p_model.py:266 +
# next-step zero_grad). Each phase therefore declares the param set to SKIP;
# this also saves the skipped kernels' work entirely.
_WG_SKIP = None  # Optional[frozenset[int]] of param ids to skip


class weight_grad_scope:
    """Context manager: suppress in-kernel weight/bias/BN-param gradient
    accumulation for the given parameters."""

    def __init__(self, params):
        self._ids = frozenset(id(p) for p in params)

    def __enter__(self):
        global _WG_SKIP
        self._prev = _WG_SKIP
        _WG_SKIP = self._ids
        return self

    def __exit__(self, *a):
        global _WG_SKIP
        _WG_SKIP = self._prev
        return False


class no_weight_grads:
    """Suppress ALL weight-grad work (kept for generation/eval paths)."""

    def __enter__(self):
        global _WG_SKIP
        self._prev = _WG_SKIP
        _WG_SKIP = _ALL_PARAMS
        return self

    def __exit__(self, *a):
        global _WG_SKIP
        _WG_SKIP = self._prev
        return False


class _AllParams(frozenset):
    def __contains__(self, item):  # noqa: D105
        return True


_ALL_PARAMS = _AllParams()


def weight_grads_enabled(p=None) -> bool:
    if _WG_SKIP is None:
        return True
    if isinstance(_WG_SKIP, _AllParams):
        return False
    return p is None or id(p) not in _WG_SKIP


def _acc_target(p):
    """The managed fp32 .grad buffer of a param, if accumulate-in-kernel is
    possible (buffer exists — materialized by P2PModel.zero_grads — dense
    fp32, matching element count). None -> caller falls back to returning
    the grad through autograd's accumulation."""
    if p is None or p.dtype != torch.float32:
        return None
    g = p.grad
    if g is None or g.dtype != torch.float32 or g.numel() != p.numel():
        return None
    return g


def _ext():
    from . import _hip_ext_loader

    return _hip_ext_loader.load()


def _to_cl_bf16(t: torch.Tensor) -> torch.Tensor:
    if t.dtype != torch.bfloat16:
        t = t.to(torch.bfloat16)
    return t.contiguous(memory_format=CL)


def _nhwc_flat(t: torch.Tensor) -> torch.Tensor:
    """(N,C,H,W) channels_last -> (N, H*W*C) view (no copy)."""
    n, c, h, w = t.shape
    return t.permute(0, 2, 3, 1).reshape(n, h * w * c)


def _act_fwd_torch(out, act):
    if act == 1:
        return torch.where(out > 0, out, out * 0.2)
    if act == 2:
        return torch.tanh(out)
    if act == 3:
        return torch.sigmoid(out)
    return out


def _act_bwd_from_y(dy, y, act):
    """grad through the fused epilogue activation, from the post-act value."""
    if act == 1:
        return torch.where(y > 0, dy, dy * 0.2)
    if act == 2:
        return dy * (1 - y.float() * y.float()).to(dy.dtype)
    if act == 3:
        yf = y.float()
        return dy * (yf * (1 - yf)).to(dy.dtype)
    return dy


def _gemm_stats(out2):
    f = out2.float()
    return torch.stack([f.sum(0), (f * f).sum(0)]).contiguous()


def _channel_sum(gout, acc=None):
    """db = sum over (N,H,W). ATen's strided (0,2,3) reduction on channels_last
    runs ~8x off bandwidth; use the NHWC channel-sum kernel when eligible.
    With acc (the bias .grad): accumulate in kernel, return None."""
    if (gout.is_cuda and gout.dtype == torch.bfloat16
            and gout.is_contiguous(memory_format=CL)):
        C = gout.shape[1]
        if C <= 4 or (C % 8 == 0 and 256 % (C // 8) == 0):
            if acc is not None:
                _ext().channel_sum_nhwc(gout, acc)
                return None
            return _ext().channel_sum_nhwc(gout)
    db = gout.sum(dim=(0, 2, 3), dtype=torch.float32)
    if acc is not None:
        acc.add_(db)
        return None
    return db


@torch.no_grad()
def _fill_conv_shadow(conv, sh):
    w = conv.weight
    if isinstance(conv, nn.ConvTranspose2d):
        sh["b"].copy_(w)  # plain (Ci,Co,k,k)
        if conv.stride[0] == 1:
            sh["f"].copy_(w.flip(2, 3).transpose(0, 1))
        else:
            sh["f"].copy_(w.transpose(0, 1))
    else:
        sh["f"].copy_(w)  # plain (K,C,k,k)
        if conv.stride[0] == 1:
            sh["b"].copy_(w.flip(2, 3).transpose(0, 1))
        else:
            sh["b"].copy_(w.transpose(0, 1))
    sh["ver"] = conv.weight._version


def _conv_shadows(conv):
    """Per-step bf16 weight shadows: 'f' = forward operand, 'b' = backward
    (dgrad / gemm) operand, both channels_last bf16 and refreshed ONCE per
    weight version instead of cast+flip+transposed at every one of the ~29
    timestep uses. refresh_conv_shadows() force-refreshes (needed inside a
    hipGraph capture, where replays never re-run this Python)."""
    sh = getattr(conv, "_p2pvg_shadow", None)
    if sh is None:
        w = conv.weight
        a, bdim, k, _ = w.shape
        if isinstance(conv, nn.ConvTranspose2d):
            fshape, bshape = (bdim, a, k, k), (a, bdim, k, k)
        else:
            fshape, bshape = (a, bdim, k, k), (bdim, a, k, k)
        mk = lambda s: torch.empty(  # noqa: E731
            s, device=w.device, dtype=torch.bfloat16, memory_format=CL
        )
        conv._p2pvg_shadow = sh = {"f": mk(fshape), "b": mk(bshape), "ver": -1}
    if sh["ver"] != conv.weight._version:
        _fill_conv_shadow(conv, sh)
    return sh


def refresh_conv_shadows(module):
    """Force-refresh every conv shadow under `module`. Call once per training
    step BEFORE the forward (recorded inside the hipGraph, so replays refresh
    too) and at the start of generation."""
    from . import backend_mode, hip_available

    if backend_mode() == "torch" or not hip_available():
        return
    for m in module.modules():
        if (isinstance(m, (nn.Conv2d, nn.ConvTranspose2d))
                and (isinstance(m, (Conv2d, ConvTranspose2d))
                     or hasattr(m, "_p2pvg_shadow"))
                and m.weight.is_cuda):
            sh = _conv_shadows(m)
            _fill_conv_shadow(m, sh)


class Conv2dNHWCFn(torch.autograd.Function):
    """y = act(conv2d(x, w, b, stride, pad)); optional per-channel sum/sumsq
    stats of the output (for the fused BatchNorm). Returns (y, stats)."""

    @staticmethod
    def forward(ctx, x, w, b, stride: int, pad: int, act: int = 0,
                want_stats: bool = False, w_fwd=None, w_bwd=None,
                in_ring: int = 0, out_ring: int = 0):
        # w receives the weight gradient. With w_fwd/w_bwd (per-step bf16
        # shadows, see refresh_conv_shadows), w is the raw fp32 param and
        # the per-use cast / flip / transpose kernels disappear; without
        # them w must already be bf16 channels_last (legacy direct calls).
        # in_ring: x is physically (H+2r, W+2r) with a ZERO ring realizing
        # the conv padding (must equal pad) — gathers become in-bounds and
        # the glds pipeline applies. out_ring: write the logical output as
        # the interior of a (HO+2r, WO+2r) map (ring left garbage; the
        # following bn_act writes the zeros).
        ext = _ext()
        wf = w_fwd if w_fwd is not None else w
        k = wf.shape[2]
        b32 = b.float() if b is not None else None
        stats = None
        assert in_ring in (0, pad), "in_ring must equal the conv padding"
        # degenerate whole-image conv (k == H, pad 0): plain GEMM
        gemm = pad == 0 and in_ring == 0 and out_ring == 0 \
            and k == x.shape[2] and k == x.shape[3]
        if gemm:
            out = torch.mm(_nhwc_flat(x), _nhwc_flat(wf).t())
            if b is not None:
                out = out + b.to(out.dtype)
            out = _act_fwd_torch(out, act)
            if want_stats:
                stats = _gemm_stats(out).unsqueeze(0)
            out = out.view(x.shape[0], wf.shape[0], 1, 1).contiguous(
                memory_format=CL
            )
            ctx.save_for_backward(x, wf, out if act != 0 else None)
            ctx.shadow_bwd = False
        else:
            oh = ow = oy = 0
            if out_ring:
                H = x.shape[2] - 2 * in_ring
                ho = (H + 2 * pad - k) // stride + 1
                oh = ow = ho + 2 * out_ring
                oy = out_ring
            out, stats = ext.conv2d_nhwc_fwd(
                x, wf, b32, stride, pad - in_ring, act, want_stats,
                oh, ow, oy, oy)
            wb = w_bwd if w_bwd is not None else wf
            ctx.save_for_backward(x, wb, out if act != 0 else None)
            ctx.shadow_bwd = w_bwd is not None
        ctx.with_shadows = w_fwd is not None or w_bwd is not None
        ctx.stride, ctx.pad, ctx.has_bias, ctx.gemm = stride, pad, b is not None, gemm
        ctx.act = act
        ctx.wdtype = w.dtype
        ctx.wref, ctx.bref = w, b
        ctx.rings = (in_ring, out_ring)
        if stats is None or not want_stats:
            stats = torch.empty(0, device=x.device)
        ctx.mark_non_differentiable(stats)
        return out, stats

    @staticmethod
    def backward(ctx, gout, _gstats):
        x, w, y = ctx.saved_tensors
        stride, pad = ctx.stride, ctx.pad
        k = w.shape[2]
        ext = _ext()
        gout = gout.contiguous(memory_format=CL)
        if gout.dtype != torch.bfloat16:
            gout = gout.to(torch.bfloat16)
        if ctx.act != 0:
            gout = _act_bwd_from_y(gout, y, ctx.act).contiguous(memory_format=CL)

        wge = weight_grads_enabled(ctx.wref)
        dx = dw = db = None
        if ctx.gemm:
            g2 = gout.reshape(gout.shape[0], gout.shape[1])  # (N, K)
            if ctx.needs_input_grad[0]:
                dx = torch.mm(g2, _nhwc_flat(w)).view(
                    x.shape[0], x.shape[2], x.shape[3], x.shape[1]
                ).permute(0, 3, 1, 2)
            if ctx.needs_input_grad[1] and wge:
                dw = torch.mm(g2.t(), _nhwc_flat(x)).view(
                    w.shape[0], w.shape[2], w.shape[3], w.shape[1]
                ).permute(0, 3, 1, 2)
                if dw.dtype != ctx.wdtype:
                    dw = dw.to(ctx.wdtype)
            if ctx.has_bias and ctx.needs_input_grad[2] and wge:
                db = g2.sum(0, dtype=torch.float32)
            return (dx, dw, db) + (None,) * 8

        in_ring, out_ring = ctx.rings
        if ctx.needs_input_grad[0]:
            # dx shape must equal x's (padded) shape; the ring is garbage
            # the interior-reading consumers (bn backward) ignore
            oh = x.shape[2] if in_ring else 0
            ow = x.shape[3] if in_ring else 0
            if stride == 1:
                wt = w if ctx.shadow_bwd else \
                    w.flip(2, 3).transpose(0, 1).contiguous(memory_format=CL)
                # gout's zero ring (out_ring) realizes that much of the
                # dgrad padding k-1-pad
                dx = ext.conv2d_nhwc_fwd(gout, wt, None, 1,
                                         k - 1 - pad - out_ring, 0, False,
                                         oh, ow, in_ring, in_ring)[0]
            else:
                wt = w if ctx.shadow_bwd else \
                    w.transpose(0, 1).contiguous(memory_format=CL)  # (C,K,k,k)
                dx = ext.conv2d_nhwc_fracstride(
                    gout, wt, None, stride, pad, x.shape[2] - 2 * in_ring,
                    x.shape[3] - 2 * in_ring, 0, False, out_ring, in_ring
                )[0]
        if ctx.needs_input_grad[1] and wge:
            wg = _acc_target(ctx.wref)
            if wg is not None and ctx.wdtype == torch.float32 \
                    and wg.is_contiguous(memory_format=CL):
                # accumulate straight into the managed fp32 .grad: autograd's
                # per-use AccumulateGrad adds never run for conv weights
                ext.conv2d_nhwc_wgrad(gout, x, k, k, stride, pad - in_ring,
                                      0, wg, out_ring)
            else:
                ws = ext.conv2d_nhwc_wgrad(gout, x, k, k, stride,
                                           pad - in_ring, 0, None, out_ring)
                if ctx.wdtype == torch.float32:
                    # (B,R,S,A) contiguous permuted to (K,C,k,k) IS the
                    # standard channels_last layout: the fp32 workspace is
                    # the grad, no bf16 round trip, no copy.
                    dw = ws.permute(0, 3, 1, 2)
                else:
                    dw = ws.permute(0, 3, 1, 2).to(torch.bfloat16) \
                        .contiguous(memory_format=CL)
        if ctx.has_bias and ctx.needs_input_grad[2] and wge:
            db = _channel_sum(gout, _acc_target(ctx.bref))
        return (dx, dw, db) + (None,) * 8


class CatConv2dFn(torch.autograd.Function):
    """y = conv2d(cat([x1, x2], 1), w) WITHOUT materializing the concat
    (SURVEY §2.6 K8): the glds conv gathers from two source tensors (the
    chunk's channel picks the pointer — every skip split is 64-aligned),
    its dgrad scatters into two grad tensors (dual-destination epilogue),
    and the wgrad gathers dual-X. Requires the padded-ring path (both
    inputs padded by the conv's padding) and the managed-grad flow for w."""

    @staticmethod
    def forward(ctx, x1, x2, w, stride: int, pad: int, act: int,
                want_stats: bool, w_fwd, w_bwd, ring: int, out_ring: int):
        ext = _ext()
        wf = w_fwd if w_fwd is not None else w
        k = wf.shape[2]
        assert ring == pad and ring > 0, "CatConv2dFn needs the padded path"
        oh = ow = oy = 0
        if out_ring:
            H = x1.shape[2] - 2 * ring
            ho = (H + 2 * pad - k) // stride + 1
            oh = ow = ho + 2 * out_ring
            oy = out_ring
        out, stats = ext.conv2d_glds_fwd(x1, wf, None, stride, act,
                                         want_stats, oh, ow, oy, oy, x2)
        wb = w_bwd if w_bwd is not None else wf
        ctx.save_for_backward(x1, x2, wb)
        ctx.shadow_bwd = w_bwd is not None
        ctx.meta = (stride, pad, act, ring, out_ring, k)
        ctx.wref = w
        ctx.wdtype = w.dtype
        if stats is None or not want_stats:
            stats = torch.empty(0, device=x1.device)
        ctx.mark_non_differentiable(stats)
        return out, stats

    @staticmethod
    def backward(ctx, gout, _gstats):
        x1, x2, w = ctx.saved_tensors
        stride, pad, act, ring, out_ring, k = ctx.meta
        ext = _ext()
        gout = gout.contiguous(memory_format=CL)
        if gout.dtype != torch.bfloat16:
            gout = gout.to(torch.bfloat16)
        wge = weight_grads_enabled(ctx.wref)
        dx1 = dx2 = dw = None
        if ctx.needs_input_grad[0] or ctx.needs_input_grad[1]:
            wt = w if ctx.shadow_bwd else \
                w.flip(2, 3).transpose(0, 1).contiguous(memory_format=CL)
            dx2 = torch.empty_like(x2)
            dx1, _ = ext.conv2d_glds_fwd(gout, wt, None, 1,
                                         0, False, x1.shape[2], x1.shape[3],
                                         ring, ring, None, dx2)
        if ctx.needs_input_grad[2] and wge:
            wg = _acc_target(ctx.wref)
            if wg is not None and ctx.wdtype == torch.float32 \
                    and wg.is_contiguous(memory_format=CL):
                ext.conv2d_nhwc_wgrad(gout, x1, k, k, stride, 0, 0, wg,
                                      out_ring, x2)
            else:
                ws = ext.conv2d_nhwc_wgrad(gout, x1, k, k, stride, 0, 0,
                                           None, out_ring, x2)
                dw = ws.permute(0, 3, 1, 2)
                if ctx.wdtype != torch.float32:
                    dw = dw.to(torch.bfloat16).contiguous(memory_format=CL)
        return (dx1, dx2, dw) + (None,) * 8


def cat_conv_eligible(x1, x2, conv) -> bool:
    """Can conv consume (x1, x2) via the dual-pointer glds path?"""
    ring = getattr(x1, "_pvg_pad", 0)
    if ring == 0 or ring != getattr(x2, "_pvg_pad", 0):
        return False
    if not isinstance(conv, nn.Conv2d) or isinstance(conv, nn.ConvTranspose2d):
        return False
    C = x1.shape[1] + x2.shape[1]
    H = x1.shape[2] - 2 * ring
    W = x1.shape[3] - 2 * ring
    pow2 = H > 0 and (H & (H - 1)) == 0 and (W & (W - 1)) == 0
    return (conv.kernel_size[0] == 3 and conv.stride[0] == 1
            and conv.padding[0] == ring and x1.shape[1] % 64 == 0
            and x2.shape[1] % 64 == 0 and conv.out_channels % 64 == 0
            and conv.in_channels == C and x1.shape[2:] == x2.shape[2:]
            and pow2)   # the dual-X wgrad needs the shift-only pixel decode


class ConvT2dNHWCFn(torch.autograd.Function):
    """y = conv_transpose2d(x, w, b, stride, pad); w logical (Ci, Co, k, k)."""

    @staticmethod
    def forward(ctx, x, w, b, stride: int, pad: int, act: int = 0,
                want_stats: bool = False, w_fwd=None, w_bwd=None,
                in_ring: int = 0, out_ring: int = 0):
        # With shadows: w is the raw fp32 param (receives the grad), w_fwd is
        # the pre-flipped/transposed (Co,Ci,k,k) bf16 forward form, w_bwd the
        # plain-cast (Ci,Co,k,k) bf16 form (dgrad + gemm operand).
        # in_ring: x physically padded with a zero ring (stride-1 convT:
        # must equal k-1-pad, the flipped conv's padding); out_ring: write
        # the output as the interior of a padded map.
        ext = _ext()
        wp = w_bwd if w_bwd is not None else w  # plain (Ci,Co,k,k) bf16 form
        k = wp.shape[2]
        co = wp.shape[1]
        n, _, hp, wdtp = x.shape
        h, wdt = hp - 2 * in_ring, wdtp - 2 * in_ring
        b32 = b.float() if b is not None else None
        stats = None
        gemm = stride == 1 and pad == 0 and in_ring == 0 and out_ring == 0 \
            and h == 1 and wdt == 1
        if gemm:
            # 1x1 -> kxk: out[n, y, x, co] = sum_ci in[n,ci] w[ci,co,y,x].
            # The channels_last (Ci,Co,k,k) weight is physically (Ci,k,k,Co),
            # so its NHWC flattening is already (Ci, [y,x,co]).
            out = torch.mm(_nhwc_flat(x), _nhwc_flat(wp))
            if b is not None:
                out = out.view(n, k * k, co) + b.to(out.dtype)
            out = _act_fwd_torch(out, act)
            if want_stats:
                stats = _gemm_stats(out.reshape(n * k * k, co)).unsqueeze(0)
            out = out.view(n, k, k, co).permute(0, 3, 1, 2)
        elif stride == 1:
            assert in_ring in (0, k - 1 - pad), "convT in_ring mismatch"
            wt = w_fwd if w_fwd is not None else \
                w.flip(2, 3).transpose(0, 1).contiguous(memory_format=CL)  # (Co,Ci,k,k)
            ho = h + k - 1 - 2 * pad
            oh = ho + 2 * out_ring if out_ring else 0
            out, stats = ext.conv2d_nhwc_fwd(x, wt, b32, 1,
                                             k - 1 - pad - in_ring, act,
                                             want_stats, oh, oh,
                                             out_ring, out_ring)
        else:
            oh = (h - 1) * stride - 2 * pad + k
            ow = (wdt - 1) * stride - 2 * pad + k
            wt = w_fwd if w_fwd is not None else \
                w.transpose(0, 1).contiguous(memory_format=CL)  # (Co, Ci, k, k)
            out, stats = ext.conv2d_nhwc_fracstride(x, wt, b32, stride, pad,
                                                    oh, ow, act, want_stats,
                                                    in_ring, out_ring)
        ctx.save_for_backward(x, wp, out if act != 0 else None)
        ctx.shadow_bwd = w_bwd is not None
        ctx.with_shadows = w_fwd is not None or w_bwd is not None
        ctx.stride, ctx.pad, ctx.has_bias, ctx.gemm = stride, pad, b is not None, gemm
        ctx.act = act
        ctx.wdtype = w.dtype
        ctx.wref, ctx.bref = w, b
        ctx.rings = (in_ring, out_ring)
        if stats is None or not want_stats:
            stats = torch.empty(0, device=x.device)
        ctx.mark_non_differentiable(stats)
        return out, stats

    @staticmethod
    def backward(ctx, gout, _gstats):
        x, w, y = ctx.saved_tensors
        stride, pad = ctx.stride, ctx.pad
        k = w.shape[2]
        ext = _ext()
        gout = gout.contiguous(memory_format=CL)
        if gout.dtype != torch.bfloat16:
            gout = gout.to(torch.bfloat16)
        if ctx.act != 0:
            gout = _act_bwd_from_y(gout, y, ctx.act).contiguous(memory_format=CL)

        wge = weight_grads_enabled(ctx.wref)
        dx = dw = db = None
        if ctx.gemm:
            n = x.shape[0]
            g2 = _nhwc_flat(gout)  # (N, k*k*Co)
            wf = _nhwc_flat(w)     # (Ci, k*k*Co)
            if ctx.needs_input_grad[0]:
                dx = torch.mm(g2, wf.t()).view(n, 1, 1, x.shape[1]).permute(0, 3, 1, 2)
            if ctx.needs_input_grad[1] and wge:
                # dwf (Ci, k*k*Co) -> (Ci, k, k, Co) physical = CL (Ci,Co,k,k)
                dwf = torch.mm(_nhwc_flat(x).t(), g2)
                dw = dwf.view(x.shape[1], k, k, w.shape[1]).permute(0, 3, 1, 2)
                if dw.dtype != ctx.wdtype:
                    dw = dw.to(ctx.wdtype)
            if ctx.has_bias and ctx.needs_input_grad[2] and wge:
                db = gout.sum(dim=(0, 2, 3), dtype=torch.float32)
            return (dx, dw, db) + (None,) * 8

        in_ring, out_ring = ctx.rings
        if ctx.needs_input_grad[0]:
            # dgrad of convT = plain conv with the untransposed weight;
            # gout's zero ring realizes that much of the conv padding, dx
            # is written as the interior of x's padded shape (garbage ring)
            wl = w if ctx.shadow_bwd else w.contiguous(memory_format=CL)
            oh = x.shape[2] if in_ring else 0
            ow = x.shape[3] if in_ring else 0
            dx = ext.conv2d_nhwc_fwd(gout, wl, None, stride, pad - out_ring,
                                     0, False, oh, ow, in_ring, in_ring)[0]
        if ctx.needs_input_grad[1] and wge:
            wg = _acc_target(ctx.wref)
            if wg is not None and ctx.wdtype == torch.float32 \
                    and wg.is_contiguous(memory_format=CL):
                ext.conv2d_nhwc_wgrad(x, gout, k, k, stride, pad - out_ring,
                                      0, wg, in_ring)
            else:
                ws = ext.conv2d_nhwc_wgrad(x, gout, k, k, stride,
                                           pad - out_ring, 0, None, in_ring)
                if ctx.wdtype == torch.float32:
                    dw = ws.permute(0, 3, 1, 2)
                else:
                    dw = ws.permute(0, 3, 1, 2).to(torch.bfloat16) \
                        .contiguous(memory_format=CL)
        if ctx.has_bias and ctx.needs_input_grad[2] and wge:
            db = _channel_sum(gout, _acc_target(ctx.bref))
        return (dx, dw, db) + (None,) * 8


def interior_view(t: torch.Tensor) -> torch.Tensor:
    """Logical (interior) view of a padded zero-ring map; identity on dense
    tensors. Safety net for padded tensors reaching consumers that don't
    speak the ring protocol."""
    p = getattr(t, "_pvg_pad", 0)
    return t[:, :, p:-p, p:-p] if p else t


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
            in_ring = getattr(x, "_pvg_pad", 0)
            if in_ring and in_ring != self.padding[0]:
                x = interior_view(x)
                in_ring = 0
            with torch.autocast("cuda", enabled=False):
                xl = _to_cl_bf16(x)
                sh = _conv_shadows(self)
                out, _ = Conv2dNHWCFn.apply(
                    xl, self.weight, self.bias, self.stride[0],
                    self.padding[0], 0, False, sh["f"], sh["b"], in_ring, 0
                )
                return out
        return super().forward(interior_view(x))


class ConvTranspose2d(nn.ConvTranspose2d):
    """nn.ConvTranspose2d on the gfx950 fracstride kernels."""

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
            k = self.kernel_size[0]
            want = k - 1 - self.padding[0] if self.stride[0] == 1 else \
                self.padding[0]
            in_ring = getattr(x, "_pvg_pad", 0)
            if in_ring and (self.stride[0] != 1 or in_ring != want):
                x = interior_view(x)
                in_ring = 0
            with torch.autocast("cuda", enabled=False):
                xl = _to_cl_bf16(x)
                sh = _conv_shadows(self)
                out, _ = ConvT2dNHWCFn.apply(
                    xl, self.weight, self.bias, self.stride[0],
                    self.padding[0], 0, False, sh["f"], sh["b"], in_ring, 0
                )
                return out
        return super().forward(interior_view(x), output_size)
