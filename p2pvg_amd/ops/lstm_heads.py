"""Autograd wrappers for the fused LSTM stack heads.

- Affine4Fn: the embed projection consuming [h | global | t | Δt] WITHOUT
  materializing the concat (reference models/p2p_model.py:241-247 cats, then
  models/lstm.py:13 runs the Linear).
- GaussHeadFn: mu/logvar heads + reparameterization in one kernel (the eps
  draw stays a torch philox op so hipGraph RNG capture semantics hold).
- TanhHeadFn: the predictor's Linear+Tanh output head.

All heads accumulate their weight/bias grads straight into the managed fp32
.grad buffers when available (same flow as the conv kernels: autograd's
per-use accumulation adds never dispatch), and compute in fp32 under
autocast (the recurrent path is fp32 throughout).
"""
from __future__ import annotations

import torch

from .conv import _acc_target, weight_grads_enabled


def _ext():
    from . import _hip_ext_loader

    return _hip_ext_loader.load()


class Affine4Fn(torch.autograd.Function):
    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.float32)
    def forward(ctx, h, g, s1, s2, w, b):
        h = h.contiguous()
        g = g.contiguous()
        s1 = s1.contiguous()
        s2 = s2.contiguous()
        out = _ext().affine4_fwd(h, g, s1, s2, w.contiguous(), b)
        ctx.save_for_backward(h, g, s1, s2, w)
        ctx.wref, ctx.bref = w, b
        return out

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, gout):
        h, g, s1, s2, w = ctx.saved_tensors
        gout = gout.contiguous()
        need_w = ctx.needs_input_grad[4] and weight_grads_enabled(ctx.wref)
        wg = _acc_target(ctx.wref) if need_w else None
        bg = _acc_target(ctx.bref) if need_w else None
        dw = db = None
        if need_w and (wg is None or (ctx.bref is not None and bg is None)):
            # unmanaged fallback: fresh zero buffers double as the grads
            wg = torch.zeros_like(w)
            bg = torch.zeros_like(ctx.bref) if ctx.bref is not None else None
            dw, db = wg, bg
        dh, dg = _ext().affine4_bwd(
            gout, h, g, s1, s2, w, wg if need_w else None,
            bg if need_w else None,
            ctx.needs_input_grad[0], ctx.needs_input_grad[1],
        )
        return (dh if ctx.needs_input_grad[0] else None,
                dg if ctx.needs_input_grad[1] else None,
                None, None, dw, db)


class GaussHeadFn(torch.autograd.Function):
    """ws/bs: optional per-step stacked-weight cache (rows [Wm; Wl]) owned by
    the module and refreshed once per step — when given, the forward skips
    its cat and the backward reuses the SAME stacked tensor (saved in ctx),
    so no weight re-stacking happens anywhere on the timestep loop."""

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.float32)
    def forward(ctx, hin, wm, bm, wl, bl, eps, ws=None, bs=None):
        hin = hin.contiguous()
        mu, lv, z, ws_used = _ext().gauss_head_fwd(
            hin, wm.contiguous(), bm.contiguous(), wl.contiguous(),
            bl.contiguous(), eps.contiguous(), ws, bs,
        )
        ctx.save_for_backward(hin, ws_used, eps, lv)
        ctx.refs = (wm, bm, wl, bl)
        return z, mu, lv

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, dz, dmu, dlv):
        hin, ws, eps, lv = ctx.saved_tensors
        wm_p, bm_p, wl_p, bl_p = ctx.refs
        need_w = weight_grads_enabled(wm_p) and (
            ctx.needs_input_grad[1] or ctx.needs_input_grad[3])
        accs = [(_acc_target(p) if need_w else None)
                for p in (wm_p, bm_p, wl_p, bl_p)]
        rets = [None, None, None, None]
        if need_w and any(a is None for a in accs):
            accs = [torch.zeros_like(p) for p in (wm_p, bm_p, wl_p, bl_p)]
            rets = accs
        dh = _ext().gauss_head_bwd(
            dz.contiguous() if dz is not None else None,
            dmu.contiguous() if dmu is not None else None,
            dlv.contiguous() if dlv is not None else None,
            eps, lv, hin, ws, wm_p.shape[0], *accs,
        )
        return (dh if ctx.needs_input_grad[0] else None,
                rets[0], rets[1], rets[2], rets[3], None, None, None)


class TanhHeadFn(torch.autograd.Function):
    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.float32)
    def forward(ctx, hin, w, b):
        hin = hin.contiguous()
        y = _ext().tanh_head_fwd(hin, w.contiguous(), b.contiguous())
        ctx.save_for_backward(hin, w, y)
        ctx.refs = (w, b)
        return y

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, dy):
        hin, w, y = ctx.saved_tensors
        w_p, b_p = ctx.refs
        need_w = weight_grads_enabled(w_p) and ctx.needs_input_grad[1]
        wg = _acc_target(w_p) if need_w else None
        bg = _acc_target(b_p) if need_w else None
        dw = db = None
        if need_w and (wg is None or bg is None):
            wg, bg = torch.zeros_like(w_p), torch.zeros_like(b_p)
            dw, db = wg, bg
        dh = _ext().tanh_head_bwd(dy.contiguous(), y, hin, w,
                                  wg if need_w else None,
                                  bg if need_w else None)
        return dh if ctx.needs_input_grad[0] else None, dw, db
