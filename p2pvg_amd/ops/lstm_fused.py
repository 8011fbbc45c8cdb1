"""Autograd wrapper for the fused gfx950 LSTM cell.

Forward is ONE kernel (both GEMMs + gates + state update,
csrc/lstm_cell.hip). Backward runs the fused pointwise gate-grad kernel and
four plain GEMMs (hipBLASLt via torch.mm — plain library GEMMs are the one
place library BLAS is used). The cell computes in fp32 regardless of autocast
(recurrent state stability); bf16 inputs are cast in.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch


def _ext():
    from . import _hip_ext_loader

    return _hip_ext_loader.load()


def lstm_cell_hip(x, hidden, w_ih, w_hh, b_ih, b_hh) -> Tuple[torch.Tensor, torch.Tensor]:
    """Inference-only fused path (no saved tensors)."""
    h, c = hidden
    x = x.float().contiguous()
    h_out, c_out, _ = _ext().lstm_cell_fwd(
        x, h.contiguous(), c.contiguous(), w_ih.contiguous(), w_hh.contiguous(),
        b_ih.contiguous(), b_hh.contiguous(),
    )
    return h_out, c_out


class LSTMCellFn(torch.autograd.Function):
    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.float32)
    def forward(ctx, x, h, c, w_ih, w_hh, b_ih, b_hh):
        x = x.contiguous()
        h = h.contiguous()
        c = c.contiguous()
        h_out, c_out, gates = _ext().lstm_cell_fwd(
            x, h, c, w_ih.contiguous(), w_hh.contiguous(), b_ih.contiguous(),
            b_hh.contiguous(),
        )
        ctx.save_for_backward(x, h, c, c_out, gates, w_ih, w_hh)
        return h_out, c_out

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, dh, dc):
        x, h, c_prev, c_new, gates, w_ih, w_hh = ctx.saved_tensors
        dh = dh.contiguous()
        dc = dc.contiguous() if dc is not None else None
        dgates, dc_prev = _ext().lstm_cell_bwd_pointwise(dh, dc, gates, c_prev, c_new)
        dx = dgates.mm(w_ih)
        dh_prev = dgates.mm(w_hh)
        dw_ih = dgates.t().mm(x)
        dw_hh = dgates.t().mm(h)
        db = dgates.sum(0)
        return dx, dh_prev, dc_prev, dw_ih, dw_hh, db, db.clone()
