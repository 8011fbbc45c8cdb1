"""Recurrent stacks: `lstm` and `gaussian_lstm`.

Same capability as the reference stacks (reference models/lstm.py:5-94):
embed Linear -> N x LSTMCell -> output head (Linear+Tanh, or mu/logvar heads with
reparameterization). Differences by design:

- Device-agnostic: hidden state is created on the input's device/dtype at
  `init_hidden` time. The reference hard-codes `.cuda()` at construction
  (reference models/lstm.py:24-25,58,63-64), which makes CPU runs impossible;
  we do not.
- The per-timestep cell math dispatches to the fused HIP cell kernel on gfx950
  (p2pvg_amd.ops.lstm_cell) and to torch.nn.functional on CPU.
- Hidden state is module-owned and re-initialized per sequence, exactly like the
  reference.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from .. import ops


class LSTMStack(nn.Module):
    """Shared base: embed -> n_layers x LSTMCell."""

    def __init__(self, input_size: int, hidden_size: int, n_layers: int):
        super().__init__()
        self.input_size = input_size
        self.hidden_size = hidden_size
        self.n_layers = n_layers
        self.embed = nn.Linear(input_size, hidden_size)
        self.lstm = nn.ModuleList(
            [nn.LSTMCell(hidden_size, hidden_size) for _ in range(n_layers)]
        )
        self.hidden: Optional[List[Tuple[torch.Tensor, torch.Tensor]]] = None

    def init_hidden(self, batch_size: int = 1, device=None, dtype=None):
        p = next(self.parameters())
        device = device if device is not None else p.device
        dtype = dtype if dtype is not None else p.dtype
        self.hidden = [
            (
                torch.zeros(batch_size, self.hidden_size, device=device, dtype=dtype),
                torch.zeros(batch_size, self.hidden_size, device=device, dtype=dtype),
            )
            for _ in range(self.n_layers)
        ]
        return self.hidden

    def _fused_heads(self, x) -> bool:
        import os

        t = x[0] if isinstance(x, tuple) else x
        # dispatch policy: the fused heads win in the LATENCY-bound regime
        # (small batch, replay-gap dominated); above ~512 rows the hipBLASLt
        # GEMM heads are faster (measured: h36m batch 1024 -21% fused,
        # profiles/MEASUREMENTS.md), so large batches keep the library GEMMs
        return (t.is_cuda and t.shape[0] <= 512
                and ops.backend_mode() != "torch"
                and os.environ.get("P2PVG_LSTM_HEADS", "1") != "0"
                and ops.hip_available())

    def _embed_in(self, x) -> torch.Tensor:
        if isinstance(x, tuple):
            # concat-free embed: the kernel gathers [h | global | t | dt]
            # from the four sources directly (SURVEY K9+K10)
            if self._fused_heads(x):
                from ..ops.lstm_heads import Affine4Fn

                h, g, s1, s2 = x
                return Affine4Fn.apply(h, g, s1, s2, self.embed.weight,
                                       self.embed.bias)
            x = torch.cat(list(x), 1)
        return self.embed(x.view(-1, self.input_size))

    def _run_cells(self, x) -> torch.Tensor:
        assert self.hidden is not None, "call init_hidden() before forward()"
        h_in = self._embed_in(x)
        for i, cell in enumerate(self.lstm):
            self.hidden[i] = ops.lstm_cell(
                h_in,
                self.hidden[i],
                cell.weight_ih,
                cell.weight_hh,
                cell.bias_ih,
                cell.bias_hh,
            )
            h_in = self.hidden[i][0]
        return h_in


class lstm(LSTMStack):
    """Deterministic stack with Linear+Tanh output head (reference models/lstm.py:5-44)."""

    def __init__(self, input_size, output_size, hidden_size, n_layers, batch_size=None):
        super().__init__(input_size, hidden_size, n_layers)
        self.output_size = output_size
        self.batch_size = batch_size  # kept for checkpoint/API parity; unused
        self.output = nn.Sequential(nn.Linear(hidden_size, output_size), nn.Tanh())

    def forward(self, x) -> torch.Tensor:
        h = self._run_cells(x)
        if self._fused_heads(x):
            from ..ops.lstm_heads import TanhHeadFn

            return TanhHeadFn.apply(h, self.output[0].weight,
                                    self.output[0].bias)
        return self.output(h)


class gaussian_lstm(LSTMStack):
    """Stochastic stack: mu/logvar heads + reparameterization
    (reference models/lstm.py:46-94)."""

    def __init__(self, input_size, output_size, hidden_size, n_layers, batch_size=None):
        super().__init__(input_size, hidden_size, n_layers)
        self.output_size = output_size
        self.batch_size = batch_size
        self.mu_net = nn.Linear(hidden_size, output_size)
        self.logvar_net = nn.Linear(hidden_size, output_size)
        # per-step stacked-weight cache [Wm; Wl] for the fused head: built
        # lazily, refreshed once per init_hidden (i.e. once per training
        # step / generation sequence) instead of re-catting per timestep
        self._stk: Optional[Tuple[torch.Tensor, torch.Tensor]] = None
        self._stk_dirty = True

    def init_hidden(self, batch_size: int = 1, device=None, dtype=None):
        self._stk_dirty = True
        return super().init_hidden(batch_size, device, dtype)

    def _stacked(self) -> Tuple[torch.Tensor, torch.Tensor]:
        n = self.output_size
        w = self.mu_net.weight
        if self._stk is None or self._stk[0].device != w.device:
            self._stk = (
                torch.empty(2 * n, w.shape[1], device=w.device, dtype=torch.float32),
                torch.empty(2 * n, device=w.device, dtype=torch.float32),
            )
            self._stk_dirty = True
        if self._stk_dirty:
            ws, bs = self._stk
            with torch.no_grad():  # grads route via the managed accumulation
                ws[:n].copy_(self.mu_net.weight)
                ws[n:].copy_(self.logvar_net.weight)
                bs[:n].copy_(self.mu_net.bias)
                bs[n:].copy_(self.logvar_net.bias)
            self._stk_dirty = False
        return self._stk

    def reparameterize(self, mu: torch.Tensor, logvar: torch.Tensor) -> torch.Tensor:
        std = logvar.mul(0.5).exp()
        eps = torch.randn_like(std)
        return eps * std + mu

    def forward(self, x):
        h = self._run_cells(x)
        if self._fused_heads(x):
            from ..ops.lstm_heads import GaussHeadFn

            eps = torch.randn(h.shape[0], self.output_size, device=h.device,
                              dtype=torch.float32)
            ws, bs = self._stacked()
            return GaussHeadFn.apply(h, self.mu_net.weight, self.mu_net.bias,
                                     self.logvar_net.weight,
                                     self.logvar_net.bias, eps, ws, bs)
        mu = self.mu_net(h)
        logvar = self.logvar_net(h)
        z = self.reparameterize(mu, logvar)
        return z, mu, logvar
