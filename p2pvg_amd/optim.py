"""Optimizers.

The reference keeps five independent Adam instances inside the model
(reference models/p2p_model.py:51-57) — that contract is preserved. On gfx950
the step dispatches to the in-tree fused multi-tensor Adam HIP kernel
(SURVEY §2.6 K15); on CPU (or with P2PVG_KERNELS=torch) it is torch.optim.Adam
with foreach=True.
"""
from __future__ import annotations

import os
from typing import Iterable, Tuple

import torch


class HIPFusedAdam(torch.optim.Adam):
    """torch.optim.Adam whose step() runs the in-tree multi-tensor HIP kernel
    for CUDA(ROCm) params. State layout ('step', 'exp_avg', 'exp_avg_sq') is
    identical to torch.optim.Adam, so checkpoints are interchangeable."""

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        from . import ops as _ops

        ext = _ops._load_hip_ext()
        for group in self.param_groups:
            params, grads, exp_avgs, exp_avg_sqs, steps = [], [], [], [], []
            cpu_params = []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = torch.zeros((), dtype=torch.float32)
                    state["exp_avg"] = torch.zeros_like(p, memory_format=torch.preserve_format)
                    state["exp_avg_sq"] = torch.zeros_like(p, memory_format=torch.preserve_format)
                if p.is_cuda and ext is not None and p.dtype == torch.float32:
                    params.append(p)
                    grads.append(p.grad)
                    exp_avgs.append(state["exp_avg"])
                    exp_avg_sqs.append(state["exp_avg_sq"])
                    state["step"] += 1
                    steps.append(int(state["step"].item()) if state["step"].dim() == 0 else 0)
                else:
                    cpu_params.append(p)
            if params:
                beta1, beta2 = group["betas"]
                # all params in one group share the same step count in practice
                ext.multi_tensor_adam(
                    params, grads, exp_avgs, exp_avg_sqs,
                    group["lr"], beta1, beta2, group["eps"],
                    group["weight_decay"], steps[0],
                )
            if cpu_params:
                # fall back to the stock single-tensor path for these
                torch.optim.Adam.step(self, None)
                return loss
        return loss


def make_adam(params: Iterable, lr: float, betas: Tuple[float, float]):
    """Factory: fused HIP Adam on GPU when the extension is present, stock
    torch.optim.Adam otherwise."""
    params = list(params)
    use_hip = (
        os.environ.get("P2PVG_KERNELS", "auto") != "torch"
        and torch.cuda.is_available()
    )
    if use_hip:
        from . import ops as _ops

        if _ops.hip_available() and _ops.fused_adam_available():
            return HIPFusedAdam(params, lr=lr, betas=betas)
    return torch.optim.Adam(params, lr=lr, betas=betas, foreach=True)
