"""Optimizers.

The reference keeps five independent Adam instances inside the model
(reference models/p2p_model.py:51-57) — that contract is preserved. On gfx950
the step dispatches to the in-tree fused multi-tensor Adam HIP kernel
(one launch per group) with the step count held in DEVICE
memory, so the whole optimizer step is hipGraph-capturable and never syncs
the host. On CPU (or with P2PVG_KERNELS=torch) it is torch.optim.Adam.
"""
from __future__ import annotations

import os
from typing import Iterable, Tuple

import torch


class HIPFusedAdam(torch.optim.Optimizer):
    """Fused multi-tensor Adam on the in-tree gfx950 kernel.

    State layout mirrors torch.optim.Adam with capturable=True ('step' is a
    device fp32 scalar per param; 'exp_avg'/'exp_avg_sq' tensors), so
    checkpoints interchange with the stock optimizer.
    """

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
                        amsgrad=False, capturable=True, foreach=None,
                        maximize=False, differentiable=False, fused=None,
                        decoupled_weight_decay=False)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        from . import ops as _ops

        ext = _ops._load_hip_ext()
        assert ext is not None, "HIPFusedAdam requires the gfx950 extension"

        for group in self.param_groups:
            params, grads, ms, vs, steps = [], [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                assert p.is_cuda and p.dtype == torch.float32, (
                    "HIPFusedAdam: fp32 CUDA params only"
                )
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = torch.zeros((), dtype=torch.float32, device=p.device)
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                g = p.grad
                if g.stride() != p.stride():
                    # rare: autograd left a grad in a different dense layout
                    # (e.g. a transposed or channels-mismatched view) — copy
                    # into the param's layout so the fused kernel can walk
                    # param/grad/state with one index
                    g = torch.empty_like(p).copy_(g)
                params.append(p)
                grads.append(g)
                ms.append(state["exp_avg"])
                vs.append(state["exp_avg_sq"])
                steps.append(state["step"])
            if not params:
                continue
            # one device-side increment for all step scalars (capture-safe)
            torch._foreach_add_(steps, 1.0)
            beta1, beta2 = group["betas"]
            ext.multi_tensor_adam(
                params, grads, ms, vs, group["lr"], beta1, beta2,
                group["eps"], group["weight_decay"], steps[0],
            )
        return loss

    def load_state_dict(self, state_dict):
        super().load_state_dict(state_dict)
        # normalize 'step' entries (torch stores cpu scalars when
        # capturable=False) onto the param device as fp32
        for group in self.param_groups:
            for p in group["params"]:
                st = self.state.get(p)
                if st and "step" in st:
                    s = st["step"]
                    if not torch.is_tensor(s):
                        s = torch.tensor(float(s))
                    st["step"] = s.detach().to(
                        device=p.device, dtype=torch.float32
                    ).reshape(())


def make_adam(params: Iterable, lr: float, betas: Tuple[float, float],
              device: str = "cpu", capturable: bool = False):
    """Factory: fused HIP Adam on GPU when the extension is present, stock
    torch.optim.Adam otherwise. `device` is the device the model will run on
    (optimizers are created before model.to(device); .to moves params in
    place, so the references stay valid). `capturable` forces the stock Adam
    into its hipGraph-capturable mode (HIPFusedAdam always is)."""
    params = list(params)
    use_hip = (
        os.environ.get("P2PVG_KERNELS", "auto") != "torch"
        and torch.cuda.is_available()
        and str(device).startswith("cuda")
        and all(p.dtype == torch.float32 for p in params)
    )
    if use_hip:
        from . import ops as _ops

        if _ops.hip_available() and _ops.fused_adam_available():
            return HIPFusedAdam(params, lr=lr, betas=betas)
    return torch.optim.Adam(
        params, lr=lr, betas=betas, foreach=True,
        capturable=capturable and str(device).startswith("cuda"),
    )
