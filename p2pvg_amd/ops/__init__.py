"""Hot-op dispatch: hand-written gfx950 HIP kernels on GPU, PyTorch on CPU.

Every op the model launches per timestep (SURVEY §2.6 kernel inventory) routes
through this module. Dispatch policy:

- On a ROCm GPU (`tensor.is_cuda`), the in-tree HIP extension (`p2pvg_amd/ops/_hip`)
  is REQUIRED for ops it implements: if the .so failed to build/load, the op
  raises rather than silently falling back to ATen — a GPU run must exercise the
  native path. Set P2PVG_KERNELS=torch (or Config.kernels="torch") to explicitly
  allow the ATen path for A/B comparison.
- On CPU, ops use the plain PyTorch fp32 reference implementation, which is also
  the numerics oracle the GPU kernels are tested against (tests/test_ops_gpu.py).
"""
from __future__ import annotations

import os
from typing import Optional, Tuple

import torch

_HIP_EXT = None
_HIP_TRIED = False


def _load_hip_ext():
    """Load the in-tree HIP extension if present (built by setup_ext.py / __graft_entry__.build)."""
    global _HIP_EXT, _HIP_TRIED
    if _HIP_TRIED:
        return _HIP_EXT
    _HIP_TRIED = True
    try:
        from . import _hip_ext_loader

        _HIP_EXT = _hip_ext_loader.load()
    except Exception as e:  # noqa: BLE001
        _HIP_EXT = None
        _load_hip_ext.error = e  # type: ignore[attr-defined]
    return _HIP_EXT


def backend_mode() -> str:
    """auto | hip | torch — process-wide kernel backend policy."""
    return os.environ.get("P2PVG_KERNELS", "auto")


def hip_available() -> bool:
    return _load_hip_ext() is not None


def frame_mse(a, b):
    """MSE over frame tensors: fused one-pass reduction on the HIP path
    (large bf16 tensors), F.mse_loss otherwise."""
    from .losses import fused_mse

    return fused_mse(a, b)


def _want_hip(t: torch.Tensor) -> bool:
    mode = backend_mode()
    if mode == "torch":
        return False
    if not t.is_cuda:
        return False
    ext = _load_hip_ext()
    if ext is None:
        if mode in ("hip", "auto"):
            err = getattr(_load_hip_ext, "error", None)
            raise RuntimeError(
                "p2pvg_amd HIP extension is not available on a GPU device "
                f"(build it with `python setup_ext.py build_ext --inplace`): {err!r}. "
                "Set P2PVG_KERNELS=torch to explicitly run the ATen fallback."
            )
        return False
    return True


# ---------------------------------------------------------------------------
# LSTM cell: x(B,H) h,c(B,H) -> fused gates + state update
# ---------------------------------------------------------------------------

def lstm_cell(
    x: torch.Tensor,
    hidden: Tuple[torch.Tensor, torch.Tensor],
    w_ih: torch.Tensor,
    w_hh: torch.Tensor,
    b_ih: Optional[torch.Tensor],
    b_hh: Optional[torch.Tensor],
) -> Tuple[torch.Tensor, torch.Tensor]:
    if _want_hip(x) and not torch.is_grad_enabled():
        from .lstm_fused import lstm_cell_hip

        return lstm_cell_hip(x, hidden, w_ih, w_hh, b_ih, b_hh)
    if _want_hip(x):
        from .lstm_fused import LSTMCellFn

        return LSTMCellFn.apply(x, hidden[0], hidden[1], w_ih, w_hh, b_ih, b_hh)
    return torch._VF.lstm_cell(x, hidden, w_ih, w_hh, b_ih, b_hh)


# ---------------------------------------------------------------------------
# Gaussian KL: closed-form KL(N1 || N2), sum / batch_size_cfg
# ---------------------------------------------------------------------------

def gaussian_kl(
    mu1: torch.Tensor,
    logvar1: torch.Tensor,
    mu2: torch.Tensor,
    logvar2: torch.Tensor,
    denom: float,
) -> torch.Tensor:
    """KL(N(mu1, e^lv1) || N(mu2, e^lv2)).sum() / denom.

    Matches the reference closed form (reference misc/criterion.py:10-15), which
    normalizes by the CONFIGURED batch size, not the runtime batch.
    """
    if _want_hip(mu1) and torch.is_grad_enabled():
        from .losses import GaussianKLFn

        return GaussianKLFn.apply(mu1, logvar1, mu2, logvar2, denom)
    kld = 0.5 * (logvar2 - logvar1) + (torch.exp(logvar1) + (mu1 - mu2) ** 2) / (
        2.0 * torch.exp(logvar2)
    ) - 0.5
    return kld.sum() / denom


# ---------------------------------------------------------------------------
# Fused multi-tensor Adam
# ---------------------------------------------------------------------------

def fused_adam_available() -> bool:
    ext = _load_hip_ext()
    return ext is not None and hasattr(ext, "multi_tensor_adam")


__all__ = [
    "lstm_cell",
    "gaussian_kl",
    "backend_mode",
    "hip_available",
    "fused_adam_available",
]
