"""Quantitative metrics: MSE / PSNR / SSIM, end-frame SSIM.

The reference imports skimage's compare_psnr/compare_ssim but never calls them
(reference misc/metrics.py:1-16, a stub); end-frame SSIM is the BASELINE
quality metric, so it is implemented here for real, in torch (works on CPU and
GPU, batched).
"""
from __future__ import annotations

import math

import torch
import torch.nn.functional as F


def mse(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    return ((a - b) ** 2).flatten(1).mean(dim=1)


def psnr(a: torch.Tensor, b: torch.Tensor, data_range: float = 1.0) -> torch.Tensor:
    m = mse(a, b).clamp_min(1e-12)
    return 10.0 * torch.log10(data_range**2 / m)


def _gaussian_window(size: int, sigma: float, device, dtype):
    coords = torch.arange(size, device=device, dtype=dtype) - (size - 1) / 2.0
    g = torch.exp(-(coords**2) / (2 * sigma**2))
    g = g / g.sum()
    return g.outer(g)


def ssim(
    a: torch.Tensor,
    b: torch.Tensor,
    data_range: float = 1.0,
    window_size: int = 11,
    sigma: float = 1.5,
) -> torch.Tensor:
    """Mean SSIM per batch element. a, b: (B,C,H,W) in [0, data_range].

    Standard Wang et al. formulation: 11x11 gaussian window (sigma 1.5),
    K1=0.01, K2=0.03, per-channel then averaged.
    """
    assert a.shape == b.shape and a.dim() == 4
    B, C, H, W = a.shape
    win = _gaussian_window(window_size, sigma, a.device, torch.float32)
    win = win.expand(C, 1, window_size, window_size).contiguous()
    a = a.float()
    b = b.float()

    pad = window_size // 2
    mu_a = F.conv2d(a, win, padding=pad, groups=C)
    mu_b = F.conv2d(b, win, padding=pad, groups=C)
    mu_a2, mu_b2, mu_ab = mu_a * mu_a, mu_b * mu_b, mu_a * mu_b
    sigma_a2 = F.conv2d(a * a, win, padding=pad, groups=C) - mu_a2
    sigma_b2 = F.conv2d(b * b, win, padding=pad, groups=C) - mu_b2
    sigma_ab = F.conv2d(a * b, win, padding=pad, groups=C) - mu_ab

    c1 = (0.01 * data_range) ** 2
    c2 = (0.03 * data_range) ** 2
    ssim_map = ((2 * mu_ab + c1) * (2 * sigma_ab + c2)) / (
        (mu_a2 + mu_b2 + c1) * (sigma_a2 + sigma_b2 + c2)
    )
    return ssim_map.flatten(1).mean(dim=1)


def end_frame_ssim(gen_seq, target_frame: torch.Tensor) -> torch.Tensor:
    """SSIM between the generated end frame and the control-point frame.

    gen_seq: list of (B,C,H,W) frames (p2p_generate output) or (T,B,C,H,W).
    """
    if isinstance(gen_seq, (list, tuple)):
        end = gen_seq[-1]
    else:
        end = gen_seq[-1]
    return ssim(end, target_frame)


class Metric:
    """Reference-named facade (reference misc/metrics.py:11) with the metrics
    actually implemented."""

    @staticmethod
    def compare_mse(x1, x2):
        return mse(x1, x2)

    @staticmethod
    def compare_psnr(x1, x2, data_range: float = 1.0):
        return psnr(x1, x2, data_range)

    @staticmethod
    def compare_ssim(x1, x2, data_range: float = 1.0):
        return ssim(x1, x2, data_range)
