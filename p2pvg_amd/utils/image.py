"""Image grid / PNG / GIF helpers (torchvision- and imageio-free).

Replaces the reference's uses of torchvision.utils.make_grid/save_image and
imageio.mimsave (reference misc/visualize.py:10, misc/utils.py:118-127) with
self-contained implementations on PIL.
"""
from __future__ import annotations

import math
from typing import List, Sequence

import numpy as np
import torch


def make_grid(
    tensor: torch.Tensor, nrow: int = 8, padding: int = 2, pad_value: float = 0.0
) -> torch.Tensor:
    """Arrange a (N,C,H,W) batch into a (C, gh, gw) grid, nrow images per row."""
    if tensor.dim() == 3:
        tensor = tensor.unsqueeze(0)
    n, c, h, w = tensor.shape
    if c == 1:
        tensor = tensor.expand(n, 3, h, w)
        c = 3
    ncols = min(nrow, n)
    nrows = int(math.ceil(n / ncols))
    gh = nrows * h + padding * (nrows + 1)
    gw = ncols * w + padding * (ncols + 1)
    grid = tensor.new_full((c, gh, gw), pad_value)
    k = 0
    for r in range(nrows):
        for cidx in range(ncols):
            if k >= n:
                break
            y = padding + r * (h + padding)
            x = padding + cidx * (w + padding)
            grid[:, y : y + h, x : x + w] = tensor[k]
            k += 1
    return grid


def to_uint8_hwc(img: torch.Tensor) -> np.ndarray:
    """(C,H,W) float [0,1] -> (H,W,C) uint8."""
    arr = (img.detach().float().cpu().clamp(0, 1).numpy() * 255.0).astype(np.uint8)
    return np.transpose(arr, (1, 2, 0))


def save_image(tensor: torch.Tensor, fname: str) -> None:
    from PIL import Image

    arr = to_uint8_hwc(tensor)
    if arr.shape[2] == 1:
        arr = arr[:, :, 0]
    Image.fromarray(arr).save(fname)


def save_gif(fname: str, frames: Sequence[np.ndarray], duration: float = 0.25) -> None:
    """frames: list of (H,W,C) uint8 arrays; C=1 saved as grayscale."""
    from PIL import Image

    frames = [f[:, :, 0] if f.ndim == 3 and f.shape[2] == 1 else f
              for f in frames]
    imgs: List = [Image.fromarray(f) for f in frames]
    imgs[0].save(
        fname,
        save_all=True,
        append_images=imgs[1:],
        duration=int(duration * 1000),
        loop=0,
    )
