"""Logging / experiment bookkeeping.

Capability parity with reference misc/utils.py:211-252 (console+file logger,
cmd.txt reproduction file) plus a tensorboard-free scalar writer: scalars go
to <log_dir>/scalars.jsonl (and to torch.utils.tensorboard if that package
happens to be importable). Histograms are summarized (mean/std/min/max)
instead of full DtoH dumps, and only when enabled (SURVEY §5.5 notes the
reference's per-50-iter full param/grad DtoH as a major sync).
"""
from __future__ import annotations

import json
import logging
import os
import sys
import time
from typing import Optional

import torch


def get_logger(logpath: str, name: str = "p2pvg_amd", displaying=True, saving=True):
    logger = logging.getLogger(name)
    logger.setLevel(logging.INFO)
    logger.handlers.clear()
    logger.propagate = False
    fmt = logging.Formatter("%(asctime)s %(message)s", "%H:%M:%S")
    if saving:
        os.makedirs(os.path.dirname(logpath) or ".", exist_ok=True)
        fh = logging.FileHandler(logpath, mode="a")
        fh.setFormatter(fmt)
        logger.addHandler(fh)
    if displaying:
        sh = logging.StreamHandler()
        sh.setFormatter(fmt)
        logger.addHandler(sh)
    return logger


def store_cmd(log_dir: str) -> str:
    """Write the reproduction command to <log_dir>/cmd.txt and snapshot the
    invoking script's source next to it (reference misc/utils.py:227-252
    stores both the command line and the training source for provenance)."""
    cmd = "python " + " ".join(sys.argv)
    os.makedirs(log_dir, exist_ok=True)
    with open(os.path.join(log_dir, "cmd.txt"), "w") as f:
        f.write(cmd + "\n")
    script = sys.argv[0] if sys.argv else ""
    if script and os.path.isfile(script):
        import shutil

        dst = os.path.join(log_dir, "src")
        os.makedirs(dst, exist_ok=True)
        shutil.copy2(script, os.path.join(dst, os.path.basename(script)))
    return cmd


class ScalarWriter:
    """JSONL scalar/image/video logger with optional tensorboard passthrough."""

    def __init__(self, log_dir: str):
        self.log_dir = log_dir
        os.makedirs(log_dir, exist_ok=True)
        self._f = open(os.path.join(log_dir, "scalars.jsonl"), "a")
        self._tb = None
        try:
            from torch.utils.tensorboard import SummaryWriter

            self._tb = SummaryWriter(log_dir=os.path.join(log_dir, "tboard"))
        except Exception:  # noqa: BLE001
            pass

    def add_scalar(self, tag: str, value, step: int):
        if torch.is_tensor(value):
            value = value.item()
        self._f.write(json.dumps({"t": time.time(), "tag": tag, "v": float(value), "step": step}) + "\n")
        self._f.flush()
        if self._tb is not None:
            self._tb.add_scalar(tag, value, step)

    def add_histogram_summary(self, tag: str, tensor: torch.Tensor, step: int):
        with torch.no_grad():
            t = tensor.detach().float()
            stats = {
                "mean": t.mean().item(),
                "std": t.std().item() if t.numel() > 1 else 0.0,
                "min": t.min().item(),
                "max": t.max().item(),
            }
        self._f.write(json.dumps({"t": time.time(), "tag": tag, "hist": stats, "step": step}) + "\n")
        if self._tb is not None:
            self._tb.add_histogram(tag, tensor.detach().cpu(), step)

    def add_image(self, tag: str, img: torch.Tensor, step: int):
        from .image import save_image

        d = os.path.join(self.log_dir, "images")
        os.makedirs(d, exist_ok=True)
        save_image(img, os.path.join(d, f"{tag.replace('/', '_')}-{step}.png"))
        if self._tb is not None:
            self._tb.add_image(tag, img.clamp(0, 1), step)

    def add_video(self, tag: str, frames, step: int, fps: int = 2):
        from .image import save_gif, to_uint8_hwc

        d = os.path.join(self.log_dir, "videos")
        os.makedirs(d, exist_ok=True)
        arrs = [to_uint8_hwc(f) for f in frames]
        save_gif(os.path.join(d, f"{tag.replace('/', '_')}-{step}.gif"), arrs, duration=1.0 / fps)

    def close(self):
        self._f.close()
        if self._tb is not None:
            self._tb.close()
