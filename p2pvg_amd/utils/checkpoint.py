"""Checkpoint I/O keeping the reference's 12-key dict contract.

Improvements over the reference: the latest-copy is atomic (write temp +
rename, vs reference train.py:279's `os.system("cp ...")`), and the serialized
`opt` is a plain dict (Config.to_dict()) rather than a pickled argparse
Namespace — but reference-written checkpoints (Namespace `opt`) still load.
"""
from __future__ import annotations

import os
import tempfile
from typing import Dict, Optional

import torch

from ..core import Config


def save_checkpoint(model, epoch: int, fname: str, latest_name: Optional[str] = None) -> None:
    states = model.state_for_checkpoint(epoch)
    d = os.path.dirname(fname) or "."
    os.makedirs(d, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=d, suffix=".tmp")
    os.close(fd)
    try:
        torch.save(states, tmp)
        os.replace(tmp, fname)
    finally:
        if os.path.exists(tmp):
            os.unlink(tmp)
    if latest_name:
        fd, tmp = tempfile.mkstemp(dir=d, suffix=".tmp")
        os.close(fd)
        try:
            torch.save(states, tmp)
            os.replace(tmp, latest_name)
        finally:
            if os.path.exists(tmp):
                os.unlink(tmp)


def config_from_states(states: Dict) -> Config:
    """Recover a Config from a checkpoint's `opt` entry, accepting both our
    dict form and a reference-style argparse Namespace."""
    opt = states["opt"]
    if isinstance(opt, dict):
        return Config.from_dict(opt)
    # Namespace (or anything attribute-shaped)
    return Config.from_dict({k: v for k, v in vars(opt).items()})


def load_checkpoint(pth: str, map_location="cpu") -> Dict:
    return torch.load(pth, map_location=map_location, weights_only=False)
