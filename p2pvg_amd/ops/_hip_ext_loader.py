"""Loader for the in-tree compiled extension (p2pvg_amd/ops/_C*.so)."""
from __future__ import annotations


def load():
    from . import _C  # in-tree .so built by setup_ext.py

    return _C
