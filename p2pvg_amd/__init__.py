"""p2pvg_amd — MI355X-native point-to-point video generation framework.

A from-scratch framework with the capabilities of yccyenchicheng/p2pvg
(ICCV 2019, arXiv:1904.02912), built MI355X-first: PyTorch-ROCm driver,
hand-written CDNA4 (gfx950) HIP kernels for the hot ops, hipGraph-captured
recurrence, RCCL-over-xGMI data parallelism.
"""

__version__ = "1.0.0"

from .core import Config  # noqa: F401


def __getattr__(name):  # lazy top-level exports: keep `import p2pvg_amd` light
    if name == "P2PModel":
        from .models import P2PModel

        return P2PModel
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
