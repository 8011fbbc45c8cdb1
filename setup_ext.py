"""Build the in-tree gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup_ext.py build_ext --inplace

Produces p2pvg_amd/ops/_C*.so next to the sources (IN-TREE: the .so travels to
the GPU box with the repo snapshot; a JIT cache under ~/.cache would not).
hipcc cross-compiles on CPU-only hosts.
"""
import os
import sys
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = Path(__file__).parent
CSRC = ROOT / "p2pvg_amd" / "ops" / "csrc"

# hipify writes *_hip.hip / *_hip.h shadow copies next to the sources;
# exclude them or a rebuild would compile everything twice.
sources = [str(CSRC / "ext.cpp")] + sorted(
    str(p) for p in CSRC.glob("*.hip") if not p.name.endswith("_hip.hip")
)

setup(
    name="p2pvg_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="p2pvg_amd.ops._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
