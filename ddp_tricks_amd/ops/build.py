"""In-tree build of the _hip_ops extension for gfx950.

Uses torch.utils.cpp_extension (which drives hipcc for the .hip sources
under PYTORCH_ROCM_ARCH=gfx950) with the build directory inside the package
so the resulting _hip_ops.so travels with the repo snapshot to GPU boxes.
hipcc cross-compiles without a GPU — this runs on CPU-only machines too.
"""
from __future__ import annotations

import os
import shutil
import sys

OPS_DIR = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(OPS_DIR, "hip")
BUILD_DIR = os.path.join(OPS_DIR, "build")

SOURCES = [
    "bindings.cpp",
    "elementwise.hip",
    "ce.hip",
    "pool.hip",
    "bn.hip",
    "gemm.hip",
    "conv.hip",
]


def build(verbose: bool = False):
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils.cpp_extension import load

    os.makedirs(BUILD_DIR, exist_ok=True)
    module = load(
        name="_hip_ops",
        sources=[os.path.join(HIP_DIR, s) for s in SOURCES],
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950", "-std=c++17"],
        build_directory=BUILD_DIR,
        verbose=verbose,
        is_python_module=True,
    )
    # copy the .so next to ops/__init__.py so `import _hip_ops` finds it
    so_path = os.path.join(BUILD_DIR, "_hip_ops.so")
    if os.path.exists(so_path):
        shutil.copy2(so_path, os.path.join(OPS_DIR, "_hip_ops.so"))
    return module


if __name__ == "__main__":
    build(verbose="-v" in sys.argv)
    print("built _hip_ops")
