"""ops — the MI355X compute path.

The hand-written HIP/CDNA4 kernels (gfx950) live in ``ops/hip/`` and are
compiled in-tree to ``_hip_ops*.so`` (see ops/build.py, driven by
``__graft_entry__.build()``).  This module loads that extension and exposes
the dispatch policy:

  * tensors on CUDA (= ROCm/HIP) devices run the HIP kernels — if the
    extension is missing on a GPU machine the op RAISES instead of silently
    falling back to eager PyTorch (the framework's GPU path is the native
    one, never a shim);
  * CPU tensors use reference torch ops (the CPU path exists for the
    no-GPU test tier and as the numerics oracle).

Set DDPX_ALLOW_TORCH_FALLBACK=1 to permit torch ops on GPU (bring-up and
A/B debugging only).
"""
from __future__ import annotations

import importlib
import os
import sys

_EXT = None
_EXT_TRIED = False


def _package_dir() -> str:
    return os.path.dirname(os.path.abspath(__file__))


class _SyncDebugProxy:
    """DDPX_SYNC_DEBUG=1: every extension call is followed by a
    device synchronize + error check, so an async kernel fault surfaces at
    the faulting op's Python call site instead of a later sync point."""

    def __init__(self, ext):
        self._ext = ext

    def __getattr__(self, name):
        fn = getattr(self._ext, name)
        if not callable(fn):
            return fn

        def wrapped(*args, **kwargs):
            import torch
            out = fn(*args, **kwargs)
            torch.cuda.synchronize()
            return out

        return wrapped


def load_extension(required: bool = False):
    """Load the in-tree _hip_ops extension; cache the module object."""
    global _EXT, _EXT_TRIED
    if os.environ.get("DDPX_FORCE_TORCH", "0") == "1":
        # A/B benchmarking: pretend the HIP extension is absent so the ops
        # take the torch-ROCm (MIOpen/hipBLASLt) bring-up path.  Pair with
        # DDPX_ALLOW_TORCH_FALLBACK=1.
        return None
    if _EXT is not None:
        return _EXT
    if _EXT_TRIED and not required:
        return None
    _EXT_TRIED = True
    pkg_dir = _package_dir()
    if pkg_dir not in sys.path:
        sys.path.insert(0, pkg_dir)
    try:
        import torch  # noqa: F401  (extension links against torch libs)
        _EXT = importlib.import_module("_hip_ops")
        if os.environ.get("DDPX_SYNC_DEBUG", "0") == "1":
            _EXT = _SyncDebugProxy(_EXT)
    except ImportError as e:
        _EXT = None
        if required:
            raise RuntimeError(
                "ddp_tricks_amd HIP extension (_hip_ops) is not built/loadable "
                "on this machine. Build it with `python -c \"import __graft_entry__ as g; g.build()\"` "
                f"from the repo root. Original error: {e}"
            ) from e
    return _EXT


def have_extension() -> bool:
    return load_extension(required=False) is not None


def allow_torch_fallback() -> bool:
    return (os.environ.get("DDPX_ALLOW_TORCH_FALLBACK", "0") == "1"
            or os.environ.get("DDPX_FORCE_TORCH", "0") == "1")


def require_ext_for(tensor):
    """Dispatch guard for GPU tensors: HIP ext or loud failure."""
    if not tensor.is_cuda:
        return None
    ext = load_extension(required=False)
    if ext is not None:
        return ext
    if allow_torch_fallback():
        return None
    raise RuntimeError(
        "ddp_tricks_amd: GPU tensor reached an op but the _hip_ops HIP "
        "extension is not loaded. The GPU compute path is HIP-native by "
        "policy; build the extension (see ops/build.py) or set "
        "DDPX_ALLOW_TORCH_FALLBACK=1 explicitly for bring-up runs."
    )


from .functional import (  # noqa: E402,F401
    conv2d, linear, batch_norm, max_pool2d, relu, cross_entropy_loss,
)
from .optim import FusedSGD  # noqa: E402,F401
