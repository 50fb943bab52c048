"""Loader for the in-tree gfx950 HIP extension.

The extension is built in-tree (`python setup.py build_ext --inplace` or
`__graft_entry__.build()`) so the resulting .so travels to the GPU box with
the repo snapshot. On a CUDA(ROCm)-visible machine a missing extension is a
hard error — the HIP path must be the path that runs (no silent eager
fallback); CPU-only machines never require it.
"""

import importlib
import os

import torch

_EXT = None
_TRIED = False


def load_extension():
    """Import the compiled extension module, caching the result."""
    global _EXT, _TRIED
    if _EXT is not None:
        return _EXT
    if _TRIED:
        _raise_missing()
    _TRIED = True
    try:
        _EXT = importlib.import_module("deepfake_detection_amd._hip_ops")
    except ImportError:
        _raise_missing()
    return _EXT


def _raise_missing():
    raise RuntimeError(
        "deepfake_detection_amd._hip_ops (gfx950 HIP extension) is not built. "
        "Run `python setup.py build_ext --inplace` (or __graft_entry__.build()) "
        "from the repo root. GPU execution without the HIP extension is "
        "disabled by design."
    )


def has_extension() -> bool:
    global _EXT
    if _EXT is not None:
        return True
    try:
        load_extension()
        return True
    except RuntimeError:
        return False


def gpu_ops_required() -> bool:
    """True when running on a ROCm-visible device (HIP path mandatory)."""
    if os.environ.get("DFD_AMD_FORCE_TORCH_OPS", "0") == "1":
        # explicit escape hatch for A/B benchmarking only
        return False
    return torch.cuda.is_available()
