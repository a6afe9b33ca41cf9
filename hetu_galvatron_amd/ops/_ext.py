"""HIP extension loader.

The CDNA4 kernels live in `hetu_galvatron_amd/ops/csrc/*.hip`, built IN-TREE
(`python setup.py build_ext --inplace`, PYTORCH_ROCM_ARCH=gfx950) into
`hetu_galvatron_amd/ops/_galvatron_hip.*.so` so the .so travels with the
repo snapshot to GPU boxes.

Policy: on a GPU ("cuda" is ROCm here) the native kernels are REQUIRED —
a missing extension raises instead of silently falling back to eager
PyTorch.  CPU execution (tests on GPU-less boxes) uses the torch reference
implementations in reference_ops.py.  Set GALVATRON_EAGER_FALLBACK=1 to
explicitly allow eager on GPU (debug only).
"""
from __future__ import annotations

import importlib
import os

_EXT = None
_TRIED = False


def get_ext(optional: bool = True):
    """Return the native module or None (if optional) / raise (if not)."""
    global _EXT, _TRIED
    if not _TRIED:
        _TRIED = True
        try:
            _EXT = importlib.import_module("hetu_galvatron_amd.ops._galvatron_hip")
        except ImportError as e:
            _EXT = None
            _IMPORT_ERROR[0] = e
    if _EXT is None and not optional:
        raise RuntimeError(
            "galvatron HIP extension not built "
            f"(import error: {_IMPORT_ERROR[0]}). Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950)."
        )
    return _EXT


_IMPORT_ERROR = [None]


def native_available() -> bool:
    return get_ext(optional=True) is not None


def eager_fallback_allowed() -> bool:
    return os.environ.get("GALVATRON_EAGER_FALLBACK", "0") == "1"


def use_native(tensor) -> bool:
    """Decide native vs eager for this tensor's device; raise loudly if a GPU
    tensor has no native path and fallback was not explicitly allowed."""
    if not tensor.is_cuda:
        return False
    if native_available():
        return True
    if eager_fallback_allowed():
        return False
    # loud failure: GPU execution without the CDNA4 kernels is a silent
    # perf/parity bug, not a fallback.
    get_ext(optional=False)
    return True
