"""HIP extension loader + dispatch policy.

Policy (per the build contract): on CUDA/HIP tensors the hand-written
gfx950 kernels MUST run — if the extension is missing on a GPU machine we
raise instead of silently falling back to eager PyTorch. On CPU tensors the
pure-torch reference implementations (ops/reference.py) are used; they also
serve as the numerics oracle in tests.
"""
from __future__ import annotations

import os

_ext = None
_tried = False


def get_ext():
    global _ext, _tried
    if not _tried:
        _tried = True
        try:
            from bnsgcn_amd import _C  # built in-tree by setup.py build_ext --inplace
            _ext = _C
        except ImportError:
            _ext = None
    return _ext


def has_ext() -> bool:
    return get_ext() is not None


def require_ext():
    e = get_ext()
    if e is None:
        raise RuntimeError(
            "bnsgcn_amd HIP extension (bnsgcn_amd/_C*.so) is not built but a "
            "GPU tensor reached a compute op. Build it with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            "Refusing to fall back to eager PyTorch on GPU.")
    return e


def use_hip(t) -> bool:
    """True if op dispatch for tensor `t` should go to the HIP kernels."""
    if not t.is_cuda:
        return False
    if os.environ.get("BNSGCN_FORCE_TORCH") == "1":  # perf A/B escape hatch
        return False
    require_ext()
    return True
