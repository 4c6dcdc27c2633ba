from . import reference, functional, philox, csr_torch
from ._ext import get_ext, has_ext, require_ext, use_hip

__all__ = ["reference", "functional", "philox", "csr_torch",
           "get_ext", "has_ext", "require_ext", "use_hip"]
