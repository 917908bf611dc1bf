"""``ddlw_amd.ops`` — hand-written CDNA4 (gfx950) HIP kernels.

The implicit GPU-kernel inventory of the reference rebuilt
as native HIP, loaded via :mod:`ddlw_amd.ops.runtime` (raw ``hipcc``-built
shared library, ctypes-bound — no hipify, no CUDA path, no Triton).

On a GPU box the HIP extension is REQUIRED: ops raise if the library is
missing (no silent eager fallback). On CPU the same ops fall back to stock
PyTorch, which doubles as the numerics oracle for the parity tests.
"""
from .runtime import lib, has_lib, require_lib, KernelLibError
from . import functional

__all__ = ["lib", "has_lib", "require_lib", "KernelLibError", "functional"]
