"""``ddlw_amd.ops`` — hand-written CDNA4 (gfx950) HIP kernels.

The implicit GPU-kernel inventory of the reference rebuilt
as native HIP, built by ``ddlw_amd/ops/build.py`` (raw hipcc — no hipify, no
CUDA path, no Triton) and bound via ctypes in ``binding.py``.

On a GPU box the HIP extension is REQUIRED: ops raise if the library is
missing (no silent eager fallback; override only via DDLW_DISABLE_HIP_OPS=1).
On CPU the same layers fall back to stock PyTorch, which doubles as the
numerics oracle for the parity tests.
"""
from .runtime import lib, has_lib, require_lib, KernelLibError
from .layers import (
    BatchNormAct2d,
    MaxPool3x3s2,
    GlobalAvgPool2d,
    softmax_cross_entropy,
    normalize_u8_bf16,
)
from .optim import FusedSGD, FusedAdam
from .conv import Conv2d

__all__ = [
    "lib",
    "has_lib",
    "require_lib",
    "KernelLibError",
    "BatchNormAct2d",
    "MaxPool3x3s2",
    "GlobalAvgPool2d",
    "softmax_cross_entropy",
    "normalize_u8_bf16",
    "FusedSGD",
    "FusedAdam",
    "Conv2d",
]
