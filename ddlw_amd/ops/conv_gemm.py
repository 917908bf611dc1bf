"""MFMA implicit-GEMM convolution dispatch (fwd/dgrad/wgrad).

The hand-written CDNA4 kernels live in ``hip/conv_gemm.hip``; this module
decides per-shape whether the ddlw kernel or MIOpen runs (measured dispatch
table) and wires the autograd Function. Until a shape is covered by the HIP
kernels, ``available`` returns False and Conv2d falls back to the library.
"""
from __future__ import annotations

import torch

# Populated as the implicit-GEMM kernel suite lands.


def available(mod, x: torch.Tensor, mode: str) -> bool:
    return False


def conv2d(x, weight, bias, stride, padding):
    raise NotImplementedError
