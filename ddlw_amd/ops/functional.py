"""Functional index of the HIP-backed ops (see the modules that own them):

- fused BN(+add)+ReLU, maxpool, GAP, softmax-CE, u8 normalize: ``ops.layers``
- conv fwd/dgrad/wgrad implicit-GEMM + depthwise: ``ops.conv`` / ``ops.conv_gemm``
- fused SGD / Adam: ``ops.optim``
- raw tensor-level wrappers: ``ops.binding``
"""
from .layers import (  # noqa: F401
    normalize_u8_bf16,
    softmax_cross_entropy,
)
