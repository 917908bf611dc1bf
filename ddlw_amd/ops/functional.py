"""Python-facing fused ops, HIP-backed on GPU with CPU (stock-op) oracles.

Populated kernel by kernel; each function documents the reference op it
replaces (SURVEY.md §2.4 kernel table) and has a parity test in
``tests/test_kernels.py`` comparing the HIP path against the fp32 stock op.
"""
from __future__ import annotations

import torch

# Kernel-backed autograd functions are registered here as the HIP suite lands
# (bn_relu, maxpool, softmax_ce, fused_sgd, gap, conv2d implicit-GEMM).
