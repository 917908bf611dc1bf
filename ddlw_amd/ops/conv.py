"""Conv2d with dispatch between the ddlw MFMA implicit-GEMM HIP kernels and
the library path (MIOpen via F.conv2d).

The reference's convolutions live behind TF/Keras (kernels K1-K3 in
SURVEY.md §2.4); here they are first-class. Dispatch policy via
``DDLW_CONV`` env: ``auto`` (default — per-shape pick from the measured
table), ``hip`` (force hand-written kernels), ``stock`` (force MIOpen).
"""
from __future__ import annotations

import os
import torch
import torch.nn as nn
import torch.nn.functional as F


def conv_mode() -> str:
    return os.environ.get("DDLW_CONV", "auto")


class Conv2d(nn.Conv2d):
    """nn.Conv2d subclass so init/state_dict/bias handling are inherited;
    forward dispatches per policy. HIP implicit-GEMM path: ops.conv_gemm."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        mode = conv_mode()
        hip_ok = (
            mode != "stock"
            and x.is_cuda
            and x.dtype == torch.bfloat16
            and self.dilation == (1, 1)
            and os.environ.get("DDLW_DISABLE_HIP_OPS", "0") != "1"
        )
        if (
            hip_ok
            and self.groups == self.in_channels == self.out_channels
            and self.in_channels % 8 == 0
            and self.bias is None
            and (
                not torch.is_grad_enabled()  # inference: fwd-only kernel safe
                or not (x.requires_grad or self.weight.requires_grad)
            )
        ):
            # depthwise (K2 — MobileNetV2 blocks), inference/frozen path
            from . import binding

            st = self.stride[0] if isinstance(self.stride, tuple) else self.stride
            pd = self.padding[0] if isinstance(self.padding, tuple) else self.padding
            return binding.depthwise_fwd(
                x.contiguous(memory_format=torch.channels_last), self.weight, st, pd
            )
        if (
            hip_ok
            and self.groups == 1
            and self.in_channels == 3
            and self.out_channels == 64
            and self.kernel_size == (7, 7)
            and self.stride == (2, 2)
            and self.padding == (3, 3)
            and self.bias is None
            and os.environ.get("DDLW_STEM", "1") == "1"
        ):
            # dedicated C=3 stem kernel (K1): c4+halo repack + 2x8x4 igemm
            from . import conv_gemm

            return conv_gemm.stem_conv2d(x, self.weight)
        if hip_ok and self.groups == 1:
            from . import conv_gemm

            if conv_gemm.available(self, x, mode):
                hip_fwd, hip_dgrad, hip_wgrad = self._ddlw_route
                return conv_gemm.conv2d(
                    x, self.weight, self.bias, self.stride, self.padding,
                    hip_fwd, hip_dgrad, hip_wgrad,
                )
        return F.conv2d(
            x, self.weight.to(x.dtype),
            None if self.bias is None else self.bias.to(x.dtype),
            self.stride, self.padding, self.dilation, self.groups,
        )
