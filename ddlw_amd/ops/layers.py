"""HIP-backed layers with stock-PyTorch fallbacks (the fallback doubles as
the fp32 numerics oracle on CPU — SURVEY.md §4 test plan item 2).

Dispatch rule: the hand-written CDNA4 kernel path runs when (a) the tensor is
on a CUDA/HIP device, (b) dtype is bf16 with channels_last memory, and
(c) the channel count is kernel-supported. On a GPU box the kernel library is
REQUIRED (ops.runtime.require_lib raises if missing) — there is no silent
eager fallback on GPU unless DDLW_DISABLE_HIP_OPS=1 is set explicitly.
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import binding


def _hip_ops_disabled() -> bool:
    return os.environ.get("DDLW_DISABLE_HIP_OPS", "0") == "1"


def _use_hip(x: torch.Tensor, C: Optional[int] = None) -> bool:
    if not x.is_cuda or _hip_ops_disabled():
        return False
    if x.dtype != torch.bfloat16:
        return False
    if C is not None and not binding.supported_channels(C):
        return False
    return True  # require_lib will raise loudly if the .so is missing


def _cl(t: torch.Tensor) -> torch.Tensor:
    if t.dim() == 4:
        return t.contiguous(memory_format=torch.channels_last)
    return t.contiguous()


# --------------------------------------------------------------------------- #
# Fused BatchNorm (+residual add) (+ReLU)
# --------------------------------------------------------------------------- #


class _BnActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, res, weight, bias, running_mean, running_var,
                training: bool, momentum: float, eps: float, relu: bool):
        x = _cl(x)
        if res is not None:
            res = _cl(res)
        if training:
            mean, rstd = binding.bn_stats(x, eps, momentum, running_mean, running_var)
        else:
            mean = running_mean.to(torch.float32)
            rstd = (running_var.to(torch.float32) + eps).rsqrt()
        y, mask = binding.bn_apply(x, res, mean, rstd, weight, bias, relu)
        if mask is None:
            mask = x.new_empty(0, dtype=torch.uint8)
        ctx.save_for_backward(x, mask, mean, rstd, weight)
        ctx.relu = relu
        ctx.has_res = res is not None
        ctx.bn_training = training
        return y

    @staticmethod
    def backward(ctx, dy):
        x, mask, mean, rstd, weight = ctx.saved_tensors
        dy = _cl(dy)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        dbeta, dgamma = binding.bn_bwd_reduce(dy, mask, x, mean, rstd, ctx.relu)
        if ctx.bn_training:
            cb, cg = dbeta, dgamma
        else:
            # eval-mode BN: mean/var are constants -> no batch-stat terms
            cb = torch.zeros_like(dbeta)
            cg = torch.zeros_like(dgamma)
        dx, dres = binding.bn_bwd_dx(dy, mask, x, mean, rstd, weight, cb, cg,
                                     ctx.relu, ctx.has_res)
        return (dx, dres if ctx.has_res else None, dgamma, dbeta,
                None, None, None, None, None, None)


class BatchNormAct2d(nn.Module):
    """BatchNorm2d with optional fused residual-add and ReLU.

    Replaces the reference's BN (+ReLU) pairs (kernels K4+K5)
    with ONE fused NHWC pass each way. Parameter/buffer names match
    nn.BatchNorm2d so state dicts interchange.
    """

    def __init__(self, num_features: int, eps: float = 1e-5, momentum: float = 0.1,
                 relu: bool = False):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.relu = relu
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked", torch.tensor(0, dtype=torch.long))

    def _apply(self, fn, recurse=True):
        # keep the step counter on CPU: a device-resident counter costs one
        # scalar add KERNEL per BN per training step (53 launches/step on
        # ResNet-50, ~0.8% of step time); nothing reads it on-device
        out = super()._apply(fn, recurse)
        if self.num_batches_tracked.is_cuda:
            self.num_batches_tracked = self.num_batches_tracked.cpu()
        return out

    def forward(self, x: torch.Tensor, residual: Optional[torch.Tensor] = None):
        if _use_hip(x, self.num_features):
            if self.training:
                self.num_batches_tracked += 1
            # kernels take fp32 parameter/stat pointers; guard against a
            # model-wide .bfloat16()/.half() cast (silent OOB otherwise)
            w, b = self.weight, self.bias
            rm, rv = self.running_mean, self.running_var
            if w.dtype != torch.float32:
                w, b = w.float(), b.float()
            if rm.dtype != torch.float32:
                if self.training:
                    raise TypeError(
                        "BatchNormAct2d training requires fp32 running stats "
                        f"(got {rm.dtype}); keep BN buffers in fp32"
                    )
                rm, rv = rm.float(), rv.float()
            return _BnActFn.apply(
                x, residual, w, b, rm, rv,
                self.training, self.momentum, self.eps, self.relu,
            )
        # stock fallback (CPU oracle / non-bf16 path)
        y = F.batch_norm(
            x, self.running_mean, self.running_var,
            self.weight.to(x.dtype), self.bias.to(x.dtype),
            self.training, self.momentum, self.eps,
        )
        if residual is not None:
            y = y + residual
        return F.relu(y) if self.relu else y

    def folded_scale_bias(self):
        """Eval-mode BN as one affine transform: y = x*scale + bias with
        scale = gamma*rsqrt(rv+eps), bias = beta - rm*scale. Cached until
        any source tensor mutates (torch _version counters)."""
        key = (self.weight._version, self.bias._version,
               self.running_mean._version, self.running_var._version)
        cached = getattr(self, "_fold_cache", None)
        if cached is not None and cached[0] == key:
            return cached[1], cached[2]
        with torch.no_grad():
            scale = (self.weight.float()
                     * (self.running_var.float() + self.eps).rsqrt())
            bias = self.bias.float() - self.running_mean.float() * scale
        scale, bias = scale.contiguous(), bias.contiguous()
        self._fold_cache = (key, scale, bias)
        return scale, bias

    def extra_repr(self) -> str:
        return f"{self.num_features}, relu={self.relu}"


# --------------------------------------------------------------------------- #
# Dropout (K7 — head path, reference P1/02:174)
# --------------------------------------------------------------------------- #


class _DropoutFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p: float, seed: int):
        y, mask = binding.dropout_fwd(x.contiguous(), p, seed)
        ctx.save_for_backward(mask)
        ctx.p = p
        return y

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        return binding.dropout_bwd(dy.contiguous(), mask, ctx.p), None, None


class Dropout(nn.Module):
    """Bitmask dropout on the ddlw kernel (counter-based RNG, deterministic
    per (base_seed, call_counter)); stock fallback off-GPU / odd sizes."""

    def __init__(self, p: float = 0.5):
        super().__init__()
        self.p = float(p)
        self._calls = 0
        self._base_seed = int(torch.initial_seed()) & (2**63 - 1)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not self.training or self.p <= 0.0:
            return x
        if (x.is_cuda and x.dtype == torch.bfloat16 and x.numel() % 8 == 0
                and not _hip_ops_disabled()):
            self._calls += 1
            seed = self._base_seed + self._calls * 0x9E3779B97F4A7C15
            return _DropoutFn.apply(x, self.p, seed)
        return F.dropout(x, self.p, self.training)

    def extra_repr(self) -> str:
        return f"p={self.p}"


# --------------------------------------------------------------------------- #
# MaxPool 3x3 s2 p1
# --------------------------------------------------------------------------- #


class _MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = _cl(x)
        y, argmax = binding.maxpool3x3s2_fwd(x)
        ctx.save_for_backward(argmax)
        ctx.in_shape = x.shape
        return y

    @staticmethod
    def backward(ctx, dy):
        (argmax,) = ctx.saved_tensors
        dy = _cl(dy)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        return binding.maxpool3x3s2_bwd(dy, argmax, ctx.in_shape)


class MaxPool3x3s2(nn.Module):
    """kernel_size=3, stride=2, padding=1 (the ResNet stem pool)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if _use_hip(x, x.shape[1]):
            return _MaxPoolFn.apply(x)
        return F.max_pool2d(x, 3, stride=2, padding=1)


# --------------------------------------------------------------------------- #
# Global average pooling -> (N, C)
# --------------------------------------------------------------------------- #


class _GapFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = _cl(x)
        ctx.in_shape = x.shape
        return binding.gap_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        return binding.gap_bwd(dy, ctx.in_shape)


class GlobalAvgPool2d(nn.Module):
    """AdaptiveAvgPool2d(1) + flatten, fused (kernel K6)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if _use_hip(x, x.shape[1]):
            return _GapFn.apply(x)
        return F.adaptive_avg_pool2d(x, 1).flatten(1)


# --------------------------------------------------------------------------- #
# Fused softmax cross-entropy (K9)
# --------------------------------------------------------------------------- #


class _SoftmaxCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):
        loss, dlogits = binding.softmax_ce(logits, labels, 1.0 / logits.shape[0])
        ctx.save_for_backward(dlogits)
        return loss.squeeze(0)

    @staticmethod
    def backward(ctx, dloss):
        (dlogits,) = ctx.saved_tensors
        return dlogits * dloss, None


def softmax_cross_entropy(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Sparse CE from logits; fused fwd+bwd on GPU."""
    if logits.is_cuda and not _hip_ops_disabled():
        return _SoftmaxCEFn.apply(logits.float(), labels)
    return F.cross_entropy(logits.float(), labels)


# --------------------------------------------------------------------------- #
# preprocess normalize (uint8 -> bf16 in [-1, 1])
# --------------------------------------------------------------------------- #


def normalize_u8_bf16(x_u8: torch.Tensor) -> torch.Tensor:
    """uint8 NHWC (channels_last) -> bf16 NHWC, x/127.5-1 in one pass."""
    if x_u8.is_cuda and not _hip_ops_disabled() and x_u8.numel() % 16 == 0:
        if x_u8.dim() == 4:
            assert x_u8.is_contiguous(memory_format=torch.channels_last)
            out = torch.empty(
                x_u8.shape, dtype=torch.bfloat16, device=x_u8.device,
                memory_format=torch.channels_last,
            )
        else:
            out = torch.empty(x_u8.shape, dtype=torch.bfloat16, device=x_u8.device)
        binding.normalize_u8(x_u8, out)
        return out
    return (x_u8.to(torch.bfloat16) / 127.5) - 1.0
