"""ctypes bindings for libddlw_kernels.so (typed, tensor-level wrappers).

All activation tensors must be channels_last (NHWC memory) bf16; statistics
and parameters fp32. Every call launches onto torch's current HIP stream, so
the kernels are stream-ordered with PyTorch ops and hipGraph-capturable.
"""
from __future__ import annotations

import ctypes
from typing import Optional

import torch

from .runtime import check, current_stream_ptr, require_lib


def _p(t: Optional[torch.Tensor]):
    return ctypes.c_void_p(0 if t is None else t.data_ptr())


def _nhwc_ptr(t: torch.Tensor):
    """Data pointer of a channels_last activation (N,C,H,W logical)."""
    if t.dim() == 4:
        assert t.is_contiguous(memory_format=torch.channels_last), "need channels_last"
    else:
        assert t.is_contiguous()
    assert t.dtype == torch.bfloat16, f"need bf16, got {t.dtype}"
    return _p(t)


def _rows_c(t: torch.Tensor):
    if t.dim() == 4:
        n, c, h, w = t.shape
        return n * h * w, c
    rows, c = t.shape[0], t.shape[-1]
    return rows, c


def supported_channels(c: int) -> bool:
    return c % 8 == 0


import functools


@functools.lru_cache(maxsize=512)
def _nparts(rows: int, C: int) -> int:
    lib = require_lib()
    return int(lib.ddlw_bn_nparts(ctypes.c_long(rows), ctypes.c_int(C)))


def bn_stats(x: torch.Tensor, eps: float, momentum: float,
             running_mean: Optional[torch.Tensor], running_var: Optional[torch.Tensor]):
    lib = require_lib()
    rows, C = _rows_c(x)
    dev = x.device
    ny = _nparts(rows, C)
    part = torch.empty(2, ny, C, dtype=torch.float32, device=dev)
    mean = torch.empty(C, dtype=torch.float32, device=dev)
    rstd = torch.empty(C, dtype=torch.float32, device=dev)
    s = current_stream_ptr()
    check(lib.ddlw_bn_stats(_nhwc_ptr(x), _p(part[0]), _p(part[1]),
                            ctypes.c_long(rows), ctypes.c_int(C),
                            ctypes.c_void_p(s)), "bn_stats")
    check(lib.ddlw_bn_finalize(_p(part[0]), _p(part[1]), _p(mean), _p(rstd),
                               _p(running_mean), _p(running_var),
                               ctypes.c_long(rows), ctypes.c_int(C),
                               ctypes.c_float(eps), ctypes.c_float(momentum),
                               ctypes.c_void_p(s)), "bn_finalize")
    return mean, rstd


def bn_finalize_parts(part_sum: torch.Tensor, part_sumsq: torch.Tensor,
                      nparts: int, rows: int, C: int, eps: float,
                      momentum: float, running_mean, running_var):
    """Finalize mean/rstd from conv-epilogue-fused partial sums (the
    k_bn_stats read pass is skipped entirely)."""
    lib = require_lib()
    dev = part_sum.device
    mean = torch.empty(C, dtype=torch.float32, device=dev)
    rstd = torch.empty(C, dtype=torch.float32, device=dev)
    if nparts > 512:
        # chip-filling fold first: k_bn_finalize's grid is ceil(C/32)
        # blocks and would serialize a 12k-row partial read on 2 CUs
        G = max(8, 512 // max(1, (C + 31) // 32))
        stage = torch.empty(2, G, C, dtype=torch.float32, device=dev)
        check(lib.ddlw_bn_parts_fold(_p(part_sum), _p(part_sumsq),
                                     _p(stage[0]), _p(stage[1]),
                                     ctypes.c_long(nparts), ctypes.c_int(C),
                                     ctypes.c_int(G),
                                     ctypes.c_void_p(current_stream_ptr())),
              "bn_parts_fold")
        part_sum, part_sumsq, nparts = stage[0], stage[1], G
    check(lib.ddlw_bn_finalize_n(_p(part_sum), _p(part_sumsq), _p(mean),
                                 _p(rstd), _p(running_mean), _p(running_var),
                                 ctypes.c_long(rows), ctypes.c_int(C),
                                 ctypes.c_int(nparts), ctypes.c_float(eps),
                                 ctypes.c_float(momentum),
                                 ctypes.c_void_p(current_stream_ptr())),
          "bn_finalize_n")
    return mean, rstd


def bn_grad_finalize_parts(part_db: torch.Tensor, part_dg: torch.Tensor,
                           nparts: int, C: int):
    """(dbeta, dgamma) from dgrad-epilogue-fused reduce partials."""
    lib = require_lib()
    dev = part_db.device
    if nparts > 512:
        G = max(8, 512 // max(1, (C + 31) // 32))
        stage = torch.empty(2, G, C, dtype=torch.float32, device=dev)
        check(lib.ddlw_bn_parts_fold(_p(part_db), _p(part_dg), _p(stage[0]),
                                     _p(stage[1]), ctypes.c_long(nparts),
                                     ctypes.c_int(C), ctypes.c_int(G),
                                     ctypes.c_void_p(current_stream_ptr())),
              "bn_parts_fold")
        part_db, part_dg, nparts = stage[0], stage[1], G
    dbeta = torch.empty(C, dtype=torch.float32, device=dev)
    dgamma = torch.empty(C, dtype=torch.float32, device=dev)
    check(lib.ddlw_bn_grad_finalize_n(_p(part_db), _p(part_dg), _p(dbeta),
                                      _p(dgamma), ctypes.c_int(C),
                                      ctypes.c_int(nparts),
                                      ctypes.c_void_p(current_stream_ptr())),
          "bn_grad_finalize_n")
    return dbeta, dgamma


def bn_apply(x: torch.Tensor, res: Optional[torch.Tensor], mean, rstd, gamma, beta,
             relu: bool):
    """Returns (y, mask): mask is a uint8 tensor of rows*C/8 relu bits (one
    byte per 8-channel vector) when relu, else None — backward reads the
    mask instead of re-reading y."""
    lib = require_lib()
    rows, C = _rows_c(x)
    y = torch.empty_like(x)
    mask = (
        torch.empty(rows * (C // 8), dtype=torch.uint8, device=x.device)
        if relu
        else None
    )
    check(
        lib.ddlw_bn_apply(
            _nhwc_ptr(x), _p(res), _nhwc_ptr(y), _p(mask), _p(mean), _p(rstd),
            _p(gamma), _p(beta), ctypes.c_long(rows), ctypes.c_int(C),
            ctypes.c_int(1 if relu else 0), ctypes.c_void_p(current_stream_ptr())
        ),
        "bn_apply",
    )
    return y, mask


def bn_bwd_reduce(dy, mask, x, mean, rstd, relu: bool):
    lib = require_lib()
    rows, C = _rows_c(x)
    dev = x.device
    ny = _nparts(rows, C)
    part = torch.empty(2, ny, C, dtype=torch.float32, device=dev)
    dbeta = torch.empty(C, dtype=torch.float32, device=dev)
    dgamma = torch.empty(C, dtype=torch.float32, device=dev)
    s = current_stream_ptr()
    check(
        lib.ddlw_bn_bwd_reduce(
            _nhwc_ptr(dy), _p(mask), _nhwc_ptr(x), _p(mean), _p(rstd), _p(part[0]),
            _p(part[1]), ctypes.c_long(rows), ctypes.c_int(C),
            ctypes.c_int(1 if relu else 0), ctypes.c_void_p(s)
        ),
        "bn_bwd_reduce",
    )
    check(
        lib.ddlw_bn_grad_finalize(
            _p(part[0]), _p(part[1]), _p(dbeta), _p(dgamma), ctypes.c_long(rows),
            ctypes.c_int(C), ctypes.c_void_p(s)
        ),
        "bn_grad_finalize",
    )
    return dbeta, dgamma


def bn_bwd_dx(dy, mask, x, mean, rstd, gamma, dbeta, dgamma, relu: bool,
              want_dres: bool):
    lib = require_lib()
    rows, C = _rows_c(x)
    dx = torch.empty_like(x)
    dres = torch.empty_like(x) if want_dres else None
    check(
        lib.ddlw_bn_bwd_dx(
            _nhwc_ptr(dy), _p(mask), _nhwc_ptr(x), _p(mean), _p(rstd), _p(gamma),
            _p(dbeta), _p(dgamma), _p(dx), _p(dres), ctypes.c_long(rows),
            ctypes.c_int(C), ctypes.c_int(1 if relu else 0),
            ctypes.c_void_p(current_stream_ptr())
        ),
        "bn_bwd_dx",
    )
    return dx, dres


def maxpool3x3s2_fwd(x: torch.Tensor):
    lib = require_lib()
    n, c, h, w = x.shape
    ho, wo = (h + 1) // 2, (w + 1) // 2
    y = torch.empty((n, c, ho, wo), dtype=x.dtype, device=x.device,
                    memory_format=torch.channels_last)
    argmax = torch.empty(n * ho * wo * c, dtype=torch.uint8, device=x.device)
    check(
        lib.ddlw_maxpool3x3s2_fwd(
            _nhwc_ptr(x), _nhwc_ptr(y), _p(argmax), n, h, w, c, ho, wo,
            ctypes.c_void_p(current_stream_ptr())
        ),
        "maxpool_fwd",
    )
    return y, argmax


def maxpool3x3s2_bwd(dy: torch.Tensor, argmax: torch.Tensor, in_shape):
    lib = require_lib()
    n, c, h, w = in_shape
    ho, wo = dy.shape[2], dy.shape[3]
    dx = torch.empty((n, c, h, w), dtype=dy.dtype, device=dy.device,
                     memory_format=torch.channels_last)
    check(
        lib.ddlw_maxpool3x3s2_bwd(
            _nhwc_ptr(dy), _p(argmax), _nhwc_ptr(dx), n, h, w, c, ho, wo,
            ctypes.c_void_p(current_stream_ptr())
        ),
        "maxpool_bwd",
    )
    return dx


def gap_fwd(x: torch.Tensor) -> torch.Tensor:
    lib = require_lib()
    n, c, h, w = x.shape
    y = torch.empty((n, c), dtype=x.dtype, device=x.device)
    check(lib.ddlw_gap_fwd(_nhwc_ptr(x), _p(y), n, h * w, c,
                           ctypes.c_void_p(current_stream_ptr())), "gap_fwd")
    return y


def gap_bwd(dy: torch.Tensor, in_shape) -> torch.Tensor:
    lib = require_lib()
    n, c, h, w = in_shape
    dx = torch.empty((n, c, h, w), dtype=dy.dtype, device=dy.device,
                     memory_format=torch.channels_last)
    check(lib.ddlw_gap_bwd(_p(dy.contiguous()), _nhwc_ptr(dx), n, h * w, c,
                           ctypes.c_void_p(current_stream_ptr())), "gap_bwd")
    return dx


def softmax_ce(logits: torch.Tensor, labels: torch.Tensor, grad_scale: float):
    """Returns (mean loss tensor [1], dlogits). logits fp32 [B,K]."""
    lib = require_lib()
    B, K = logits.shape
    logits = logits.contiguous()
    dlogits = torch.empty_like(logits)
    loss_sum = torch.zeros(1, dtype=torch.float32, device=logits.device)
    check(
        lib.ddlw_softmax_ce(
            _p(logits), _p(labels.contiguous()), _p(dlogits), _p(loss_sum), B, K,
            ctypes.c_float(grad_scale), ctypes.c_void_p(current_stream_ptr())
        ),
        "softmax_ce",
    )
    return loss_sum / B, dlogits


def fused_sgd(chunk_desc: torch.Tensor, nchunks: int, max_numel: int, lr: float,
              momentum: float, weight_decay: float, first_step: bool):
    """chunk_desc: int64 device tensor [nchunks, 5] of
    (p_ptr, g_ptr, m_ptr, p_bf16_ptr_or_0, numel)."""
    lib = require_lib()
    check(
        lib.ddlw_fused_sgd(
            _p(chunk_desc), nchunks, ctypes.c_long(max_numel), ctypes.c_float(lr),
            ctypes.c_float(momentum), ctypes.c_float(weight_decay),
            ctypes.c_int(1 if first_step else 0),
            ctypes.c_void_p(current_stream_ptr())
        ),
        "fused_sgd",
    )


def normalize_u8(x_u8: torch.Tensor, out_bf16: torch.Tensor):
    lib = require_lib()
    n = x_u8.numel()
    assert n % 16 == 0
    check(lib.ddlw_normalize_u8(_p(x_u8), _p(out_bf16), ctypes.c_long(n),
                                ctypes.c_void_p(current_stream_ptr())), "normalize_u8")


def fused_adam(chunk_desc: torch.Tensor, nchunks: int, max_numel: int,
               lr: float, beta1: float, beta2: float, eps: float,
               weight_decay: float, bc1: float, bc2: float):
    """chunk_desc: int64 device tensor [nchunks, 7] of
    (p_ptr, g_ptr, m_ptr, v_ptr, p_bf16_ptr_or_0, numel, flags)."""
    lib = require_lib()
    check(
        lib.ddlw_fused_adam(
            _p(chunk_desc), nchunks, ctypes.c_long(max_numel),
            ctypes.c_float(lr), ctypes.c_float(beta1), ctypes.c_float(beta2),
            ctypes.c_float(eps), ctypes.c_float(weight_decay),
            ctypes.c_float(bc1), ctypes.c_float(bc2),
            ctypes.c_void_p(current_stream_ptr())
        ),
        "fused_adam",
    )


def depthwise_fwd(x: torch.Tensor, weight: torch.Tensor, stride: int, padding: int) -> torch.Tensor:
    """Depthwise conv2d forward (groups == C). weight [C,1,R,S] -> transposed
    to [R*S][C] host-side so taps read contiguous channel vectors."""
    lib = require_lib()
    n, c, h, w = x.shape
    _, _, r, s = weight.shape
    ho = (h + 2 * padding - r) // stride + 1
    wo = (w + 2 * padding - s) // stride + 1
    w_t = weight.reshape(c, r * s).t().contiguous()  # [R*S][C]
    if w_t.dtype != torch.bfloat16:
        w_t = w_t.to(torch.bfloat16)
    y = torch.empty((n, c, ho, wo), dtype=torch.bfloat16, device=x.device,
                    memory_format=torch.channels_last)
    check(
        lib.ddlw_depthwise_fwd(
            _nhwc_ptr(x), _p(w_t), _nhwc_ptr(y), n, h, w, c, ho, wo, r, s,
            stride, padding, ctypes.c_void_p(current_stream_ptr())
        ),
        "depthwise_fwd",
    )
    return y


def dropout_fwd(x: torch.Tensor, p: float, seed: int):
    """K7: bitmask dropout (counter-based RNG, deterministic per seed).
    Returns (y, mask); x flat-size must be a multiple of 8."""
    lib = require_lib()
    n = x.numel()
    y = torch.empty_like(x)
    mask = torch.empty(n // 8, dtype=torch.uint8, device=x.device)
    check(lib.ddlw_dropout_fwd(_p(x), _p(y), _p(mask), ctypes.c_long(n),
                               ctypes.c_float(p),
                               ctypes.c_ulonglong(seed & (2**64 - 1)),
                               ctypes.c_void_p(current_stream_ptr())),
          "dropout_fwd")
    return y, mask


def dropout_bwd(dy: torch.Tensor, mask: torch.Tensor, p: float):
    lib = require_lib()
    dx = torch.empty_like(dy)
    check(lib.ddlw_dropout_bwd(_p(dy), _p(mask), _p(dx),
                               ctypes.c_long(dy.numel()), ctypes.c_float(p),
                               ctypes.c_void_p(current_stream_ptr())),
          "dropout_bwd")
    return dx


def argmax_rows(logits: torch.Tensor) -> torch.Tensor:
    """K12: per-row argmax (first-max tie-break, like torch.argmax)."""
    lib = require_lib()
    lf = logits.float().contiguous()
    n, c = lf.shape
    out = torch.empty(n, dtype=torch.long, device=lf.device)
    check(lib.ddlw_argmax_rows(_p(lf), _p(out), ctypes.c_long(n),
                               ctypes.c_int(c),
                               ctypes.c_void_p(current_stream_ptr())),
          "argmax_rows")
    return out


def accuracy(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """K11: mean(argmax(logits) == labels) via per-block partial counts."""
    lib = require_lib()
    lf = logits.float().contiguous()
    n, c = lf.shape
    nblocks = (n + 3) // 4
    partial = torch.empty(nblocks, dtype=torch.int32, device=lf.device)
    check(lib.ddlw_accuracy(_p(lf), _p(labels.contiguous()), _p(partial),
                            ctypes.c_long(n), ctypes.c_int(c),
                            ctypes.c_void_p(current_stream_ptr())),
          "accuracy")
    return partial.sum().float() / n
