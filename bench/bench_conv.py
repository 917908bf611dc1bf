#!/usr/bin/env python3
"""Per-shape conv microbenchmark: ddlw MFMA implicit-GEMM vs library (MIOpen).

For every unique conv layer shape of ResNet-50 @224 (the BASELINE model):
  1. parity-check the ddlw fwd (and stride-1 dgrad) kernel against the fp32
     stock op;
  2. time both paths;
  3. emit ``ddlw_amd/ops/conv_dispatch.json`` routing each (shape, direction)
     to the measured winner.

Run on an MI355X:  python bench/bench_conv.py [--batch 256] [--iters 10]
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time
from pathlib import Path

import torch
import torch.nn.functional as F

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from ddlw_amd.ops import conv_gemm  # noqa: E402

# (H, W, C, K, R, S, stride) — unique ResNet-50 conv shapes (stem excluded:
# C=3 unsupported by the implicit-GEMM tile, routed to the library)
RESNET50_SHAPES = [
    (56, 56, 64, 64, 1, 1, 1),
    (56, 56, 64, 64, 3, 3, 1),
    (56, 56, 64, 256, 1, 1, 1),
    (56, 56, 256, 64, 1, 1, 1),
    (56, 56, 256, 128, 1, 1, 1),
    (56, 56, 128, 128, 3, 3, 2),
    (56, 56, 256, 512, 1, 1, 2),
    (28, 28, 512, 128, 1, 1, 1),
    (28, 28, 128, 128, 3, 3, 1),
    (28, 28, 128, 512, 1, 1, 1),
    (28, 28, 512, 256, 1, 1, 1),
    (28, 28, 256, 256, 3, 3, 2),
    (28, 28, 512, 1024, 1, 1, 2),
    (14, 14, 1024, 256, 1, 1, 1),
    (14, 14, 256, 256, 3, 3, 1),
    (14, 14, 256, 1024, 1, 1, 1),
    (14, 14, 1024, 512, 1, 1, 1),
    (14, 14, 512, 512, 3, 3, 2),
    (14, 14, 1024, 2048, 1, 1, 2),
    (7, 7, 2048, 512, 1, 1, 1),
    (7, 7, 512, 512, 3, 3, 1),
    (7, 7, 512, 2048, 1, 1, 1),
]


def _cl(t):
    return t.contiguous(memory_format=torch.channels_last)


def time_fn(fn, iters: int, warmup: int = 3) -> float:
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--out", type=str, default=str(conv_gemm._DISPATCH_PATH))
    ap.add_argument("--parity-only", action="store_true")
    args = ap.parse_args()
    dev = torch.device("cuda:0")
    table = {}
    report = []
    pad_of = {1: 0, 3: 1, 7: 3}
    for (H, W, C, K, R, S, st) in RESNET50_SHAPES:
        pad = pad_of[R]
        B = args.batch
        key = conv_gemm._key(B, H, W, C, K, R, S, st)
        Ho = (H + 2 * pad - R) // st + 1
        flops = 2.0 * B * Ho * Ho * K * C * R * S

        # ---------- parity at small batch ----------
        torch.manual_seed(0)
        xs = _cl(torch.randn(4, C, H, W, device=dev).to(torch.bfloat16))
        ws = _cl(torch.randn(K, C, R, S, device=dev).to(torch.bfloat16))
        ref = F.conv2d(xs.float(), ws.float(), None, st, pad)
        mine = conv_gemm.conv_fwd_kernel(xs, ws, st, pad).float()
        err = (mine - ref).abs().max().item()
        scale = ref.abs().max().item() + 1e-6
        fwd_ok = err / scale < 5e-2
        dgrad_ok = None
        dy4 = _cl(torch.randn(4, K, Ho, Ho, device=dev).to(torch.bfloat16))
        if conv_gemm.dgrad_supported(C, K, st, R, pad):
            ref_dx = torch.nn.grad.conv2d_input(
                xs.shape, ws.float(), dy4.float(), stride=st, padding=pad
            )
            dx = conv_gemm.conv_dgrad_kernel(dy4, ws, xs.shape, pad, st).float()
            derr = (dx - ref_dx).abs().max().item() / (ref_dx.abs().max().item() + 1e-6)
            dgrad_ok = derr < 5e-2
        # wgrad parity (split-K fp32 slabs -> bf16 dW)
        ref_dw = torch.nn.grad.conv2d_weight(
            xs.float(), ws.shape, dy4.float(), stride=st, padding=pad
        )
        dwh = conv_gemm.conv_wgrad_kernel(dy4, xs, ws.shape, st, pad).float()
        werr = (dwh - ref_dw).abs().max().item() / (ref_dw.abs().max().item() + 1e-6)
        wgrad_ok = werr < 5e-2

        if args.parity_only:
            report.append({"key": key, "fwd_ok": fwd_ok, "fwd_err": err / scale,
                           "dgrad_ok": dgrad_ok})
            print(report[-1], flush=True)
            continue

        # ---------- timing at full batch ----------
        x = _cl(torch.randn(B, C, H, W, device=dev).to(torch.bfloat16))
        w = _cl(torch.randn(K, C, R, S, device=dev).to(torch.bfloat16))
        t_stock_f = time_fn(lambda: F.conv2d(x, w, None, st, pad), args.iters)
        t_hip_f = time_fn(lambda: conv_gemm.conv_fwd_kernel(x, w, st, pad), args.iters) if fwd_ok else None

        dy = _cl(torch.randn(B, K, Ho, Ho, device=dev).to(torch.bfloat16))
        t_stock_d = time_fn(
            lambda: torch.ops.aten.convolution_backward(
                dy, x, w, None, [st, st], [pad, pad], [1, 1], False, [0, 0], 1,
                [True, False, False])[0],
            args.iters,
        )
        t_hip_d = (
            time_fn(lambda: conv_gemm.conv_dgrad_kernel(dy, w, x.shape, pad, st), args.iters)
            if dgrad_ok
            else None
        )
        t_stock_w = time_fn(
            lambda: torch.ops.aten.convolution_backward(
                dy, x, w, None, [st, st], [pad, pad], [1, 1], False, [0, 0], 1,
                [False, True, False])[1],
            args.iters,
        )
        t_hip_w = (
            time_fn(lambda: conv_gemm.conv_wgrad_kernel(dy, x, w.shape, st, pad), args.iters)
            if wgrad_ok
            else None
        )
        ent = {
            "fwd": "hip" if (t_hip_f is not None and t_hip_f < t_stock_f) else "stock",
            "dgrad": "hip" if (t_hip_d is not None and t_hip_d < t_stock_d) else "stock",
            "wgrad": "hip" if (t_hip_w is not None and t_hip_w < t_stock_w) else "stock",
        }
        table[key] = ent
        rec = {
            "key": key,
            "fwd_ok": fwd_ok,
            "dgrad_ok": dgrad_ok,
            "wgrad_ok": wgrad_ok,
            "tflops_stock_fwd": round(flops / t_stock_f / 1e9, 1),
            "tflops_hip_fwd": round(flops / t_hip_f / 1e9, 1) if t_hip_f else None,
            "tflops_stock_dgrad": round(flops / t_stock_d / 1e9, 1),
            "tflops_hip_dgrad": round(flops / t_hip_d / 1e9, 1) if t_hip_d else None,
            "tflops_stock_wgrad": round(flops / t_stock_w / 1e9, 1),
            "tflops_hip_wgrad": round(flops / t_hip_w / 1e9, 1) if t_hip_w else None,
            "route": ent,
        }
        report.append(rec)
        print(json.dumps(rec), flush=True)

    if not args.parity_only:
        Path(args.out).write_text(json.dumps(table, indent=2))
        print(f"wrote {args.out}")
    Path("gpurun_out").mkdir(exist_ok=True)
    Path("gpurun_out/conv_report.json").write_text(json.dumps(report, indent=2))


if __name__ == "__main__":
    main()
