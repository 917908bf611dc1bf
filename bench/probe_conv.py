#!/usr/bin/env python3
"""Focused probe of ddlw conv shapes (run under rocprofv3 for PMC/trace)."""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from ddlw_amd.ops import conv_gemm  # noqa: E402


def _cl(t):
    return t.contiguous(memory_format=torch.channels_last)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--shapes", type=str, default="64-256,256-64,2048-512")
    ap.add_argument("--batch", type=int, default=256)
    args = ap.parse_args()
    dev = torch.device("cuda:0")
    catalog = {
        "64-256": (56, 56, 64, 256, 1, 1, 1),
        "256-64": (56, 56, 256, 64, 1, 1, 1),
        "2048-512": (7, 7, 2048, 512, 1, 1, 1),
        "512-2048": (7, 7, 512, 2048, 1, 1, 1),
        "3x3s1": (28, 28, 128, 128, 3, 3, 1),
    }
    for name in args.shapes.split(","):
        H, W, C, K, R, S, st = catalog[name]
        pad = 1 if R == 3 else 0
        x = _cl(torch.randn(args.batch, C, H, W, device=dev).to(torch.bfloat16))
        w = _cl(torch.randn(K, C, R, S, device=dev).to(torch.bfloat16))
        for _ in range(3):
            conv_gemm.conv_fwd_kernel(x, w, st, pad)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            conv_gemm.conv_fwd_kernel(x, w, st, pad)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        Ho = (H + 2 * pad - R) // st + 1
        fl = 2.0 * args.batch * Ho * Ho * K * C * R * S
        print(f"{name}: {dt*1e3:.3f} ms  {fl/dt/1e12:.1f} TF", flush=True)


if __name__ == "__main__":
    main()
