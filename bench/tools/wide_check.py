#!/usr/bin/env python3
"""A/B + parity: wide (BM=256, 3-buf counted-vmcnt) vs base conv kernel."""
import os, sys, time
sys.path.insert(0, "/root/repo")
import torch
from ddlw_amd.ops import conv_gemm

def _cl(t): return t.contiguous(memory_format=torch.channels_last)

dev = torch.device("cuda:0")
shapes = [  # H, W, C, K, R, S, st, pad
    (56, 56, 64, 64, 3, 3, 1, 1),
    (28, 28, 128, 128, 3, 3, 1, 1),
    (14, 14, 256, 256, 3, 3, 1, 1),
    (7, 7, 512, 512, 3, 3, 1, 1),
    (14, 14, 1024, 256, 1, 1, 1, 0),
    (7, 7, 2048, 512, 1, 1, 1, 0),
    (14, 14, 256, 1024, 1, 1, 1, 0),
    (28, 28, 512, 128, 1, 1, 1, 0),
]
B = 256
for H, W, C, K, R, S, st, pad in shapes:
    torch.manual_seed(0)
    x = _cl(torch.randn(B, C, H, W, device=dev).to(torch.bfloat16))
    w = _cl(torch.randn(K, C, R, S, device=dev).to(torch.bfloat16) * 0.05)
    # fp32 reference
    ref = torch.nn.functional.conv2d(x.float(), w.float(), stride=st, padding=pad)
    res = {}
    for mode in ("0", "1"):
        os.environ["DDLW_CONV_WIDE"] = mode
        import ctypes
        conv_gemm.require_lib()  # reset static? env read once per proc... 
        y = conv_gemm.conv_fwd_kernel(x, w, st, pad)
        err = (y.float() - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
        for _ in range(3): conv_gemm.conv_fwd_kernel(x, w, st, pad)
        torch.cuda.synchronize(); t0 = time.perf_counter(); it = 20
        for _ in range(it): conv_gemm.conv_fwd_kernel(x, w, st, pad)
        torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / it
        Ho = (H + 2 * pad - R) // st + 1
        fl = 2.0 * B * Ho * Ho * K * C * R * S
        res[mode] = (dt * 1e3, fl / dt / 1e12, err)
    print(f"{H}x{W}x{C}->{K} {R}x{S}: base {res['0'][0]:.3f} ms {res['0'][1]:.0f} TF err {res['0'][2]:.2e} | "
          f"wide {res['1'][0]:.3f} ms {res['1'][1]:.0f} TF err {res['1'][2]:.2e}", flush=True)
