import sys, time
sys.path.insert(0, "/root/repo")
import os, torch
from ddlw_amd.ops import conv_gemm
import torch.nn.functional as F
dev = torch.device("cuda:0")
def _cl(t): return t.contiguous(memory_format=torch.channels_last)
shapes = [(56,56,64,64,1,1,1),(56,56,64,64,3,3,1),(56,56,64,256,1,1,1),(56,56,256,64,1,1,1)]
B=256
for H,W,C,K,R,S,st in shapes:
    pad = 1 if R==3 else 0
    torch.manual_seed(0)
    x = _cl(torch.randn(B,C,H,W,device=dev).to(torch.bfloat16))
    Ho=(H+2*pad-R)//st+1
    dy = _cl(torch.randn(B,K,Ho,Ho,device=dev).to(torch.bfloat16))
    ref = torch.nn.grad.conv2d_weight(x.float(), (K,C,R,S), dy.float(), stride=st, padding=pad)
    out = {}
    for bm in ("64","128"):
        os.environ["DDLW_WGRAD_BM"] = bm
        dw = conv_gemm.conv_wgrad_kernel(dy, x, (K,C,R,S), st, pad).float()
        err = ((dw-ref).abs().max()/(ref.abs().max()+1e-6)).item()
        for _ in range(3): conv_gemm.conv_wgrad_kernel(dy, x, (K,C,R,S), st, pad)
        torch.cuda.synchronize(); t0=time.perf_counter(); it=20
        for _ in range(it): conv_gemm.conv_wgrad_kernel(dy, x, (K,C,R,S), st, pad)
        torch.cuda.synchronize(); dt=(time.perf_counter()-t0)/it
        fl = 2.0*B*Ho*Ho*K*C*R*S
        out[bm]=(dt*1e3, fl/dt/1e12, err)
    # stock
    for _ in range(3): torch.nn.grad.conv2d_weight(x, (K,C,R,S), dy, stride=st, padding=pad)
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(20): torch.nn.grad.conv2d_weight(x, (K,C,R,S), dy, stride=st, padding=pad)
    torch.cuda.synchronize(); dts=(time.perf_counter()-t0)/20
    print(f"{H}x{W}x{C}->{K} {R}x{S}: bm64 {out['64'][1]:.0f}TF e{out['64'][2]:.0e} | bm128 {out['128'][1]:.0f}TF e{out['128'][2]:.0e} | stock {fl/dts/1e12:.0f}TF")
