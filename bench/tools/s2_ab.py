import sys, time
sys.path.insert(0, "/root/repo")
import torch
from ddlw_amd.ops import conv_gemm
dev = torch.device("cuda:0")
def _cl(t): return t.contiguous(memory_format=torch.channels_last)
B=256
for (H,C,K) in ((56,128,128),(28,256,256),(14,512,512)):
    st, pad = 2, 1
    torch.manual_seed(0)
    w = _cl(torch.randn(K,C,3,3,device=dev).to(torch.bfloat16)*0.1)
    Ho=(H+2*pad-3)//st+1
    dy = _cl(torch.randn(B,K,Ho,Ho,device=dev).to(torch.bfloat16))
    dx = conv_gemm.conv_dgrad_kernel(dy, w, (B,C,H,H), pad, st).float()
    ref = torch.nn.grad.conv2d_input((B,C,H,H), w.float(), dy.float(), stride=st, padding=pad)
    err = ((dx-ref).abs().max()/(ref.abs().max()+1e-6)).item()
    for _ in range(3): conv_gemm.conv_dgrad_kernel(dy, w, (B,C,H,H), pad, st)
    torch.cuda.synchronize(); t0=time.perf_counter(); it=20
    for _ in range(it): conv_gemm.conv_dgrad_kernel(dy, w, (B,C,H,H), pad, st)
    torch.cuda.synchronize(); dt=(time.perf_counter()-t0)/it
    fl = 2.0*B*Ho*Ho*K*C*9
    # stock
    for _ in range(3): torch.nn.grad.conv2d_input((B,C,H,H), w, dy, stride=st, padding=pad)
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(20): torch.nn.grad.conv2d_input((B,C,H,H), w, dy, stride=st, padding=pad)
    torch.cuda.synchronize(); dts=(time.perf_counter()-t0)/20
    print(f"{H}x{H}x{C}<-{K} 3x3s2 dgrad: merged {fl/dt/1e12:.0f}TF err {err:.0e} | stock {fl/dts/1e12:.0f}TF")
