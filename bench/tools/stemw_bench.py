import sys, time
sys.path.insert(0, "/root/repo")
import torch
from ddlw_amd.ops import conv_gemm
dev = torch.device("cuda:0")
def _cl(t): return t.contiguous(memory_format=torch.channels_last)
B=256
torch.manual_seed(0)
x = _cl(torch.randn(B,3,224,224,device=dev).to(torch.bfloat16))
dy = _cl(torch.randn(B,64,112,112,device=dev).to(torch.bfloat16))
dw = conv_gemm.stem_wgrad_kernel(dy, x).float()
ref = torch.nn.grad.conv2d_weight(x.float(), (64,3,7,7), dy.float(), stride=2, padding=3)
print("rel err:", ((dw-ref).abs().max()/(ref.abs().max()+1e-6)).item())
for name, fn in (("stem_wgrad", lambda: conv_gemm.stem_wgrad_kernel(dy, x)),
                 ("miopen", lambda: torch.nn.grad.conv2d_weight(x, (64,3,7,7), dy, stride=2, padding=3))):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0=time.perf_counter(); it=20
    for _ in range(it): fn()
    torch.cuda.synchronize(); print(f"{name}: {(time.perf_counter()-t0)/it*1e3:.3f} ms")
