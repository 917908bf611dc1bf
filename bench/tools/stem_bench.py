import sys, time, os
sys.path.insert(0, "/root/repo")
import torch, torch.nn.functional as F
from ddlw_amd.ops import conv_gemm
dev = torch.device("cuda:0")
torch.manual_seed(0)
B = 256
x = torch.randn(B, 3, 224, 224, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
w = (torch.randn(64, 3, 7, 7, device=dev).to(torch.bfloat16) * 0.2).contiguous(memory_format=torch.channels_last)
y = conv_gemm.stem_fwd_kernel(x, w).float()
ref = F.conv2d(x.float(), w.float(), None, 2, 3)
print("rel err:", ((y - ref).abs().max() / ref.abs().max()).item())
for name, fn in (("stem", lambda: conv_gemm.stem_fwd_kernel(x, w)),
                 ("miopen", lambda: F.conv2d(x, w, None, 2, 3))):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter(); it = 20
    for _ in range(it): fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter()-t0)/it
    print(f"{name}: {dt*1e3:.3f} ms")
