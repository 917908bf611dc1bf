import sys, time
sys.path.insert(0, "/root/repo")
import torch
from ddlw_amd.ops import binding
import torch.nn.functional as F
dev = torch.device("cuda:0")
def _cl(t): return t.contiguous(memory_format=torch.channels_last)
torch.manual_seed(0)
x = _cl(torch.randn(256, 64, 112, 112, device=dev).to(torch.bfloat16))
y, am = binding.maxpool3x3s2_fwd(x)
dy = _cl(torch.randn_like(y))
dx = binding.maxpool3x3s2_bwd(dy, am, x.shape).float()
# fp32 reference
x32 = x.float().detach().requires_grad_(True)
y32 = F.max_pool2d(x32, 3, 2, 1)
y32.backward(dy.float())
err = ((dx - x32.grad).abs().max() / (x32.grad.abs().max() + 1e-6)).item()
print("rel err:", err)
for _ in range(3): binding.maxpool3x3s2_bwd(dy, am, x.shape)
torch.cuda.synchronize(); t0 = time.perf_counter(); it = 50
for _ in range(it): binding.maxpool3x3s2_bwd(dy, am, x.shape)
torch.cuda.synchronize()
print(f"bwd: {(time.perf_counter()-t0)/it*1e6:.1f} us")
