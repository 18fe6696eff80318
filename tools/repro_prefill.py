import sys, time
sys.path.insert(0, "/root/repo")
import torch
from mlx_sharding_amd import ops
torch.manual_seed(0)
E, H, I, K = 64, 2048, 1408, 6
N = 16384
x = torch.randn(N, H, dtype=torch.bfloat16, device="cuda")
gw = torch.randn(E, I, H, dtype=torch.bfloat16, device="cuda") * 0.03
uw = torch.randn(E, I, H, dtype=torch.bfloat16, device="cuda") * 0.03
dw = torch.randn(E, H, I, dtype=torch.bfloat16, device="cuda") * 0.03
wts = torch.rand(N, K, dtype=torch.bfloat16, device="cuda")
idx = torch.randint(0, E, (N, K), device="cuda")
torch.cuda.synchronize(); t0 = time.time()
out = ops.grouped_expert_mlp(x, gw, uw, dw, wts, idx)
torch.cuda.synchronize(); print("first call", time.time()-t0)
t0 = time.time()
for _ in range(3):
    out = ops.grouped_expert_mlp(x, gw, uw, dw, wts, idx)
torch.cuda.synchronize(); print("per call", (time.time()-t0)/3)
print(out.shape, out.dtype, torch.isfinite(out.float()).all().item())
