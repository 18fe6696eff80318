"""Standalone gemm_kseg run for PMC collection ([8192, 28672], ks4)."""
import sys, pathlib
sys.path.insert(0, str(pathlib.Path(__file__).parent.parent))
import torch
from mlx_sharding_amd import ops

ext = ops.hip_ext()
dev = "cuda"
M, N, K = 64, 8192, 28672
torch.manual_seed(0)
x = torch.randn(M, K, device=dev, dtype=torch.bfloat16) * 0.1
w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
for _ in range(30):
    y = ext.gemm_m64_kseg(x, w, 4)
torch.cuda.synchronize()
print("done", y.shape)
