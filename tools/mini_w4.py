import sys
sys.path.insert(0, "/root/repo")
import torch
from mlx_sharding_amd import ops
ext = ops.hip_ext()
torch.manual_seed(0)
E, H, I, N, K = 64, 2048, 1408, 32, 6
x = torch.randn(N, H, dtype=torch.bfloat16, device="cuda")
wq = torch.randint(0, 2**31 - 1, (E, I, H // 8), device="cuda", dtype=torch.int32)
sc = torch.rand(E, I, H // 64, dtype=torch.bfloat16, device="cuda") * 0.01
bi = torch.rand(E, I, H // 64, dtype=torch.bfloat16, device="cuda") * 0.01
logits = torch.randn(N, E, dtype=torch.bfloat16, device="cuda")
s32 = ops.moe_gate_subranges(logits, K, max_tok=32)
P = N * K
for _ in range(3):
    y = ext.moe_w4_mfma(x, wq, sc, bi, *s32[:4], P, 64, 4)
# dense lm_head shape too
O = 102400
wq2 = torch.randint(0, 2**31 - 1, (O, H // 8), device="cuda", dtype=torch.int32)
sc2 = torch.rand(O, H // 64, dtype=torch.bfloat16, device="cuda") * 0.01
bi2 = torch.rand(O, H // 64, dtype=torch.bfloat16, device="cuda") * 0.01
for _ in range(3):
    y2 = ext.w4a16_gemv(x, wq2, sc2, bi2, 64, 4)
torch.cuda.synchronize()
print("done")
