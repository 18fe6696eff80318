"""Tiny standalone run of moe_w4_mfma for PMC collection."""
import torch

from mlx_sharding_amd import ops

ext = ops.hip_ext()
E, H, I, N, K = 64, 2048, 1408, 64, 6
dev = "cuda"
torch.manual_seed(0)
x = torch.randn(N, H, dtype=torch.bfloat16, device=dev)
wq = torch.randint(0, 2**31 - 1, (E, I, H // 8), device=dev, dtype=torch.int32)
sc = torch.rand(E, I, H // 64, dtype=torch.bfloat16, device=dev) * 0.01
bi = torch.rand(E, I, H // 64, dtype=torch.bfloat16, device=dev) * 0.01
logits = torch.randn(N, E, dtype=torch.bfloat16, device=dev)
subs = ops.moe_gate_subranges(logits, K, max_tok=32)
s_e, s_off, s_cnt, s_tok, _ = subs
P = N * K
for _ in range(30):
    h = ext.moe_w4_mfma(x, wq, sc, bi, s_e, s_off, s_cnt, s_tok, P, 64, 4)
torch.cuda.synchronize()
print("done")
