"""Tiny standalone run of the fp16-dequant w4 kernels for PMC collection
(gateup + down + dense gemv at DeepSeek-V2-Lite decode shapes)."""
import sys, pathlib
sys.path.insert(0, str(pathlib.Path(__file__).parent.parent))
import torch

from mlx_sharding_amd import ops

ext = ops.hip_ext()
E, H, I, N, K = 64, 2048, 1408, 64, 6
dev = "cuda"
torch.manual_seed(0)
x = torch.randn(N, H, dtype=torch.float16, device=dev)
mk = lambda o, i: (  # noqa: E731
    ops.repack_w4(torch.randint(0, 2**31 - 1, (E, o, i // 8), device=dev,
                                dtype=torch.int32), 4),
    torch.rand(E, o, i // 64, dtype=torch.bfloat16, device=dev) * 0.01,
    torch.rand(E, o, i // 64, dtype=torch.bfloat16, device=dev) * 0.01)
g, u, d = mk(I, H), mk(I, H), mk(H, I)
logits = torch.randn(N, E, dtype=torch.bfloat16, device=dev)
s_e, s_off, s_cnt, s_tok, s_wt = ops.moe_gate_subranges(logits, K, max_tok=16)
P = N * K
wqd = ops.repack_w4(torch.randint(0, 2**31 - 1, (5632, H // 8), device=dev,
                                  dtype=torch.int32), 4)
scd = torch.rand(5632, H // 64, dtype=torch.bfloat16, device=dev) * 0.01
bid = torch.rand(5632, H // 64, dtype=torch.bfloat16, device=dev) * 0.01
for _ in range(30):
    hh = ext.moe_w4f16_gateup(x, g[0], u[0], g[1], g[2], u[1], u[2],
                              s_e, s_off, s_cnt, s_tok, P, 64, 4)
    out = ext.moe_w4f16_down(hh, d[0], d[1], d[2], s_e, s_off, s_cnt,
                             s_tok, s_wt, N, 64, 4)
    y = ext.w4f16_gemv(x, wqd, scd, bid, 64, 4)
torch.cuda.synchronize()
print("done", hh.shape, out.shape, y.shape)
