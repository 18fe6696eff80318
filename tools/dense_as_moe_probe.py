"""Probe: can the LDS-free bf16 MFMA MoE kernels (moe_gateup/down_mfma,
~4.5 TB/s on deepseek expert tiles) sustain their weight-stream rate on
llama-70B dense MLP shapes ([28672, 8192]) when driven as ONE expert
with 4x16-token sub-ranges?  If yes, routing the dense MLP through them
beats hipBLASLt's ~3.4 TB/s on these M=64 tall-skinny shapes."""
import sys
import time
import pathlib

sys.path.insert(0, str(pathlib.Path(__file__).parent.parent))
import torch

from mlx_sharding_amd import ops

ext = ops.hip_ext()
dev = "cuda"
B = 64
H = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
I = int(sys.argv[2]) if len(sys.argv) > 2 else 28672

torch.manual_seed(0)
x = torch.randn(B, H, device=dev, dtype=torch.bfloat16) * 0.1
gate_w = torch.randn(1, I, H, device=dev, dtype=torch.bfloat16) * 0.02
up_w = torch.randn(1, I, H, device=dev, dtype=torch.bfloat16) * 0.02
down_w = torch.randn(1, H, I, device=dev, dtype=torch.bfloat16) * 0.02

S = B // 16
sub_e = torch.zeros(S, dtype=torch.int32, device=dev)
sub_off = torch.arange(0, B, 16, dtype=torch.int32, device=dev)
sub_cnt = torch.full((S,), 16, dtype=torch.int32, device=dev)
sorted_tok = torch.arange(B, dtype=torch.int32, device=dev)
sorted_wt = torch.ones(B, dtype=torch.float32, device=dev)


def timeit(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


t_gu = timeit(lambda: ext.moe_gateup_grouped(
    x, gate_w, up_w, sub_e, sub_off, sub_cnt, sorted_tok, B, 16))
h = ext.moe_gateup_grouped(x, gate_w, up_w, sub_e, sub_off, sub_cnt,
                           sorted_tok, B, 16)
t_dn = timeit(lambda: ext.moe_down_grouped(
    h, down_w, sub_e, sub_off, sub_cnt, sorted_tok, sorted_wt, B, 16))

gu_gb = 2 * I * H * 2 / 1e9
dn_gb = I * H * 2 / 1e9
print(f"gateup [2x{I}x{H}]: {t_gu*1e6:.0f} us  -> {gu_gb/t_gu/1000:.2f} TB/s")
print(f"down   [{H}x{I}]:   {t_dn*1e6:.0f} us  -> {dn_gb/t_dn/1000:.2f} TB/s")

# hipBLASLt comparison on the same shapes (gate as plain mm)
wg = gate_w[0]
t_mm = timeit(lambda: x @ wg.t())
print(f"blaslt [{I}x{H}] mm: {t_mm*1e6:.0f} us -> {I*H*2/1e9/t_mm/1000:.2f} TB/s")
wd = down_w[0]
t_md = timeit(lambda: h @ wd.t())
print(f"blaslt [{H}x{I}] mm: {t_md*1e6:.0f} us -> {dn_gb/t_md/1000:.2f} TB/s")

# numerics sanity vs torch
ref = (torch.nn.functional.silu((x.float() @ wg.t().float()))
       * (x.float() @ up_w[0].t().float())) @ wd.t().float()
got = ext.moe_down_grouped(h, down_w, sub_e, sub_off, sub_cnt, sorted_tok,
                           sorted_wt, B, 16)
cos = torch.nn.functional.cosine_similarity(
    ref.flatten(), got.float().flatten(), dim=0).item()
print(f"cos vs fp32 torch: {cos:.5f}")
