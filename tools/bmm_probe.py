"""MoE prefill expert-GEMM probe: hipBLASLt bmm runs ~15% MFU at
[E=64, cap~1664, 2048] x [E, 2048, 1408].  Alternatives measured here."""
import sys
import pathlib

sys.path.insert(0, str(pathlib.Path(__file__).parent.parent))
import torch


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000


E, CAP, H, I = 64, 1664, 2048, 1408
xp = torch.randn(E, CAP, H, dtype=torch.bfloat16, device="cuda")
w = torch.randn(E, I, H, dtype=torch.bfloat16, device="cuda") * 0.02
wt = w.transpose(1, 2).contiguous()          # [E, H, I]
flop = 2.0 * E * CAP * H * I

t = timeit(lambda: torch.bmm(xp, wt))
print(f"bmm contiguous-B          {t:9.1f} us  {flop/(t*1e6)/1e3:6.0f} TF/s")

# transposed-B strided bmm memory-faults in this torch/hipBLASLt build
# (round-1 finding, see ops._cached_t) — not probed.

# single same-FLOP 2D GEMM for the MFU ceiling comparison
x2 = torch.randn(E * CAP, H, dtype=torch.bfloat16, device="cuda")
w2 = torch.randn(I, H, dtype=torch.bfloat16, device="cuda")
t = timeit(lambda: torch.nn.functional.linear(x2, w2))
print(f"same-FLOP 2D linear       {t:9.1f} us  {flop/(t*1e6)/1e3:6.0f} TF/s")

t = timeit(lambda: torch.mm(x2, w2.t()))
print(f"same-FLOP 2D mm(x, w.t()) {t:9.1f} us  {flop/(t*1e6)/1e3:6.0f} TF/s")

# per-expert mm loop (64 launches)
outs = torch.empty(E, CAP, I, dtype=torch.bfloat16, device="cuda")


def loop_mm():
    for e in range(E):
        torch.mm(xp[e], wt[e], out=outs[e])


t = timeit(loop_mm)
print(f"per-expert mm loop        {t:9.1f} us  {flop/(t*1e6)/1e3:6.0f} TF/s")

# grouped mm if this torch build has it
if hasattr(torch, "_grouped_mm"):
    try:
        offs = torch.arange(1, E + 1, device="cuda", dtype=torch.int32) * CAP
        xg = xp.reshape(E * CAP, H)
        t = timeit(lambda: torch._grouped_mm(xg, wt, offs=offs))
        print(f"_grouped_mm               {t:9.1f} us  {flop/(t*1e6)/1e3:6.0f} TF/s")
    except Exception as exc:  # noqa: BLE001
        print("_grouped_mm unavailable:", str(exc)[:120])
else:
    print("torch._grouped_mm: not present")

# wide-N combined gate|up vs two separate bmms (CAP at batch-64 scale)
CAP2 = 3200
xp2 = torch.randn(E, CAP2, H, dtype=torch.bfloat16, device="cuda")
wt1 = torch.randn(E, H, I, dtype=torch.bfloat16, device="cuda")
wt2 = torch.randn(E, H, I, dtype=torch.bfloat16, device="cuda")
wtc = torch.cat([wt1, wt2], dim=2).contiguous()
flop2 = 2.0 * E * CAP2 * H * I
t = timeit(lambda: (torch.bmm(xp2, wt1), torch.bmm(xp2, wt2)))
print(f"2x bmm [cap {CAP2}]        {t:9.1f} us  {2*flop2/(t*1e-6)/1e12:6.0f} TF/s")
t = timeit(lambda: torch.bmm(xp2, wtc))
print(f"combined bmm [2I wide]    {t:9.1f} us  {2*flop2/(t*1e-6)/1e12:6.0f} TF/s")

# in-model vs isolated: bmm right after a scatter into a FRESH xp
xp3 = torch.zeros(E, CAP2, H, dtype=torch.bfloat16, device="cuda")
src = torch.randn(E * CAP2 // 2, H, dtype=torch.bfloat16, device="cuda")
idx = torch.arange(E * CAP2 // 2, device="cuda", dtype=torch.int32)
import mlx_sharding_amd.ops as O
ext2 = O.hip_ext()

def scatter_then_bmm():
    ext2.moe_scatter_rows(src, xp3.view(-1, H), idx, idx)
    return torch.bmm(xp3, wtc)

t = timeit(scatter_then_bmm, iters=10, warmup=3)
print(f"scatter+combined bmm      {t:9.1f} us (bmm alone was ~2122)")
