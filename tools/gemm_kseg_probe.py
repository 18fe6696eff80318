"""Probe gemm_m64_kseg (coalesced-A K-segmented GEMM) vs hipBLASLt on
the K-long decode shapes.  Sweeps ksegs; checks numerics vs fp32."""
import sys
import time
import pathlib

sys.path.insert(0, str(pathlib.Path(__file__).parent.parent))
import torch

from mlx_sharding_amd import ops

ext = ops.hip_ext()
dev = "cuda"
M = 64


def timeit(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


for (N, K) in [(8192, 28672), (4096, 14336), (28672, 8192)]:
    gb = N * K * 2 / 1e9
    torch.manual_seed(0)
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16) * 0.1
    w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
    t_mm = timeit(lambda: torch.mm(x, w.t()))
    line = [f"[{N}x{K}] blaslt={t_mm*1e6:.0f}us/{gb/t_mm/1000:.2f}TB"]
    ref = (x.float() @ w.t().float())
    for ks in (1, 2, 4, 8, 16):
        t = timeit(lambda: ext.gemm_m64_kseg(x, w, ks))
        got = ext.gemm_m64_kseg(x, w, ks).float()
        err = (got - ref).abs().max().item() / ref.abs().max().item()
        line.append(f"ks{ks}={t*1e6:.0f}us/{gb/t/1000:.2f}TB(e{err:.0e})")
    print(" ".join(line))

# ---- w4 variant: packed weights (4x less DRAM) ----------------------------
from mlx_sharding_amd.ops import reference as ref

print("\n-- w4 --")
for (N, K) in [(8192, 28672), (28672, 8192), (10240, 8192)]:
    torch.manual_seed(0)
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16) * 0.1
    w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
    wq, sc, bi = [t.to(dev) for t in ref.quantize(w.cpu(), 64, 4)]
    rp = ops.repack_w4(wq, 4)
    xf = x.to(torch.float16)
    dq = ref.dequantize(wq, sc, bi, 64, 4).float()
    refo = x.float() @ dq.t()
    gb = (N * K / 2 + 2 * N * K / 64 * 2) / 1e9  # packed + scales/biases
    t_mm = timeit(lambda: torch.mm(x, w.t()))
    line = [f"[{N}x{K}] blaslt(bf16)={t_mm*1e6:.0f}us"]
    for ks in (2, 4, 8, 16):
        t = timeit(lambda: ext.gemm_m64_kseg_w4(xf, rp, sc, bi, 64, ks))
        got = ext.gemm_m64_kseg_w4(xf, rp, sc, bi, 64, ks).float()
        err = (got - refo).abs().max().item() / refo.abs().max().item()
        line.append(f"ks{ks}={t*1e6:.0f}us/{gb/t/1000:.2f}TB(e{err:.0e})")
    print(" ".join(line))
