"""Probe hipBLASLt layout/dtype variants on the llama-70B M=64 decode
GEMM shapes.  The down-proj [8192 out, 28672 k] runs at only 3.3 TB/s
via the default x @ w.t(); if a pre-transposed or fp16 variant reaches
the gate-shape's 4.9 TB/s, wiring that in is a free ~10% on 70B decode."""
import sys
import time
import pathlib

sys.path.insert(0, str(pathlib.Path(__file__).parent.parent))
import torch

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


for (N, K) in [(8192, 28672), (28672, 8192), (10240, 8192)]:
    gb = N * K * 2 / 1e9
    for dt in (torch.bfloat16, torch.float16):
        x = torch.randn(M, K, device=dev, dtype=dt) * 0.1
        w = torch.randn(N, K, device=dev, dtype=dt) * 0.02
        wt = w.t().contiguous()          # [K, N] row-major
        out = torch.empty(M, N, device=dev, dtype=dt)
        variants = {
            "x@w.t()": lambda: torch.mm(x, w.t()),
            "x@wt":    lambda: torch.mm(x, wt),
            "linear":  lambda: torch.nn.functional.linear(x, w),
            "mm_out":  lambda: torch.mm(x, w.t(), out=out),
            "(w@x.t()).t()": lambda: torch.mm(w, x.t()),
        }
        res = {k: timeit(f) for k, f in variants.items()}
        best = min(res, key=res.get)
        line = " ".join(f"{k}={v*1e6:.0f}us/{gb/v/1000:.2f}TB" for k, v in res.items())
        print(f"[{N}x{K}] {str(dt)[6:]}: {line}  BEST={best}")
