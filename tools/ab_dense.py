"""A/B: hipBLASLt F.linear vs the MFMA dense GEMV at decode shapes."""
import sys
import time

import torch

from mlx_sharding_amd import ops

ext = ops.hip_ext()


def t(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


SHAPES = {
    "lite": [(3072, 2048, "qkv"), (2048, 2048, "o_proj"), (576, 2048, "kv_a"),
             (5632, 2048, "sh_gu"), (2048, 2816, "sh_down"),
             (102400, 2048, "lm_head")],
    "llama70": [(10240, 8192, "qkv"), (8192, 8192, "o_proj"),
                (57344, 8192, "gate_up"), (8192, 28672, "down"),
                (128256, 8192, "lm_head")],
}

for (O, H, name) in SHAPES.get(sys.argv[1] if len(sys.argv) > 1 else "lite"):
    x = torch.randn(64, H, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(O, H, dtype=torch.bfloat16, device="cuda") * 0.1
    a = t(lambda: torch.nn.functional.linear(x, w))
    b = t(lambda: ext.dense_gemv(x, w))
    bw = O * H * 2 / 1e12
    print(f"{name:8s} O={O:6d} H={H:5d}  blaslt {a:7.1f}us ({bw/(a/1e6):.2f}TB/s)"
          f"  mfma {b:7.1f}us ({bw/(b/1e6):.2f}TB/s)")
