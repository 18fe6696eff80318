import sys, time
sys.path.insert(0, "/root/repo")
import torch
import torch.nn.functional as F

def timeit(fn, iters=30, warmup=10):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True); e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters): fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000

for M in (4, 16, 32):
    for (O, H, name) in [(10240, 8192, "l70-qkv"), (57344, 8192, "l70-gateup"),
                         (8192, 28672, "l70-down"), (3648, 2048, "ds-qkv"),
                         (102400, 2048, "ds-lmhead")]:
        w = torch.randn(O, H, dtype=torch.bfloat16, device="cuda") * 0.02
        x = torch.randn(M, H, dtype=torch.bfloat16, device="cuda")
        t = timeit(lambda: F.linear(x, w))
        bw = O * H * 2 / (t / 1e6) / 1e12
        print(f"M={M:3d} {name:12s} {t:8.1f} us  {bw:5.2f} TB/s")
