"""Latency-vs-throughput probe: run the w4 MFMA kernel on 1 vs 2 streams.
If two concurrent copies take ~the same wall as one, the kernel leaves
the chip mostly idle (latency-bound); if ~2x, it is resource-bound."""
import sys, time
sys.path.insert(0, "/root/repo")
import torch
from mlx_sharding_amd import ops
ext = ops.hip_ext()
torch.manual_seed(0)
O, H, M = 102400, 2048, 32
x = torch.randn(M, H, dtype=torch.bfloat16, device="cuda")
wq = torch.randint(0, 2**31 - 1, (O, H // 8), device="cuda", dtype=torch.int32)
sc = torch.rand(O, H // 64, dtype=torch.bfloat16, device="cuda") * 0.01
bi = torch.rand(O, H // 64, dtype=torch.bfloat16, device="cuda") * 0.01

def run(n_streams, iters=20):
    streams = [torch.cuda.Stream() for _ in range(n_streams)]
    for s in streams:
        with torch.cuda.stream(s):
            ext.w4a16_gemv(x, wq, sc, bi, 64, 4)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        for s in streams:
            with torch.cuda.stream(s):
                ext.w4a16_gemv(x, wq, sc, bi, 64, 4)
    torch.cuda.synchronize()
    return (time.time() - t0) / iters * 1e6

t1 = run(1)
t2 = run(2)
t4 = run(4)
print(f"1 stream: {t1:.0f} us;  2 streams: {t2:.0f} us ({t2/t1:.2f}x);  4 streams: {t4:.0f} us ({t4/t1:.2f}x)")
