"""Root-cause probe: hipBLASLt GEMMs measure ~3x slower under hipGraph
replay than eager (round-1 anomaly, docs/PERFORMANCE.md).

Hypotheses tested per shape (llama-70B / deepseek decode projections):
  A. eager F.linear                      (baseline)
  B. graph-replayed F.linear             (the anomaly)
  C. graph-replayed F.linear, TunableOp  (algo pinned before capture)
  D. eager dense_gemv (our MFMA kernel)
  E. graph-replayed dense_gemv
Run with PYTORCH_TUNABLEOP_ENABLED=1 to add C.
"""
import sys
import pathlib

sys.path.insert(0, str(pathlib.Path(__file__).parent.parent))
import torch
import torch.nn.functional as F

from mlx_sharding_amd import ops

ext = ops.hip_ext()


def timeit(fn, iters=50, warmup=10):
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


def graph_timeit(fn, iters=50):
    sstr = torch.cuda.Stream()
    sstr.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(sstr):
        for _ in range(3):
            fn()
    torch.cuda.current_stream().wait_stream(sstr)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        fn()
    return timeit(lambda: g.replay(), iters=iters)


M = 64
for (O, H, name) in [(10240, 8192, "l70-qkv"), (57344, 8192, "l70-gateup"),
                     (28672 // 2 * 2, 8192, "l70-up-ish"),
                     (8192, 28672, "l70-down"), (2048, 8192, "l70-o"),
                     (3648, 2048, "ds-qkv"), (102400, 2048, "ds-lmhead")]:
    w = (torch.randn(O, H, dtype=torch.bfloat16, device="cuda") * 0.02)
    x = torch.randn(M, H, dtype=torch.bfloat16, device="cuda")
    a = timeit(lambda: F.linear(x, w))
    b = graph_timeit(lambda: F.linear(x, w))
    d = timeit(lambda: ext.dense_gemv(x, w))
    e = graph_timeit(lambda: ext.dense_gemv(x, w))
    gb = O * H * 2 / 1e12
    print(f"{name:12s} eager {a:7.1f}us ({gb/(a/1e6):4.1f}TB/s) | graph "
          f"{b:7.1f}us x{b/a:4.2f} | mfma {d:7.1f}us | mfma-graph {e:7.1f}us",
          flush=True)


# --- k-split decomposition probe for the slow deep-k shapes ------------
print("\nk-split probe (M=64):", flush=True)
for (O, H, name) in [(8192, 28672, "l70-down"), (2048, 8192, "l70-o"),
                     (57344, 8192, "l70-gateup")]:
    w = (torch.randn(O, H, dtype=torch.bfloat16, device="cuda") * 0.02)
    x = torch.randn(M, H, dtype=torch.bfloat16, device="cuda")
    base = timeit(lambda: F.linear(x, w))
    for nk in (2, 4):
        ws = [w[:, i * H // nk:(i + 1) * H // nk].contiguous()
              for i in range(nk)]
        xs = [x[:, i * H // nk:(i + 1) * H // nk].contiguous()
              for i in range(nk)]

        def split_fn(ws=ws, xs=xs, nk=nk):
            y = F.linear(xs[0], ws[0])
            for i in range(1, nk):
                y = y + F.linear(xs[i], ws[i])
            return y
        t = timeit(split_fn)
        gb = O * H * 2 / 1e12
        print(f"  {name:12s} nk={nk}: {t:7.1f}us ({gb/(t/1e6):4.1f}TB/s) "
              f"vs base {base:7.1f}us", flush=True)


# --- operand-layout probe for the slow down-proj shape -----------------
print("\nlayout probe (M=64):", flush=True)
for (O, H, name) in [(8192, 28672, "l70-down"), (2048, 8192, "l70-o")]:
    w = (torch.randn(O, H, dtype=torch.bfloat16, device="cuda") * 0.02)
    wt = w.t().contiguous()  # [H, O]
    x = torch.randn(M, H, dtype=torch.bfloat16, device="cuda")
    base = timeit(lambda: F.linear(x, w))
    tmm = timeit(lambda: torch.mm(x, wt))
    gb = O * H * 2 / 1e12
    print(f"  {name:10s} linear {base:7.1f}us ({gb/(base/1e6):4.1f}TB/s) | "
          f"mm(x, w^T) {tmm:7.1f}us ({gb/(tmm/1e6):4.1f}TB/s)", flush=True)


# --- LDS-tiled dense_gemm64 probe -------------------------------------
print("\ndense_gemm64 probe (M=64):", flush=True)
for (O, H, name) in [(8192, 28672, "l70-down"), (57344, 8192, "l70-gateup"),
                     (10240, 8192, "l70-qkv"), (102400, 2048, "ds-lmhead")]:
    w = (torch.randn(O, H, dtype=torch.bfloat16, device="cuda") * 0.02)
    x = torch.randn(M, H, dtype=torch.bfloat16, device="cuda")
    base = timeit(lambda: F.linear(x, w))
    tg = timeit(lambda: ext.dense_gemm64(x, w))
    gb = O * H * 2 / 1e12
    print(f"  {name:10s} blaslt {base:7.1f}us ({gb/(base/1e6):4.1f}TB/s) | "
          f"gemm64 {tg:7.1f}us ({gb/(tg/1e6):4.1f}TB/s)", flush=True)
