#!/usr/bin/env python3
"""Per-kernel timing microbench (within-probe A/B, guide §5.4 rule 24).

Times the hand-written kernels at DeepSeek-Coder-V2-Lite decode shapes
with hipEvent brackets.  Run on a GPU box:
    python tools/bench_kernels.py [--iters 50]
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

import torch


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters * 1000  # us


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    from mlx_sharding_amd import ops
    from mlx_sharding_amd.ops import reference as ref
    ext = ops.hip_ext()
    assert ext is not None
    dev = "cuda"
    torch.manual_seed(0)

    E, H, I, N, K = 64, 2048, 1408, 64, 6
    x = torch.randn(N, H, dtype=torch.bfloat16, device=dev)
    gw = torch.randn(E, I, H, dtype=torch.bfloat16, device=dev) * 0.03
    uw = torch.randn(E, I, H, dtype=torch.bfloat16, device=dev) * 0.03
    dw = torch.randn(E, H, I, dtype=torch.bfloat16, device=dev) * 0.03
    logits = torch.randn(N, E, dtype=torch.bfloat16, device=dev)

    subs = ops.moe_gate_subranges(logits, K, max_tok=4)
    sub_e, sub_off, sub_cnt, sorted_tok, sorted_wt = subs
    P = N * K
    h = ext.moe_gateup_grouped(x, gw, uw, sub_e, sub_off, sub_cnt, sorted_tok,
                               P, 4)

    t = timeit(lambda: ext.moe_gateup_grouped(x, gw, uw, sub_e, sub_off,
                                              sub_cnt, sorted_tok, P, 4),
               args.iters)
    bw = 64 * (2 * I * H * 2) / (t / 1e6) / 1e12
    print(f"moe_gateup_grouped      {t:8.1f} us   ~{bw:.2f} TB/s wt")

    t = timeit(lambda: ext.moe_down_grouped(h, dw, sub_e, sub_off, sub_cnt,
                                            sorted_tok, sorted_wt, N, 4),
               args.iters)
    bw = 64 * (H * I * 2) / (t / 1e6) / 1e12
    print(f"moe_down_grouped        {t:8.1f} us   ~{bw:.2f} TB/s wt")

    subs16 = ops.moe_gate_subranges(logits, K, max_tok=16)
    s16_e, s16_off, s16_cnt, s16_tok, s16_wt = subs16
    t = timeit(lambda: ext.moe_gateup_grouped(x, gw, uw, s16_e, s16_off,
                                              s16_cnt, s16_tok, P, 16),
               args.iters)
    bw = 64 * (2 * I * H * 2) / (t / 1e6) / 1e12
    print(f"moe_gateup_mfma16       {t:8.1f} us   ~{bw:.2f} TB/s wt")
    t = timeit(lambda: ext.moe_down_grouped(h, dw, s16_e, s16_off, s16_cnt,
                                            s16_tok, s16_wt, N, 16),
               args.iters)
    bw = 64 * (H * I * 2) / (t / 1e6) / 1e12
    print(f"moe_down_mfma16         {t:8.1f} us   ~{bw:.2f} TB/s wt")

    t = timeit(lambda: ops.moe_gate_subranges(logits, K), args.iters)
    print(f"moe_gate_subranges      {t:8.1f} us")

    # w4 variants
    wq = torch.randint(0, 2**31 - 1, (E, I, H // 8), device=dev, dtype=torch.int32)
    sc = torch.rand(E, I, H // 64, dtype=torch.bfloat16, device=dev) * 0.01
    bi = torch.rand(E, I, H // 64, dtype=torch.bfloat16, device=dev) * 0.01
    subs32 = ops.moe_gate_subranges(logits, K, max_tok=32)
    s32_e, s32_off, s32_cnt, s32_tok, _ = subs32
    t = timeit(lambda: ext.moe_w4_mfma(x, wq, sc, bi, s32_e, s32_off,
                                       s32_cnt, s32_tok, P, 64, 4), args.iters)
    bw = 64 * (I * H // 2) / (t / 1e6) / 1e12
    print(f"moe_w4_mfma             {t:8.1f} us   ~{bw:.2f} TB/s wt")

    # round-2 fp16-dequant kernels (the production quant decode path)
    x16 = x.to(torch.float16)
    uq2 = torch.randint(0, 2**31 - 1, (E, I, H // 8), device=dev,
                        dtype=torch.int32)
    dq2 = torch.randint(0, 2**31 - 1, (E, H, I // 8), device=dev,
                        dtype=torch.int32)
    dsc = torch.rand(E, H, I // 64, dtype=torch.bfloat16, device=dev) * 0.01
    dbi = torch.rand(E, H, I // 64, dtype=torch.bfloat16, device=dev) * 0.01
    usc = sc.clone(); ubi = bi.clone()
    gq = ops.repack_w4(wq, 4); uq = ops.repack_w4(uq2, 4)
    dq = ops.repack_w4(dq2, 4)
    hh16 = ext.moe_w4f16_gateup(x16, gq, uq, sc, bi, usc, ubi, s16_e, s16_off,
                                s16_cnt, s16_tok, P, 64, 4)
    t = timeit(lambda: ext.moe_w4f16_gateup(x16, gq, uq, sc, bi, usc, ubi,
                                            s16_e, s16_off, s16_cnt, s16_tok,
                                            P, 64, 4), args.iters)
    bw = 64 * (2 * I * H // 2) / (t / 1e6) / 1e12
    print(f"moe_w4f16_gateup        {t:8.1f} us   ~{bw:.2f} TB/s wt(packed)")
    t = timeit(lambda: ext.moe_w4f16_down(hh16, dq, dsc, dbi, s16_e, s16_off,
                                          s16_cnt, s16_tok, s16_wt, N, 64, 4),
               args.iters)
    bw = 64 * (H * I // 2) / (t / 1e6) / 1e12
    print(f"moe_w4f16_down          {t:8.1f} us   ~{bw:.2f} TB/s wt(packed)")

    for (O, HH, name) in [(3648, 2048, "qkv"), (102400, 2048, "lm_head"),
                          (5632, 2048, "sh_gateup"), (2048, 2816, "sh_down")]:
        wq2 = torch.randint(0, 2**31 - 1, (O, HH // 8), device=dev, dtype=torch.int32)
        sc2 = torch.rand(O, HH // 64, dtype=torch.bfloat16, device=dev) * 0.01
        bi2 = torch.rand(O, HH // 64, dtype=torch.bfloat16, device=dev) * 0.01
        x2 = torch.randn(N, HH, dtype=torch.bfloat16, device=dev)
        t = timeit(lambda: ext.w4a16_gemv(x2, wq2, sc2, bi2, 64, 4), args.iters)
        bw = (O * HH // 2) / (t / 1e6) / 1e12
        print(f"w4a16_gemv {name:10s}  {t:8.1f} us   ~{bw:.2f} TB/s wt")
        rp2 = ops.repack_w4(wq2, 4)
        x216 = x2.to(torch.float16)
        t = timeit(lambda: ext.w4f16_gemv(x216, rp2, sc2, bi2, 64, 4),
                   args.iters)
        bw = O * HH // 2 / (t / 1e6) / 1e12
        print(f"w4f16_gemv {name:10s}  {t:8.1f} us   ~{bw:.2f} TB/s wt")

    # attention decode, MLA shape
    for S in (512, 2048):
        B, Hq, Dk, Dv = 32, 16, 192, 128
        Scap = ((S + 1023) // 1024) * 1024
        q = torch.randn(B, Hq, 1, Dk, dtype=torch.bfloat16, device=dev)
        kb = torch.randn(B, Hq, Scap, Dk, dtype=torch.bfloat16, device=dev)
        vb = torch.randn(B, Hq, Scap, Dv, dtype=torch.bfloat16, device=dev)
        kv_, vv_ = kb[:, :, :S], vb[:, :, :S]
        t = timeit(lambda: ext.attn_decode(q, kv_, vv_, 0.1, 0.0, 0, None),
                   args.iters)
        bw = B * Hq * S * (Dk + Dv) * 2 / (t / 1e6) / 1e12
        print(f"attn_decode S={S:5d}     {t:8.1f} us   ~{bw:.2f} TB/s kv")

    # attention decode, absorbed-MLA shape (compressed cache, V in K rows)
    for S in (512, 2048):
        B, Hq, Dk, Dv = 64, 16, 576, 512
        Scap = ((S + 1023) // 1024) * 1024
        q = torch.randn(B, Hq, 1, Dk, dtype=torch.bfloat16, device=dev)
        kb = torch.randn(B, 1, Scap, Dk, dtype=torch.bfloat16, device=dev)
        kv_ = kb[:, :, :S]
        vv_ = kv_[..., :Dv]
        t = timeit(lambda: ext.attn_decode(q, kv_, vv_, 0.04, 0.0, 0, None),
                   args.iters)
        bw = B * S * Dk * 2 / (t / 1e6) / 1e12
        print(f"attn_decode/abs S={S:5d} {t:8.1f} us   ~{bw:.2f} TB/s kv")

    # attention prefill, MLA shape
    B, Hq, T, Dk, Dv = 32, 16, 512, 192, 128
    q = torch.randn(B, Hq, T, Dk, dtype=torch.bfloat16, device=dev) * 0.3
    kb = torch.randn(B, Hq, 1024, Dk, dtype=torch.bfloat16, device=dev) * 0.3
    vb = torch.randn(B, Hq, 1024, Dv, dtype=torch.bfloat16, device=dev) * 0.3
    kv_, vv_ = kb[:, :, :T], vb[:, :, :T]
    t = timeit(lambda: ext.attn_prefill(q, kv_, vv_, 0.072, 0.0, 0, 0),
               iters=10)
    fl = 2 * B * Hq * T * T * (Dk + Dv) / 2  # causal ~half
    print(f"attn_prefill T=512      {t:8.1f} us   ~{fl/(t/1e6)/1e12:.1f} TF")

    # rmsnorm
    xr = torch.randn(N, H, dtype=torch.bfloat16, device=dev)
    wr = torch.randn(H, dtype=torch.bfloat16, device=dev)
    t = timeit(lambda: ext.rms_norm(xr, wr, 1e-6, 0.0), args.iters)
    print(f"rms_norm [32,2048]      {t:8.1f} us")


if __name__ == "__main__":
    main()
