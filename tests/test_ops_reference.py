"""Unit tests for the torch reference ops (fp32-reference semantics)."""


import torch

from mlx_sharding_amd.ops import reference as ref


def test_rms_norm_matches_manual():
    x = torch.randn(2, 5, 64)
    w = torch.randn(64)
    y = ref.rms_norm(x, w, 1e-5)
    expect = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-5) * w
    assert torch.allclose(y, expect, atol=1e-5)


def test_rms_norm_offset_gemma():
    x = torch.randn(1, 3, 16)
    w = torch.zeros(16)
    y = ref.rms_norm(x, w, 1e-6, weight_offset=1.0)
    yn = ref.rms_norm(x, torch.ones(16), 1e-6)
    assert torch.allclose(y, yn, atol=1e-6)


def test_rope_rotation_norm_preserving():
    x = torch.randn(1, 4, 2, 8)
    inv = ref.rope_freqs(8)
    cos, sin = ref.rope_cos_sin(torch.arange(4), inv)
    for inter in (False, True):
        y = ref.apply_rope(x, cos, sin, interleaved=inter)
        # rotation preserves pairwise norms
        assert torch.allclose(y.norm(dim=-1), x.norm(dim=-1), atol=1e-4)
    # position 0 is identity
    y0 = ref.apply_rope(x[:, :1], cos[:1], sin[:1])
    assert torch.allclose(y0, x[:, :1], atol=1e-5)


def test_attention_vs_sdpa():
    q = torch.randn(2, 4, 6, 16)
    k = torch.randn(2, 4, 6, 16)
    v = torch.randn(2, 4, 6, 16)
    out = ref.attention(q, k, v, scale=0.25)
    expect = torch.nn.functional.scaled_dot_product_attention(
        q, k, v, is_causal=True, scale=0.25)
    assert torch.allclose(out, expect, atol=1e-5)


def test_attention_gqa_offset():
    # decode step: Tq=1 over 5 cached keys, GQA 4:2
    q = torch.randn(1, 4, 1, 8)
    k = torch.randn(1, 2, 5, 8)
    v = torch.randn(1, 2, 5, 8)
    out = ref.attention(q, k, v, scale=1.0, causal_offset=4)
    kk = k.repeat_interleave(2, dim=1)
    vv = v.repeat_interleave(2, dim=1)
    s = torch.softmax((q.float() @ kk.float().transpose(-1, -2)), dim=-1)
    assert torch.allclose(out, (s @ vv.float()).to(out.dtype), atol=1e-5)


def test_attention_sliding_window():
    q = torch.randn(1, 1, 6, 8)
    k = torch.randn(1, 1, 6, 8)
    v = torch.randn(1, 1, 6, 8)
    out = ref.attention(q, k, v, scale=1.0, sliding_window=2)
    # last query attends only keys 5 and 4
    s = (q.float() @ k.float().transpose(-1, -2))[0, 0, 5]
    s[:4] = float("-inf")
    expect = torch.softmax(s, -1) @ v[0, 0].float()
    assert torch.allclose(out[0, 0, 5], expect.to(out.dtype), atol=1e-5)


def test_quantize_roundtrip():
    for bits, gs in [(4, 32), (4, 64), (8, 32)]:
        w = torch.randn(16, 128, dtype=torch.bfloat16)
        wq, sc, bi = ref.quantize(w, gs, bits)
        wd = ref.dequantize(wq, sc, bi, gs, bits)
        step = (w.float().reshape(16, -1, gs).max(-1).values
                - w.float().reshape(16, -1, gs).min(-1).values) / (2 ** bits - 1)
        tol = step.max().item() * 0.75 + 0.05
        assert (w.float() - wd.float()).abs().max().item() < tol


def test_quantized_linear_close_to_dense():
    torch.manual_seed(0)
    w = torch.randn(32, 64, dtype=torch.bfloat16) * 0.1
    x = torch.randn(3, 64, dtype=torch.bfloat16)
    wq, sc, bi = ref.quantize(w, 32, 4)
    y = ref.quantized_linear(x, wq, sc, bi, 32, 4)
    yd = x @ w.t()
    assert (y.float() - yd.float()).abs().max().item() < 0.5


def test_moe_gate_greedy():
    logits = torch.tensor([[1.0, 5.0, 3.0, 0.0]])
    w, idx = ref.moe_gate(logits, top_k=2)
    assert idx[0].tolist() == [1, 2]
    probs = torch.softmax(logits, -1)
    assert torch.allclose(w[0], probs[0, [1, 2]])


def test_moe_gate_group_limited():
    # 2 groups of 2 experts; pick best group then top-k within it
    logits = torch.tensor([[10.0, 0.0, 9.0, 8.9]])
    w, idx = ref.moe_gate(logits, top_k=2, n_group=2, topk_group=1)
    assert set(idx[0].tolist()) == {0, 1}


def test_grouped_expert_mlp_matches_loop():
    torch.manual_seed(0)
    E, H, I, N, K = 4, 8, 16, 5, 2
    x = torch.randn(N, H)
    gw = torch.randn(E, I, H) * 0.1
    uw = torch.randn(E, I, H) * 0.1
    dw = torch.randn(E, H, I) * 0.1
    wts = torch.rand(N, K)
    idx = torch.randint(0, E, (N, K))
    out = ref.grouped_expert_mlp(x, gw, uw, dw, wts, idx)
    expect = torch.zeros(N, H)
    for n in range(N):
        for kk in range(K):
            e = idx[n, kk].item()
            h = ref.swiglu(x[n] @ gw[e].t(), x[n] @ uw[e].t())
            expect[n] += wts[n, kk] * (h @ dw[e].t())
    assert torch.allclose(out, expect, atol=1e-4)


def test_sampling_greedy_and_topp():
    logits = torch.tensor([[0.0, 2.0, 1.0]])
    assert ref.sample(logits).item() == 1
    g = torch.Generator().manual_seed(0)
    # top_p tiny → always the argmax
    for _ in range(5):
        assert ref.sample(logits, temperature=1.0, top_p=1e-6, generator=g).item() == 1


def test_repetition_penalty():
    logits = torch.tensor([[2.0, -2.0, 1.0]])
    out = ref.apply_repetition_penalty(logits, torch.tensor([0, 1]), 2.0)
    assert torch.allclose(out, torch.tensor([[1.0, -4.0, 1.0]]))


def test_softcap():
    x = torch.randn(4, 8) * 100
    y = ref.softcap(x, 30.0)
    assert y.abs().max().item() <= 30.0
    small = torch.randn(4, 8) * 0.01
    assert torch.allclose(ref.softcap(small, 30.0), small, atol=1e-4)


def test_yarn_freqs_shape():
    inv = ref.rope_freqs(64, 10000.0, {"type": "yarn", "factor": 40.0,
                                       "original_max_position_embeddings": 4096})
    base = ref.rope_freqs(64)
    assert inv.shape == base.shape
    # interpolated (low) freqs shrink, high freqs mostly preserved
    assert inv[-1] < base[-1]
    assert torch.allclose(inv[0], base[0], rtol=1e-3)


def test_make_expert_subranges():
    from mlx_sharding_amd.ops import make_expert_subranges
    torch.manual_seed(0)
    N, K, E, MT = 7, 3, 6, 4
    idx = torch.randint(0, E, (N, K))
    wts = torch.rand(N, K)
    sub_e, sub_off, sub_cnt, sorted_tok, sorted_wt, order = \
        make_expert_subranges(idx, wts, E, MT)
    P = N * K
    flat_e = idx.reshape(-1)
    # reconstruct (expert, token, weight) triples from sub-ranges
    triples = []
    for s in range(len(sub_e)):
        for t in range(int(sub_cnt[s])):
            p = int(sub_off[s]) + t
            triples.append((int(sub_e[s]), int(sorted_tok[p]),
                            float(sorted_wt[p])))
    assert len(triples) == P
    expect = sorted((int(flat_e[p]), p // K, float(wts.reshape(-1)[p]))
                    for p in range(P))
    assert sorted(triples) == expect
    # every sub-range stays within one expert's run and <= MT tokens
    assert int(sub_cnt.max()) <= MT


def test_make_expert_subranges_empty_experts():
    from mlx_sharding_amd.ops import make_expert_subranges
    idx = torch.tensor([[5, 5], [5, 0]])  # experts 1-4 empty
    wts = torch.ones(2, 2)
    sub_e, sub_off, sub_cnt, sorted_tok, _, _ = \
        make_expert_subranges(idx, wts, 8, 4)
    total = int(sub_cnt.sum())
    assert total == 4
    used = {int(sub_e[s]) for s in range(len(sub_e)) if int(sub_cnt[s]) > 0}
    assert used == {0, 5}


def test_dq_cache_budget(monkeypatch):
    """The dequant-residency budget refuses allocations over the cap and
    counts accepted bytes."""
    from mlx_sharding_amd import ops as O
    monkeypatch.setattr(O, "_DQ_CACHE_BYTES", 0)
    monkeypatch.setenv("MLXS_AMD_DQ_CACHE_GB", "0.000001")  # ~1 KB
    assert O._dq_cache_ok(512)
    assert not O._dq_cache_ok(10_000)  # over cap now
    monkeypatch.setenv("MLXS_AMD_NO_DQ_CACHE", "1")
    assert not O._dq_cache_ok(1)


def test_streaming_detokenizer_holds_partial_utf8():
    """Deltas for multi-byte characters are held until complete
    (reference relies on mlx_lm's streaming detokenizer for this)."""
    from mlx_sharding_amd.utils.detokenizer import StreamingDetokenizer

    class ByteTok:  # token id == one utf-8 byte
        def decode(self, ids):
            return bytes(ids).decode("utf-8", errors="replace")

    det = StreamingDetokenizer(ByteTok())
    out = []
    for b in "héllo".encode("utf-8"):  # é = 2 bytes
        out.append(det.add_token(b))
    assert "".join(out) == "héllo"
    assert "�" not in "".join(out)
    # the partial é byte must have produced an empty delta
    assert "" in out
    det.reset()
    assert det.add_token(ord("x")) == "x"
    assert det.finalize() == ""


def test_repack_w4_kernel_unpack_roundtrip():
    """repack_w4's nibble/byte interleave must invert exactly under the
    moe_w4f16 kernel's unpack pattern: pair j of a repacked word is
    ((w >> 4j) & 0x000F000F) -> (elem 2j, elem 2j+1)."""
    from mlx_sharding_amd.ops import repack_w4
    torch.manual_seed(3)
    wdtype = torch.uint32 if hasattr(torch, "uint32") else torch.int32
    wq = torch.randint(-2 ** 31, 2 ** 31 - 1, (5, 7), dtype=torch.int32) \
        .view(wdtype)
    rp = repack_w4(wq, 4).view(torch.int32).to(torch.int64) & 0xFFFFFFFF
    orig = wq.view(torch.int32).to(torch.int64) & 0xFFFFFFFF
    for j in range(4):
        pair = (rp >> (4 * j)) & 0x000F000F
        lo, hi = pair & 0xF, (pair >> 16) & 0xF
        assert torch.equal(lo, (orig >> (8 * j)) & 0xF)        # elem 2j
        assert torch.equal(hi, (orig >> (8 * j + 4)) & 0xF)    # elem 2j+1
    assert repack_w4(wq, 4) is repack_w4(wq, 4)  # cached

    rp8 = repack_w4(wq.clone().view(wdtype), 8).view(torch.int32) \
        .to(torch.int64) & 0xFFFFFFFF
    for j in range(2):
        pair = (rp8 >> (8 * j)) & 0x00FF00FF
        lo, hi = pair & 0xFF, (pair >> 16) & 0xFF
        assert torch.equal(lo, (orig >> (16 * j)) & 0xFF)      # elem 2j
        assert torch.equal(hi, (orig >> (16 * j + 8)) & 0xFF)  # elem 2j+1


def test_fp16_or_trick_dequant_exact():
    """The fp16 magic-number dequant ((q | 0x6400) - 1032)*s + (b+8s)
    must match s*q + b to fp16 rounding of the FINAL value only."""
    import numpy as np
    q = np.arange(16, dtype=np.uint16)
    magic = (q | 0x6400).view(np.float16).astype(np.float64)
    assert np.array_equal(magic, 1024.0 + q)          # OR trick exact
    v2 = (magic.astype(np.float16) - np.float16(1032.0)).astype(np.float64)
    assert np.array_equal(v2, q - 8.0)                # recenter exact
    for s, b in [(0.037, -0.21), (1.5e-3, 9e-3), (0.11, 0.4)]:
        s16 = np.float16(s)
        b2 = np.float16(b + 8.0 * float(s16))
        got = (v2.astype(np.float16) * s16 + b2).astype(np.float64)
        want = float(s16) * q + float(np.float16(b + 8 * float(s16))) - \
            8.0 * float(s16)
        # single-rounding bound: error ≤ 1 ulp of the result magnitude
        ref = s * q + b
        tol = max(abs(ref).max(), 8 * s) * 2 ** -9
        assert np.abs(got - ref).max() < max(tol, 2 * abs(b - float(np.float16(b))))
