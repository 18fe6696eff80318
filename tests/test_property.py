"""Property-based tests (hypothesis) for the pure-Python subsystems:
wire codec round-trips, stop-sequence machinery, affine quantization
error bounds, top-p sampling support, and streaming detokenization.

These harden surfaces the reference leaves untested (it has no tests at
all) against the adversarial inputs a serving deployment
actually sees: odd shapes, 0-d/empty tensors, overlapping stop strings,
multi-byte unicode."""

import pytest
import torch
from hypothesis import given, settings, strategies as st

from mlx_sharding_amd.ops import reference as ref
from mlx_sharding_amd.parallel import wire

SETTINGS = dict(max_examples=30, deadline=None)


# ---------------------------------------------------------------------------
# wire codec
# ---------------------------------------------------------------------------

@given(shape=st.lists(st.integers(1, 8), min_size=1, max_size=4),
       dtype=st.sampled_from([torch.float32, torch.float16, torch.bfloat16,
                              torch.int32, torch.int64]))
@settings(**SETTINGS)
def test_wire_roundtrip_identity(shape, dtype):
    if dtype.is_floating_point:
        t = torch.randn(shape).to(dtype)
    else:
        t = torch.randint(-1000, 1000, shape, dtype=dtype)
    out = wire.msg_to_tensor(wire.tensor_to_msg(t))
    assert out.dtype == t.dtype and out.shape == t.shape
    assert torch.equal(out, t)


@given(shape=st.lists(st.integers(1, 6), min_size=1, max_size=3))
@settings(**SETTINGS)
def test_wire_fp16_downcast_on_bf16(shape):
    t = torch.randn(shape, dtype=torch.bfloat16)
    out = wire.msg_to_tensor(wire.tensor_to_msg(t, wire_fp16=True))
    assert out.dtype == torch.float16
    assert torch.allclose(out.float(), t.float(), atol=2e-3, rtol=2e-3)


# ---------------------------------------------------------------------------
# stop-sequence machinery
# ---------------------------------------------------------------------------

@given(tokens=st.lists(st.integers(0, 9), max_size=12),
       stop=st.lists(st.integers(0, 9), min_size=1, max_size=4))
@settings(**SETTINGS)
def test_stopping_criteria_matches_suffix(tokens, stop):
    from mlx_sharding_amd.server.openai_api import stopping_criteria
    met, trim = stopping_criteria(tokens, [stop], eos_token_id=None)
    is_suffix = len(tokens) >= len(stop) and tokens[-len(stop):] == stop
    assert met == is_suffix
    if met:
        assert trim == len(stop)


@given(s1=st.lists(st.integers(0, 3), max_size=8),
       s2=st.lists(st.integers(0, 3), min_size=1, max_size=8))
@settings(**SETTINGS)
def test_sequence_overlap_definition(s1, s2):
    from mlx_sharding_amd.server.openai_api import sequence_overlap
    expected = any(s1[-i:] == s2[:i]
                   for i in range(1, min(len(s1), len(s2)) + 1))
    assert sequence_overlap(s1, s2) == expected


# ---------------------------------------------------------------------------
# affine quantization error bound
# ---------------------------------------------------------------------------

@given(rows=st.integers(1, 8), groups=st.integers(1, 4),
       gs=st.sampled_from([32, 64]), bits=st.sampled_from([4, 8]))
@settings(**SETTINGS)
def test_quantize_error_bounded_by_half_scale(rows, groups, gs, bits):
    w = torch.randn(rows, groups * gs, dtype=torch.bfloat16)
    wq, sc, bi = ref.quantize(w, gs, bits)
    dq = ref.dequantize(wq, sc, bi, gs, bits).float()
    err = (dq - w.float()).abs().view(rows, groups, gs)
    # per-group |error| <= scale/2 + bf16 rounding slack
    bound = sc.float().abs().view(rows, groups, 1) * 0.5 + 0.02
    assert bool((err <= bound).all())


@given(rows=st.integers(1, 4), gs=st.sampled_from([32, 64]),
       bits=st.sampled_from([4, 8]))
@settings(**SETTINGS)
def test_quantize_constant_rows_are_exact(rows, gs, bits):
    w = torch.full((rows, 2 * gs), 0.5, dtype=torch.bfloat16)
    wq, sc, bi = ref.quantize(w, gs, bits)
    dq = ref.dequantize(wq, sc, bi, gs, bits)
    assert torch.allclose(dq.float(), w.float(), atol=1e-2)


# ---------------------------------------------------------------------------
# sampling
# ---------------------------------------------------------------------------

@given(seed=st.integers(0, 2**31 - 1),
       top_p=st.floats(0.1, 0.95))
@settings(**SETTINGS)
def test_top_p_samples_only_from_nucleus(seed, top_p):
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(1, 64, generator=g) * 3
    probs = torch.softmax(logits[0], dim=-1)
    order = torch.argsort(probs, descending=True)
    cum = torch.cumsum(probs[order], dim=0)
    # the nucleus: smallest prefix with mass >= top_p
    k = int((cum < top_p).sum().item()) + 1
    nucleus = set(order[:k].tolist())
    for trial in range(5):
        tok = int(ref.sample(logits, temperature=1.0, top_p=top_p,
                             generator=torch.Generator().manual_seed(
                                 seed + trial)).item())
        assert tok in nucleus


@given(seed=st.integers(0, 2**31 - 1))
@settings(**SETTINGS)
def test_temperature_zero_is_argmax(seed):
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(1, 100, generator=g)
    tok = int(ref.sample(logits, temperature=0.0).item())
    assert tok == int(logits.argmax().item())


# ---------------------------------------------------------------------------
# streaming detokenizer (byte-level BPE, multi-byte unicode)
# ---------------------------------------------------------------------------

@pytest.fixture(scope="module")
def bpe_tokenizer(tmp_path_factory):
    from tokenizers import Tokenizer
    from tokenizers.models import BPE
    from tokenizers.pre_tokenizers import ByteLevel as PreByteLevel
    from tokenizers.decoders import ByteLevel as DecByteLevel
    from tokenizers.trainers import BpeTrainer
    from transformers import PreTrainedTokenizerFast

    tok = Tokenizer(BPE(unk_token=None))
    tok.pre_tokenizer = PreByteLevel()
    tok.decoder = DecByteLevel()
    trainer = BpeTrainer(vocab_size=400, special_tokens=["<eos>"],
                         initial_alphabet=PreByteLevel.alphabet())
    corpus = ["hello world", "héllo wörld", "日本語 テスト", "emoji 🚀 test",
              "the quick brown fox", "años 123"]
    tok.train_from_iterator(corpus * 4, trainer)
    return PreTrainedTokenizerFast(tokenizer_object=tok, eos_token="<eos>")


@given(text=st.text(
    alphabet=st.sampled_from("helo wrd日本語🚀añ é123"), min_size=1,
    max_size=24))
@settings(**SETTINGS)
def test_streaming_detok_equals_full_decode(bpe_tokenizer, text):
    from mlx_sharding_amd.utils.detokenizer import StreamingDetokenizer
    ids = bpe_tokenizer.encode(text)
    d = StreamingDetokenizer(bpe_tokenizer)
    out = "".join(d.add_token(t) for t in ids) + d.finalize()
    assert out == bpe_tokenizer.decode(ids)


# ---------------------------------------------------------------------------
# KV cache invariants
# ---------------------------------------------------------------------------

@given(appends=st.lists(st.integers(1, 700), min_size=1, max_size=6))
@settings(max_examples=20, deadline=None)
def test_kvcache_grow_preserves_content(appends):
    """Growing across CHUNK boundaries must preserve previously
    appended K/V exactly."""
    from mlx_sharding_amd.ops.kvcache import KVCache
    c = KVCache(2, 16, 8)
    chunks = []
    for t in appends:
        k = torch.randn(1, 2, t, 16, dtype=torch.bfloat16)
        v = torch.randn(1, 2, t, 8, dtype=torch.bfloat16)
        c.update(k, v)
        chunks.append((k, v))
    k_all = torch.cat([k for k, _ in chunks], dim=2)
    v_all = torch.cat([v for _, v in chunks], dim=2)
    assert torch.equal(c.k, k_all)
    assert torch.equal(c.v, v_all)
    assert c.offset == sum(appends)


@given(n1=st.integers(1, 40), trim_to=st.integers(0, 40),
       n2=st.integers(1, 20))
@settings(max_examples=20, deadline=None)
def test_kvcache_trim_then_append(n1, trim_to, n2):
    """trim(n) + append must equal a fresh cache fed prefix+suffix
    (the prefix-cache invariant at cache level)."""
    from mlx_sharding_amd.ops.kvcache import KVCache
    trim_to = min(trim_to, n1)
    k1 = torch.randn(1, 2, n1, 8, dtype=torch.bfloat16)
    v1 = torch.randn(1, 2, n1, 8, dtype=torch.bfloat16)
    k2 = torch.randn(1, 2, n2, 8, dtype=torch.bfloat16)
    v2 = torch.randn(1, 2, n2, 8, dtype=torch.bfloat16)

    c = KVCache(2, 8, 8)
    c.update(k1, v1)
    c.trim(trim_to)
    c.update(k2, v2)

    ref = KVCache(2, 8, 8)
    ref.update(k1[:, :, :trim_to], v1[:, :, :trim_to])
    ref.update(k2, v2)
    assert c.offset == ref.offset
    assert torch.equal(c.k, ref.k)
    assert torch.equal(c.v, ref.v)


# ---------------------------------------------------------------------------
# splitter key routing: partition property
# ---------------------------------------------------------------------------

@given(total=st.integers(2, 12), cuts=st.lists(st.integers(1, 11),
                                               min_size=1, max_size=3))
@settings(max_examples=25, deadline=None)
def test_route_key_partitions_exactly(total, cuts):
    """Any multi-stage split routes every layer key to EXACTLY one
    stage, embeddings to the first, norm/head to the last."""
    from mlx_sharding_amd.utils.loading import _route_key
    bounds = sorted({0, total, *[min(c, total) for c in cuts]})
    stages = list(zip(bounds[:-1], bounds[1:]))
    keys = [f"model.layers.{i}.self_attn.q_proj.weight" for i in range(total)]
    keys += ["model.embed_tokens.weight", "model.norm.weight", "lm_head.weight"]
    for key in keys:
        owners = [i for i, (s, e) in enumerate(stages)
                  if _route_key(key, s, e, total)]
        assert len(owners) == 1, f"{key} owned by {owners} ({stages})"
