"""Model-level tests: PP=1 vs PP=N bitwise parity (the reference's most
valuable invariant), cache lifecycle, quantized path."""

import pytest
import torch

from mlx_sharding_amd.models import get_model_class
from mlx_sharding_amd.config import QuantConfig

from conftest import init_model


def _run_pp(cls, cfg, splits, ids, n_decode=3):
    """Run prefill+decode through a chain of stages; return all logits."""
    stages = []
    full = init_model(cls, cfg, cfg.shard(0, cfg.num_hidden_layers))
    sd = full.state_dict()
    for (s, e) in splits:
        st = cls(cfg, cfg.shard(s, e))
        st.load_weights(sd)
        st.eval()
        stages.append(st)
    caches = [st.make_cache() for st in stages]
    outs = []
    x = ids
    with torch.no_grad():
        for step in range(n_decode + 1):
            h = x
            for st, c in zip(stages, caches):
                h = st(h, c)
            outs.append(h)
            x = h[:, -1, :].argmax(-1, keepdim=True)
    return torch.cat([o[:, -1:, :] for o in outs], dim=1)


@pytest.mark.parametrize("arch,cfg_fixture", [
    ("llama", "tiny_llama_config"),
    ("gemma2", "tiny_gemma2_config"),
    ("deepseek_v2", "tiny_deepseek_config"),
])
def test_pp_parity(arch, cfg_fixture, request):
    cfg = request.getfixturevalue(cfg_fixture)
    cls = get_model_class(arch)
    n = cfg.num_hidden_layers
    torch.manual_seed(1)
    ids = torch.randint(0, cfg.vocab_size, (1, 6))
    ref = _run_pp(cls, cfg, [(0, n)], ids)
    mid = n // 2
    pp2 = _run_pp(cls, cfg, [(0, mid), (mid, n)], ids)
    assert torch.equal(ref, pp2), "PP=2 logits diverge from PP=1"
    pp_n = _run_pp(cls, cfg, [(i, i + 1) for i in range(n)], ids)
    assert torch.equal(ref, pp_n), "PP=n logits diverge from PP=1"


def test_greedy_tokens_match(tiny_llama_config):
    cfg = tiny_llama_config
    cls = get_model_class("llama")
    torch.manual_seed(2)
    ids = torch.randint(0, cfg.vocab_size, (1, 5))
    l1 = _run_pp(cls, cfg, [(0, 4)], ids, n_decode=8)
    l2 = _run_pp(cls, cfg, [(0, 1), (1, 4)], ids, n_decode=8)
    assert torch.equal(l1.argmax(-1), l2.argmax(-1))


def test_cache_reset_reproduces(tiny_llama_config):
    cfg = tiny_llama_config
    cls = get_model_class("llama")
    m = init_model(cls, cfg, cfg.shard(0, 4))
    ids = torch.randint(0, cfg.vocab_size, (1, 5))
    c = m.make_cache()
    with torch.no_grad():
        a = m(ids, c)
        for cc in c:
            cc.reset()
        b = m(ids, c)
    assert torch.equal(a, b)


def test_batched_forward(tiny_llama_config):
    cfg = tiny_llama_config
    cls = get_model_class("llama")
    m = init_model(cls, cfg, cfg.shard(0, 4))
    ids = torch.randint(0, cfg.vocab_size, (4, 5))
    c = m.make_cache(batch_size=4)
    with torch.no_grad():
        out = m(ids, c)
        # each row equals its unbatched run
        for b in range(4):
            cb = m.make_cache()
            ref = m(ids[b:b + 1], cb)
            assert torch.equal(out[b:b + 1], ref)


def test_quantized_model_runs(tiny_llama_config):
    cfg = tiny_llama_config
    cfg.raw["quantization"] = {"group_size": 32, "bits": 4}
    cls = get_model_class("llama")
    qc = QuantConfig(32, 4)
    m = init_model(cls, cfg, cfg.shard(0, 4), quant_for=lambda p: qc)
    # re-init quant params coherently: quantize random dense weights
    from mlx_sharding_amd.ops import reference as ref
    from mlx_sharding_amd.models.base import Linear
    torch.manual_seed(3)
    for mod in m.modules():
        if isinstance(mod, Linear) and mod.quant is not None:
            w = torch.randn(mod.out_features, mod.in_features) * 0.05
            wq, sc, bi = ref.quantize(w.bfloat16(), 32, 4)
            mod.weight.data = wq
            mod.scales.data = sc
            mod.biases.data = bi
    ids = torch.randint(0, cfg.vocab_size, (1, 5))
    with torch.no_grad():
        out = m(ids, m.make_cache())
    assert out.shape == (1, 5, cfg.vocab_size)
    assert torch.isfinite(out.float()).all()


def test_chunked_prefill_equivalence(tiny_llama_config):
    """Prefill in chunks (growing cache offset) == single-shot prefill —
    the parity answer to long-context handling."""
    cfg = tiny_llama_config
    cls = get_model_class("llama")
    m = init_model(cls, cfg, cfg.shard(0, 4), seed=13)
    torch.manual_seed(3)
    ids = torch.randint(0, cfg.vocab_size, (1, 12))
    with torch.no_grad():
        c1 = m.make_cache()
        full = m(ids, c1)
        c2 = m.make_cache()
        parts = []
        for chunk in torch.split(ids, 5, dim=1):
            parts.append(m(chunk, c2))
        chunked = torch.cat(parts, dim=1)
    assert torch.allclose(full.float(), chunked.float(), atol=1e-3)
    assert torch.equal(full[:, -1].argmax(-1), chunked[:, -1].argmax(-1))


def test_gemma2_sliding_window_long_context(tiny_gemma2_config):
    """Sliding-window layers must only attend the trailing window."""
    cfg = tiny_gemma2_config  # window = 3
    cls = get_model_class("gemma2")
    m = init_model(cls, cfg, cfg.shard(0, 4), seed=14)
    torch.manual_seed(1)
    a = torch.randint(0, cfg.vocab_size, (1, 10))
    b = a.clone()
    b[0, 0] = (b[0, 0] + 1) % cfg.vocab_size  # differs far outside the window
    with torch.no_grad():
        oa = m(a, m.make_cache())
        ob = m(b, m.make_cache())
    # global (odd) layers still see position 0, so outputs differ overall,
    # but the model runs and is finite; main check: window masking works in
    # the op (unit-tested) and end-to-end forward is stable
    assert torch.isfinite(oa.float()).all() and torch.isfinite(ob.float()).all()


def test_deepseek_group_limited_many_experts():
    """group_limited_greedy gating + >64 experts (DeepSeek-V2 full config
    shape) runs through the torch gating path."""
    from mlx_sharding_amd.config import ModelConfig
    cfg = ModelConfig.from_dict({
        "model_type": "deepseek_v2", "hidden_size": 64, "num_hidden_layers": 2,
        "intermediate_size": 128, "moe_intermediate_size": 32,
        "num_attention_heads": 4, "vocab_size": 128, "rms_norm_eps": 1e-6,
        "rope_theta": 10000.0, "q_lora_rank": 32, "kv_lora_rank": 32,
        "qk_nope_head_dim": 16, "qk_rope_head_dim": 8, "v_head_dim": 16,
        "n_routed_experts": 96, "num_experts_per_tok": 4,
        "n_shared_experts": 1, "first_k_dense_replace": 1, "moe_layer_freq": 1,
        "topk_method": "group_limited_greedy", "n_group": 8, "topk_group": 3,
        "routed_scaling_factor": 16.0, "norm_topk_prob": False,
    })
    cls = get_model_class("deepseek_v2")
    m = init_model(cls, cfg, cfg.shard(0, 2), seed=21)
    ids = torch.randint(0, 128, (2, 5))
    with torch.no_grad():
        out = m(ids, m.make_cache(batch_size=2))
    assert out.shape == (2, 5, 128)
    assert torch.isfinite(out.float()).all()
    # q_lora path exercised too (q_a/q_b projections)
    assert hasattr(m.model.layers["0"].self_attn, "q_a_proj")


def test_mistral_remap():
    """model_type "mistral" resolves to the llama stage model
    (reference MODEL_REMAPPING, shard/utils.py:14-17)."""
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.models.llama import LlamaStageModel
    assert get_model_class("mistral") is LlamaStageModel


def test_yarn_rope_attn_scale_matches_hf(tiny_deepseek_config):
    """The cos/sin multiplier must be the HF/mlx_lm RATIO
    yarn_mscale(f, mscale)/yarn_mscale(f, mscale_all_dim)
    (DeepseekV2YarnRotaryEmbedding._mscale) — for DeepSeek-V2 configs
    (mscale == mscale_all_dim) that is exactly 1.0, not ~1.26."""
    import math

    def hf_yarn_get_mscale(scale, mscale=1.0):
        if scale <= 1:
            return 1.0
        return 0.1 * mscale * math.log(scale) + 1.0

    cfg = tiny_deepseek_config
    cls = get_model_class("deepseek_v2")
    m = init_model(cls, cfg, cfg.shard(0, cfg.num_hidden_layers))
    rs = cfg.get("rope_scaling")
    want = hf_yarn_get_mscale(rs["factor"], rs["mscale"]) / \
        hf_yarn_get_mscale(rs["factor"], rs["mscale_all_dim"])
    assert abs(m.rope_attn_scale - want) < 1e-9
    assert abs(m.rope_attn_scale - 1.0) < 1e-9  # V2 configs: ratio is 1

    # cos table itself matches an HF-style reference computation
    cos, sin = m._rope_tables(torch.device("cpu"), 16)
    pos = torch.arange(cos.shape[0], dtype=torch.float32)
    ang = pos[:, None] * m.rope_inv_freq[None, :]
    assert torch.allclose(cos, torch.cos(ang) * want, atol=1e-6)
    assert torch.allclose(sin, torch.sin(ang) * want, atol=1e-6)

    # mscale_all_dim == 0 ⇒ denominator 1.0 ⇒ numerator survives
    from mlx_sharding_amd.config import ModelConfig
    raw = dict(cfg.raw)
    raw["rope_scaling"] = dict(rs, mscale_all_dim=0.0)
    cfg2 = ModelConfig.from_dict(raw)
    m2 = init_model(cls, cfg2, cfg2.shard(0, cfg2.num_hidden_layers))
    want2 = hf_yarn_get_mscale(rs["factor"], rs["mscale"])
    assert abs(m2.rope_attn_scale - want2) < 1e-9


def test_quantize_on_load(tiny_llama_config, tmp_path):
    """A dense bf16 checkpoint loaded with quantize=(4, 32) runs the
    w4a16 layout and tracks the dense model's outputs."""
    import json

    from safetensors.torch import save_file

    from mlx_sharding_amd.utils.loading import load_model

    cfg = tiny_llama_config
    cls = get_model_class("llama")
    m = init_model(cls, cfg, cfg.shard(0, cfg.num_hidden_layers), seed=3)
    d = tmp_path / "dense"
    d.mkdir()
    save_file({k: v.clone() for k, v in m.state_dict().items()
               if "rope_inv_freq" not in k}, str(d / "model.safetensors"))
    with open(d / "config.json", "w") as f:
        json.dump(cfg.raw, f)

    mq, cfg_q = load_model(d, quantize=(4, 32))
    assert cfg_q.quantization is not None
    from mlx_sharding_amd.models.base import Linear
    n_quant = sum(1 for mod in mq.modules()
                  if isinstance(mod, Linear) and mod.quant is not None)
    assert n_quant > 0, "no linear was quantized"
    md, _ = load_model(d)
    ids = torch.randint(0, 128, (1, 6), generator=torch.Generator().manual_seed(2))
    with torch.no_grad():
        yq = mq(ids, mq.make_cache())[:, -1].float()
        yd = md(ids, md.make_cache())[:, -1].float()
    # int4 tracks dense within quantization error
    cos = torch.nn.functional.cosine_similarity(yq.flatten(), yd.flatten(), dim=0)
    assert cos.item() > 0.98, cos.item()


def test_quantize_on_load_deepseek_stacked(tiny_deepseek_config, tmp_path):
    """quantize-on-load covers STACKED [E, out, in] expert weights
    (checkpoints saved by this framework keep the stacked layout)."""
    import json

    from safetensors.torch import save_file

    from mlx_sharding_amd.utils.loading import load_model

    cfg = tiny_deepseek_config
    cls = get_model_class("deepseek_v2")
    m = init_model(cls, cfg, cfg.shard(0, cfg.num_hidden_layers), seed=8)
    d = tmp_path / "dense"
    d.mkdir()
    save_file({k: v.clone() for k, v in m.state_dict().items()
               if "rope_inv_freq" not in k}, str(d / "model.safetensors"))
    with open(d / "config.json", "w") as f:
        json.dump(cfg.raw, f)

    mq, cfg_q = load_model(d, quantize=(4, 32))
    sw = mq.model.layers["1"].mlp.switch_mlp
    assert sw.quant is not None, "stacked experts were not quantized"
    md, _ = load_model(d)
    ids = torch.randint(0, 128, (1, 5),
                        generator=torch.Generator().manual_seed(1))
    with torch.no_grad():
        yq = mq(ids, mq.make_cache())[:, -1].float()
        yd = md(ids, md.make_cache())[:, -1].float()
    cos = torch.nn.functional.cosine_similarity(yq.flatten(), yd.flatten(), dim=0)
    assert cos.item() > 0.97, cos.item()


def test_qwen2_remap_and_bias_wiring():
    """Beyond-parity qwen2 family: remapped to the llama stage model
    with QKV-only biases (HF Qwen2 hardcodes qkv bias on, o_proj off);
    fused-QKV path must carry the concatenated bias."""
    from mlx_sharding_amd.config import ModelConfig
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.models.fuse import fuse_model
    from mlx_sharding_amd.models.llama import LlamaStageModel

    cfg = ModelConfig.from_dict({
        "model_type": "qwen2", "hidden_size": 64, "num_hidden_layers": 2,
        "intermediate_size": 128, "num_attention_heads": 4,
        "num_key_value_heads": 2, "vocab_size": 96, "rms_norm_eps": 1e-6,
        "rope_theta": 10000.0, "tie_word_embeddings": False})
    cls = get_model_class("qwen2")
    assert cls is LlamaStageModel
    torch.manual_seed(0)
    m = cls(cfg, cfg.shard(0, 2))
    for p in m.parameters():
        p.data = p.data.float().normal_(0, 0.05).to(p.dtype)
    attn = m.model.layers["0"].self_attn
    assert attn.q_proj.bias is not None and attn.k_proj.bias is not None
    assert attn.o_proj.bias is None

    ids = torch.randint(0, 96, (1, 6))
    with torch.no_grad():
        ref = m(ids, m.make_cache(batch_size=1)).float()
    n = fuse_model(m)
    assert n > 0  # the biased QKV group must still fuse
    with torch.no_grad():
        fused = m(ids, m.make_cache(batch_size=1)).float()
    assert torch.allclose(ref, fused, atol=1e-3, rtol=1e-3)


def _tiny_llama():
    from mlx_sharding_amd.config import ModelConfig
    from mlx_sharding_amd.models import get_model_class
    cfg = ModelConfig.from_dict({
        "model_type": "llama", "hidden_size": 64, "num_hidden_layers": 2,
        "intermediate_size": 128, "num_attention_heads": 4,
        "num_key_value_heads": 2, "vocab_size": 96, "rms_norm_eps": 1e-5,
        "rope_theta": 10000.0})
    torch.manual_seed(9)
    m = get_model_class("llama")(cfg, cfg.shard(0, 2))
    for p in m.parameters():
        p.data = p.data.float().normal_(0, 0.05).to(p.dtype)
    return m.eval()


def test_seeded_sampling_reproducible():
    """Same seed => identical sampled stream; different seed diverges
    (temperature > 0), mirroring the reference's seeded sampler."""
    from mlx_sharding_amd.parallel.engine import SamplingParams, generate_step
    m = _tiny_llama()
    ids = torch.randint(0, 96, (1, 6), generator=torch.Generator().manual_seed(1))

    def run(seed, n=8):
        g = generate_step(ids, m, m.make_cache(batch_size=1),
                          params=SamplingParams(temperature=0.9, seed=seed))
        return [next(g)[0] for _ in range(n)]

    assert run(123) == run(123)
    assert run(123) != run(321) or run(123) != run(7)  # overwhelmingly


def test_repetition_penalty_discourages_repeats():
    """With a strong penalty, an already-emitted token's logprob drops
    relative to the unpenalized run (reference utils.py:152-177)."""
    from mlx_sharding_amd.parallel.engine import SamplingParams, generate_step
    m = _tiny_llama()
    ids = torch.randint(0, 96, (1, 5), generator=torch.Generator().manual_seed(2))

    def first_two(params):
        g = generate_step(ids, m, m.make_cache(batch_size=1), params=params)
        t1, lp1 = next(g)
        t2, lp2 = next(g)
        return t1, lp1, t2, lp2

    t1, _, _, lp2_plain = first_two(SamplingParams(temperature=0.0))
    _, _, _, lp2_pen = first_two(SamplingParams(temperature=0.0,
                                                repetition_penalty=5.0))
    # token t1 was emitted at step 1; by step 2 the penalized run must
    # assign it a lower logprob than the unpenalized run does
    assert lp2_pen[t1].item() < lp2_plain[t1].item()


def test_logit_bias_forces_token():
    """A +inf-ish logit bias must force the biased token under greedy
    decoding (reference openai_api param surface)."""
    from mlx_sharding_amd.parallel.engine import SamplingParams, generate_step
    m = _tiny_llama()
    ids = torch.randint(0, 96, (1, 4), generator=torch.Generator().manual_seed(3))
    g = generate_step(ids, m, m.make_cache(batch_size=1),
                      params=SamplingParams(temperature=0.0,
                                            logit_bias={17: 1e9}))
    tid, _ = next(g)
    assert tid == 17


def test_invalid_shard_ranges_rejected():
    """Out-of-order / out-of-bounds layer ranges fail loudly at config
    time (an 8-way CPU smoke once hit 'invalid shard range [4,4)' —
    keep the validation pinned)."""
    from mlx_sharding_amd.config import ModelConfig
    cfg = ModelConfig.from_dict({
        "model_type": "llama", "hidden_size": 32, "num_hidden_layers": 4,
        "intermediate_size": 64, "num_attention_heads": 2,
        "num_key_value_heads": 2, "vocab_size": 32})
    assert cfg.shard(0, 4).n_layers == 4
    for (s, e) in [(2, 2), (3, 1), (-1, 2), (0, 99)]:
        with pytest.raises((ValueError, AssertionError)):
            cfg.shard(s, e)
