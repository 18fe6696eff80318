"""GPU numerics tests: each HIP/CDNA4 kernel vs the plain-PyTorch fp32
reference of the same op (tolerances sized for bf16 I/O, fp32 accum)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from mlx_sharding_amd.ops import reference as ref


def ext():
    from mlx_sharding_amd import ops
    e = ops.hip_ext()
    assert e is not None, "HIP extension must be built on a GPU box"
    return e


def _close(a, b, atol, rtol=1e-2):
    a = a.float().cpu()
    b = b.float().cpu()
    err = (a - b).abs().max().item()
    scale = b.abs().max().item()
    assert err <= atol + rtol * scale, f"max err {err} (scale {scale})"


@pytest.mark.parametrize("H", [256, 2048, 8192])
def test_rms_norm(H):
    x = torch.randn(5, H, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(H, dtype=torch.bfloat16, device="cuda")
    y = ext().rms_norm(x, w, 1e-5, 0.0)
    y_ref = ref.rms_norm(x.cpu(), w.cpu(), 1e-5)
    _close(y, y_ref, atol=2e-2)


def test_rms_norm_gemma_offset():
    x = torch.randn(3, 512, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(512, dtype=torch.bfloat16, device="cuda") * 0.1
    y = ext().rms_norm(x, w, 1e-6, 1.0)
    _close(y, ref.rms_norm(x.cpu(), w.cpu(), 1e-6, 1.0), atol=2e-2)


def test_rms_norm_residual():
    x = torch.randn(4, 1024, dtype=torch.bfloat16, device="cuda")
    r = torch.randn(4, 1024, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(1024, dtype=torch.bfloat16, device="cuda")
    y, h = ext().rms_norm_residual(x, r, w, 1e-5, 0.0)
    h_ref = (x.cpu().float() + r.cpu().float()).bfloat16()
    _close(h, h_ref, atol=1e-2)
    _close(y, ref.rms_norm(h_ref, w.cpu(), 1e-5), atol=2e-2)


def test_swiglu_geglu():
    g = torch.randn(3, 4096, dtype=torch.bfloat16, device="cuda")
    u = torch.randn(3, 4096, dtype=torch.bfloat16, device="cuda")
    _close(ext().glu(g, u, False), ref.swiglu(g.cpu(), u.cpu()), atol=2e-2)
    _close(ext().glu(g, u, True), ref.geglu(g.cpu(), u.cpu()), atol=2e-2)


def test_softcap():
    x = torch.randn(4, 1024, dtype=torch.bfloat16, device="cuda") * 60
    _close(ext().softcap(x, 30.0), ref.softcap(x.cpu(), 30.0), atol=0.3)


@pytest.mark.parametrize("interleaved", [False, True])
def test_rope(interleaved):
    B, T, nH, D = 2, 7, 4, 128
    x = torch.randn(B, T, nH, D, dtype=torch.bfloat16, device="cuda")
    inv = ref.rope_freqs(D)
    cos, sin = ref.rope_cos_sin(torch.arange(3, 3 + T), inv)
    y = ext().apply_rope(x, cos.cuda(), sin.cuda(), interleaved)
    y_ref = ref.apply_rope(x.cpu(), cos, sin, interleaved)
    _close(y, y_ref, atol=2e-2)


@pytest.mark.parametrize("Hq,Hkv,Dk,Dv,S", [
    (4, 4, 64, 64, 33),       # MHA
    (8, 2, 128, 128, 300),    # GQA 4
    (16, 16, 192, 128, 513),  # MLA shape
    (8, 1, 128, 128, 1024),   # GQA 8
])
def test_attn_decode(Hq, Hkv, Dk, Dv, S):
    torch.manual_seed(0)
    B = 3
    Scap = ((S + 1023) // 1024) * 1024
    q = torch.randn(B, Hq, 1, Dk, dtype=torch.bfloat16, device="cuda")
    kbuf = torch.randn(B, Hkv, Scap, Dk, dtype=torch.bfloat16, device="cuda")
    vbuf = torch.randn(B, Hkv, Scap, Dv, dtype=torch.bfloat16, device="cuda")
    k = kbuf[:, :, :S]
    v = vbuf[:, :, :S]
    out = ext().attn_decode(q, k, v, Dk ** -0.5, 0.0, 0)
    out_ref = ref.attention(q.cpu(), k.cpu(), v.cpu(), Dk ** -0.5,
                            causal_offset=S - 1)
    _close(out, out_ref, atol=3e-2)


@pytest.mark.parametrize("S", [96, 512, 1500])
def test_attn_decode_absorbed_mla_shape(S):
    """Absorbed-MLA decode: G=16 MQA over 576-dim compressed rows, with
    V = the first 512 columns of the K rows (DVT=2 + in-row V stride)."""
    torch.manual_seed(3)
    B, Hq, Dk, Dv = 4, 16, 576, 512
    Scap = ((S + 1023) // 1024) * 1024
    q = torch.randn(B, Hq, 1, Dk, dtype=torch.bfloat16, device="cuda")
    kbuf = torch.randn(B, 1, Scap, Dk, dtype=torch.bfloat16, device="cuda")
    k = kbuf[:, :, :S]
    v = k[..., :Dv]
    out = ext().attn_decode(q, k, v, Dk ** -0.5, 0.0, 0)
    out_ref = ref.attention(q.cpu(), k.cpu(), v.cpu().contiguous(),
                            Dk ** -0.5, causal_offset=S - 1)
    _close(out, out_ref, atol=3e-2)


def test_attn_decode_softcap_window():
    torch.manual_seed(1)
    B, Hq, Hkv, D, S = 2, 8, 4, 128, 700
    q = torch.randn(B, Hq, 1, D, dtype=torch.bfloat16, device="cuda")
    kbuf = torch.randn(B, Hkv, 1024, D, dtype=torch.bfloat16, device="cuda")
    vbuf = torch.randn(B, Hkv, 1024, D, dtype=torch.bfloat16, device="cuda")
    k, v = kbuf[:, :, :S], vbuf[:, :, :S]
    out = ext().attn_decode(q, k, v, 0.1, 50.0, 256)
    out_ref = ref.attention(q.cpu(), k.cpu(), v.cpu(), 0.1,
                            causal_offset=S - 1, softcap=50.0,
                            sliding_window=256)
    _close(out, out_ref, atol=3e-2)


@pytest.mark.parametrize("M,O,H", [(3, 128, 256), (33, 512, 320), (64, 2048, 2048),
                                   (64, 80, 2048)])
def test_dense_gemv_mfma(M, O, H):
    """bf16 MFMA decode GEMV vs fp32 F.linear (covers partial token
    groups, split-K small-O shapes and the k-slice tail)."""
    torch.manual_seed(2)
    x = torch.randn(M, H, dtype=torch.bfloat16, device="cuda") * 0.3
    w = torch.randn(O, H, dtype=torch.bfloat16, device="cuda") * 0.3
    y = ext().dense_gemv(x, w)
    y_ref = torch.nn.functional.linear(x.float().cpu(), w.float().cpu())
    _close(y, y_ref, atol=5e-2)


@pytest.mark.parametrize("bits,gs", [(4, 64), (4, 32), (8, 64)])
def test_w4a16_gemv(bits, gs):
    torch.manual_seed(0)
    M, O, H = 4, 384, 512
    w = torch.randn(O, H, dtype=torch.bfloat16) * 0.1
    wq, sc, bi = ref.quantize(w, gs, bits)
    x = torch.randn(M, H, dtype=torch.bfloat16)
    y_ref = ref.quantized_linear(x, wq, sc, bi, gs, bits)
    y = ext().w4a16_gemv(x.cuda(), wq.cuda(), sc.cuda(), bi.cuda(), gs, bits)
    _close(y, y_ref, atol=5e-2)


@pytest.mark.parametrize("bits,gs", [(4, 64), (8, 64)])
def test_dequant(bits, gs):
    torch.manual_seed(0)
    O, H = 128, 512
    w = torch.randn(O, H, dtype=torch.bfloat16) * 0.1
    wq, sc, bi = ref.quantize(w, gs, bits)
    wd_ref = ref.dequantize(wq, sc, bi, gs, bits)
    wd = ext().dequant(wq.cuda(), sc.cuda(), bi.cuda(), H, gs, bits)
    _close(wd, wd_ref, atol=1e-3)


def test_quantized_linear_large_m_path():
    from mlx_sharding_amd import ops as O
    torch.manual_seed(2)
    M, Od, H = 200, 256, 512  # M > GEMV threshold → dequant+GEMM path
    w = torch.randn(Od, H, dtype=torch.bfloat16) * 0.1
    wq, sc, bi = ref.quantize(w, 64, 4)
    x = torch.randn(M, H, dtype=torch.bfloat16)
    y = O.quantized_linear(x.cuda(), wq.cuda(), sc.cuda(), bi.cuda(), 64, 4)
    y_ref = ref.quantized_linear(x, wq, sc, bi, 64, 4)
    _close(y, y_ref, atol=8e-2)


def test_moe_grouped_mlp():
    from mlx_sharding_amd import ops as O
    torch.manual_seed(0)
    E, H, I, N, K = 8, 256, 512, 6, 2
    x = torch.randn(N, H, dtype=torch.bfloat16, device="cuda")
    gw = (torch.randn(E, I, H, dtype=torch.bfloat16, device="cuda") * 0.05)
    uw = (torch.randn(E, I, H, dtype=torch.bfloat16, device="cuda") * 0.05)
    dw = (torch.randn(E, H, I, dtype=torch.bfloat16, device="cuda") * 0.05)
    wts = torch.rand(N, K, dtype=torch.bfloat16, device="cuda")
    idx = torch.randint(0, E, (N, K), device="cuda")
    out = O.grouped_expert_mlp(x, gw, uw, dw, wts, idx)
    out_ref = ref.grouped_expert_mlp(x.cpu(), gw.cpu(), uw.cpu(), dw.cpu(),
                                     wts.cpu(), idx.cpu())
    _close(out, out_ref, atol=5e-2)


@pytest.mark.parametrize("N,K", [(6, 2), (40, 6), (64, 6)])
def test_moe_grouped_mlp_mfma16(N, K):
    """MFMA 16-token-sub-range kernels vs the fp32 reference (incl.
    ragged I % MF_CH chunks and partially-filled sub-ranges)."""
    from mlx_sharding_amd import ops as O
    torch.manual_seed(1)
    E, H, I = 8, 256, 320
    x = torch.randn(N, H, dtype=torch.bfloat16, device="cuda")
    gw = (torch.randn(E, I, H, dtype=torch.bfloat16, device="cuda") * 0.05)
    uw = (torch.randn(E, I, H, dtype=torch.bfloat16, device="cuda") * 0.05)
    dw = (torch.randn(E, H, I, dtype=torch.bfloat16, device="cuda") * 0.05)
    wts = torch.rand(N, K, dtype=torch.bfloat16, device="cuda")
    idx = torch.stack([torch.randperm(E, device="cuda")[:K]
                       for _ in range(N)])
    subs = O.make_expert_subranges(idx, wts, E, max_tok=16)[:5]
    out = O.grouped_expert_mlp_subs(x, gw, uw, dw, subs, max_tok=16)
    out_ref = ref.grouped_expert_mlp(x.cpu(), gw.cpu(), uw.cpu(), dw.cpu(),
                                     wts.cpu(), idx.cpu())
    _close(out, out_ref, atol=5e-2)


def test_moe_grouped_mlp_quant():
    from mlx_sharding_amd import ops as O
    torch.manual_seed(0)
    E, H, I, N, K = 4, 256, 128, 5, 2
    trip = {}
    for name, (o, i) in {"g": (I, H), "u": (I, H), "d": (H, I)}.items():
        ws, ss, bs = [], [], []
        for e in range(E):
            w = torch.randn(o, i, dtype=torch.bfloat16) * 0.05
            wq, sc, bi = ref.quantize(w, 64, 4)
            ws.append(wq); ss.append(sc); bs.append(bi)
        trip[name] = (torch.stack(ws).cuda(), torch.stack(ss).cuda(),
                      torch.stack(bs).cuda())
    x = torch.randn(N, H, dtype=torch.bfloat16, device="cuda")
    wts = torch.rand(N, K, dtype=torch.bfloat16, device="cuda")
    idx = torch.randint(0, E, (N, K), device="cuda")
    out = O.grouped_expert_mlp_quant(x, trip["g"], trip["u"], trip["d"],
                                     wts, idx, 64, 4)
    cpu = lambda t: tuple(z.cpu() for z in t)
    out_ref = O.grouped_expert_mlp_quant(x.cpu(), cpu(trip["g"]),
                                         cpu(trip["u"]), cpu(trip["d"]),
                                         wts.cpu(), idx.cpu(), 64, 4)
    _close(out, out_ref, atol=6e-2)


def test_prefill_attention_gpu_path():
    from mlx_sharding_amd import ops as O
    torch.manual_seed(0)
    B, Hq, Hkv, T, D = 2, 8, 2, 64, 128
    q = torch.randn(B, Hq, T, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(B, Hkv, T, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(B, Hkv, T, D, dtype=torch.bfloat16, device="cuda")
    out = O.attention(q, k, v, D ** -0.5)
    out_ref = ref.attention(q.cpu(), k.cpu(), v.cpu(), D ** -0.5)
    _close(out, out_ref, atol=4e-2)


def test_model_decode_gpu_matches_cpu(tiny_llama_config):
    """Whole-stage decode on GPU (HIP kernels) vs CPU reference."""
    from conftest import init_model
    from mlx_sharding_amd.models import get_model_class
    cfg = tiny_llama_config
    cls = get_model_class("llama")
    m = init_model(cls, cfg, cfg.shard(0, 4), seed=3)
    ids = torch.randint(0, cfg.vocab_size, (2, 12),
                        generator=torch.Generator().manual_seed(0))
    with torch.no_grad():
        c_cpu = m.make_cache(batch_size=2)
        out_cpu = m(ids, c_cpu)
        tok_cpu = out_cpu[:, -1].float().argmax(-1)
        d_cpu = m(tok_cpu[:, None], c_cpu)

        mg = m.to("cuda")
        c_gpu = mg.make_cache(batch_size=2)
        out_gpu = mg(ids.cuda(), c_gpu)
        tok_gpu = out_gpu[:, -1].float().argmax(-1)
        d_gpu = mg(tok_gpu[:, None], c_gpu)
    assert torch.equal(tok_cpu, tok_gpu.cpu()), "greedy tokens diverge CPU vs GPU"
    _close(d_gpu, d_cpu, atol=6e-2)


def test_moe_prefill_gemm_path():
    from mlx_sharding_amd import ops as O
    torch.manual_seed(1)
    E, H, I, N, K = 8, 256, 512, 200, 2  # N*K >= 256 → per-expert GEMM path
    x = torch.randn(N, H, dtype=torch.bfloat16, device="cuda")
    gw = torch.randn(E, I, H, dtype=torch.bfloat16, device="cuda") * 0.05
    uw = torch.randn(E, I, H, dtype=torch.bfloat16, device="cuda") * 0.05
    dw = torch.randn(E, H, I, dtype=torch.bfloat16, device="cuda") * 0.05
    wts = torch.rand(N, K, dtype=torch.bfloat16, device="cuda")
    idx = torch.randint(0, E, (N, K), device="cuda")
    out = O.grouped_expert_mlp(x, gw, uw, dw, wts, idx)
    out_ref = ref.grouped_expert_mlp(x.cpu(), gw.cpu(), uw.cpu(), dw.cpu(),
                                     wts.cpu(), idx.cpu())
    _close(out, out_ref, atol=6e-2)


def test_moe_uneven_expert_load():
    """All tokens on one expert (cnt ≫ MG_TOK) + empty experts."""
    from mlx_sharding_amd import ops as O
    torch.manual_seed(2)
    E, H, I, N, K = 8, 256, 512, 10, 2
    x = torch.randn(N, H, dtype=torch.bfloat16, device="cuda")
    gw = torch.randn(E, I, H, dtype=torch.bfloat16, device="cuda") * 0.05
    uw = torch.randn(E, I, H, dtype=torch.bfloat16, device="cuda") * 0.05
    dw = torch.randn(E, H, I, dtype=torch.bfloat16, device="cuda") * 0.05
    wts = torch.rand(N, K, dtype=torch.bfloat16, device="cuda")
    idx = torch.full((N, K), 3, device="cuda", dtype=torch.long)
    out = O.grouped_expert_mlp(x, gw, uw, dw, wts, idx)
    out_ref = ref.grouped_expert_mlp(x.cpu(), gw.cpu(), uw.cpu(), dw.cpu(),
                                     wts.cpu(), idx.cpu())
    _close(out, out_ref, atol=6e-2)


def test_moe_gate_subranges_kernel():
    """Fused gating kernel vs torch moe_gate + make_expert_subranges."""
    from mlx_sharding_amd import ops as O
    torch.manual_seed(3)
    N, E, K = 32, 64, 6
    logits = torch.randn(N, E, dtype=torch.bfloat16, device="cuda")
    sub_e, sub_off, sub_cnt, sorted_tok, sorted_wt = O.moe_gate_subranges(
        logits, K, routed_scaling_factor=1.5, norm_topk_prob=False)
    w_ref, idx_ref = ref.moe_gate(logits.float().cpu(), K,
                                  routed_scaling_factor=1.5)
    # reconstruct (expert, token) -> weight map from the kernel outputs
    got = {}
    for s in range(sub_e.shape[0]):
        for t in range(int(sub_cnt[s])):
            p = int(sub_off[s]) + t
            got[(int(sub_e[s]), int(sorted_tok[p]))] = float(sorted_wt[p])
    expect = {}
    for n in range(N):
        for k in range(K):
            expect[(int(idx_ref[n, k]), n)] = float(w_ref[n, k])
    assert set(got.keys()) == set(expect.keys())
    for key in expect:
        assert abs(got[key] - expect[key]) < 3e-2, (key, got[key], expect[key])


def test_moe_fused_path_matches_torch_path():
    """Full DeepseekV2MoE forward: fused gating path vs CPU reference."""
    from conftest import init_model
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.config import ModelConfig
    cfg = ModelConfig.from_dict({
        "model_type": "deepseek_v2", "hidden_size": 256, "num_hidden_layers": 2,
        "intermediate_size": 512, "moe_intermediate_size": 256,
        "num_attention_heads": 4, "vocab_size": 512, "rms_norm_eps": 1e-6,
        "rope_theta": 10000.0, "q_lora_rank": None, "kv_lora_rank": 64,
        "qk_nope_head_dim": 32, "qk_rope_head_dim": 16, "v_head_dim": 32,
        "n_routed_experts": 16, "num_experts_per_tok": 4, "n_shared_experts": 1,
        "first_k_dense_replace": 1, "moe_layer_freq": 1,
    })
    cls = get_model_class("deepseek_v2")
    m = init_model(cls, cfg, cfg.shard(0, 2), seed=5)
    ids = torch.randint(0, 512, (2, 4), generator=torch.Generator().manual_seed(0))
    with torch.no_grad():
        out_cpu = m(ids, m.make_cache(batch_size=2))
        mg = m.to("cuda")
        out_gpu = mg(ids.cuda(), mg.make_cache(batch_size=2))
    a = out_cpu.float()
    b = out_gpu.float().cpu()
    err = (a - b).abs().max().item()
    assert err < 0.1 + 1e-2 * a.abs().max().item(), err


def test_absorbed_mla_chunked_prefill_gpu():
    """Chunked prefill over the COMPRESSED MLA cache (later chunks
    re-expand the prefix via kv_b_proj) == single-shot GPU prefill."""
    from conftest import init_model
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.utils.presets import get_preset
    cfg = get_preset("debug-deepseek")
    cls = get_model_class("deepseek_v2")
    m = init_model(cls, cfg, cfg.shard(0, cfg.num_hidden_layers),
                   seed=7).to("cuda")
    torch.manual_seed(5)
    ids = torch.randint(0, cfg.vocab_size, (2, 12)).cuda()
    with torch.no_grad():
        c1 = m.make_cache(batch_size=2)
        assert c1[0].n_kv_heads == 1, "absorbed compressed cache expected"
        full = m(ids, c1)
        c2 = m.make_cache(batch_size=2)
        parts = [m(chunk, c2) for chunk in torch.split(ids, 5, dim=1)]
        chunked = torch.cat(parts, dim=1)
    assert torch.allclose(full.float(), chunked.float(), atol=2e-2), \
        (full.float() - chunked.float()).abs().max()
    assert torch.equal(full[:, -1].argmax(-1), chunked[:, -1].argmax(-1))


def test_graph_captured_decode_matches_eager(tiny_llama_config):
    """hipGraph-captured decode must emit the same greedy tokens as the
    eager decode path from the same prefill state."""
    from conftest import init_model
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.parallel.rccl import PipelineWorker
    cfg = tiny_llama_config
    cls = get_model_class("llama")
    ids = [torch.randint(0, cfg.vocab_size, (2, 6),
                         generator=torch.Generator().manual_seed(4)).cuda()]

    def run(graph: bool):
        m = init_model(cls, cfg, cfg.shard(0, 4), seed=9).to("cuda")
        w = PipelineWorker(m, 0, 1, torch.device("cuda"), torch.bfloat16)
        toks = w.prefill(ids, 2, 1, 6)
        seq = [toks[0].tolist()]
        if graph:
            w.enable_graph_decode(toks, 2, 1, 64)
        for _ in range(6):
            toks = w.decode_step(toks, 2, 1)
            seq.append(toks[0].tolist())
        return seq

    eager = run(False)
    captured = run(True)
    assert eager == captured, f"{eager} vs {captured}"


def test_graph_captured_decode_deepseek():
    from conftest import init_model
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.parallel.rccl import PipelineWorker
    from mlx_sharding_amd.utils.presets import get_preset
    cfg = get_preset("debug-deepseek")
    cls = get_model_class("deepseek_v2")
    ids = [torch.randint(0, cfg.vocab_size, (4, 5),
                         generator=torch.Generator().manual_seed(1)).cuda()]

    def run(graph: bool):
        m = init_model(cls, cfg, cfg.shard(0, cfg.num_hidden_layers),
                       seed=2).to("cuda")
        w = PipelineWorker(m, 0, 1, torch.device("cuda"), torch.bfloat16)
        toks = w.prefill(ids, 4, 1, 5)
        seq = [toks[0].tolist()]
        if graph:
            w.enable_graph_decode(toks, 4, 1, 64)
        for _ in range(5):
            toks = w.decode_step(toks, 4, 1)
            seq.append(toks[0].tolist())
        return seq

    assert run(False) == run(True)


def test_mfma_fragment_layout_probe():
    """A=I with asymmetric B (guide G9: transpose-detecting)."""
    A = torch.zeros(16, 32, dtype=torch.bfloat16, device="cuda")
    for i in range(16):
        A[i, i] = 1.0
    Bm = (torch.arange(32 * 16, dtype=torch.float32, device="cuda")
          .reshape(32, 16) * 0.01).bfloat16()
    D = ext().mfma_probe(A, Bm)
    expect = (A.float() @ Bm.float())
    _close(D, expect, atol=1e-2)
    A2 = torch.randn(16, 32, dtype=torch.bfloat16, device="cuda")
    B2 = torch.randn(32, 16, dtype=torch.bfloat16, device="cuda")
    D2 = ext().mfma_probe(A2, B2)
    _close(D2, A2.float() @ B2.float(), atol=5e-2)


@pytest.mark.parametrize("Hq,Hkv,Dk,Dv,T,S,offset,cap,win", [
    (4, 4, 64, 64, 64, 64, 0, 0.0, 0),        # square MHA
    (8, 2, 128, 128, 100, 100, 0, 0.0, 0),    # GQA, ragged T
    (16, 16, 192, 128, 64, 192, 128, 0.0, 0), # MLA shape, chunked offset
    (4, 2, 128, 128, 33, 97, 64, 50.0, 48),   # softcap + window + offsets
    (4, 2, 256, 256, 64, 96, 32, 50.0, 0),    # gemma2 256-dim heads
])
def test_attn_prefill_mfma(Hq, Hkv, Dk, Dv, T, S, offset, cap, win):
    torch.manual_seed(0)
    B = 2
    q = torch.randn(B, Hq, T, Dk, dtype=torch.bfloat16, device="cuda") * 0.5
    Scap = ((S + 255) // 256) * 256
    kbuf = torch.randn(B, Hkv, Scap, Dk, dtype=torch.bfloat16, device="cuda") * 0.5
    vbuf = torch.randn(B, Hkv, Scap, Dv, dtype=torch.bfloat16, device="cuda") * 0.5
    k, v = kbuf[:, :, :S], vbuf[:, :, :S]
    out = ext().attn_prefill(q, k, v, Dk ** -0.5, cap, win, offset)
    out_ref = ref.attention(q.cpu(), k.cpu(), v.cpu(), Dk ** -0.5,
                            causal_offset=offset, softcap=cap,
                            sliding_window=win)
    _close(out, out_ref, atol=4e-2)


@pytest.mark.parametrize("M,O,H", [(32, 384, 512), (17, 1408, 2048), (8, 256, 1024)])
def test_w4a16_mfma_path(M, O, H):
    """M>=8 dispatches the MFMA w4 GEMM (dequant-once, tokens as columns)."""
    torch.manual_seed(1)
    w = torch.randn(O, H, dtype=torch.bfloat16) * 0.1
    wq, sc, bi = ref.quantize(w, 64, 4)
    x = torch.randn(M, H, dtype=torch.bfloat16)
    y_ref = ref.quantized_linear(x, wq, sc, bi, 64, 4)
    y = ext().w4a16_gemv(x.cuda(), wq.cuda(), sc.cuda(), bi.cuda(), 64, 4)
    _close(y, y_ref, atol=6e-2)


def test_gemma2_model_gpu(tiny_gemma2_config):
    """gemma2 stage on GPU (softcap + sliding-window kernels) vs CPU."""
    from conftest import init_model
    from mlx_sharding_amd.models import get_model_class
    cfg = tiny_gemma2_config
    cls = get_model_class("gemma2")
    m = init_model(cls, cfg, cfg.shard(0, 4), seed=6)
    ids = torch.randint(0, cfg.vocab_size, (2, 9),
                        generator=torch.Generator().manual_seed(2))
    with torch.no_grad():
        out_cpu = m(ids, m.make_cache(batch_size=2))
        tok_cpu = out_cpu[:, -1].float().argmax(-1)
        mg = m.to("cuda")
        c = mg.make_cache(batch_size=2)
        out_gpu = mg(ids.cuda(), c)
        tok_gpu = out_gpu[:, -1].float().argmax(-1)
        d_gpu = mg(tok_gpu[:, None], c)
    assert torch.equal(tok_cpu, tok_gpu.cpu())
    assert torch.isfinite(d_gpu.float()).all()


def test_graph_captured_decode_gemma2(tiny_gemma2_config):
    from conftest import init_model
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.parallel.rccl import PipelineWorker
    cfg = tiny_gemma2_config
    cls = get_model_class("gemma2")
    ids = [torch.randint(0, cfg.vocab_size, (2, 5),
                         generator=torch.Generator().manual_seed(3)).cuda()]

    def run(graph):
        m = init_model(cls, cfg, cfg.shard(0, 4), seed=8).to("cuda")
        w = PipelineWorker(m, 0, 1, torch.device("cuda"), torch.bfloat16)
        toks = w.prefill(ids, 2, 1, 5)
        seq = [toks[0].tolist()]
        if graph:
            assert w.enable_graph_decode(toks, 2, 1, 64)
        for _ in range(5):
            toks = w.decode_step(toks, 2, 1)
            seq.append(toks[0].tolist())
        return seq

    assert run(False) == run(True)


@pytest.mark.parametrize("bits,gs,M,O,H", [(4, 64, 64, 192, 256),
                                           (4, 32, 17, 96, 128),
                                           (8, 64, 33, 64, 256),
                                           (4, 128, 64, 64, 384)])
def test_w4f16_gemv(bits, gs, M, O, H):
    """Dense fp16-dequant GEMV (repacked words + pk_fma) vs fp32
    dequant reference."""
    from mlx_sharding_amd import ops as O_
    torch.manual_seed(0)
    w = torch.randn(O, H, dtype=torch.bfloat16) * 0.05
    wq, sc, bi = ref.quantize(w, gs, bits)
    x = torch.randn(M, H, dtype=torch.bfloat16, device="cuda")
    ext = O_.hip_ext()
    y = ext.w4f16_gemv(x.to(torch.float16),
                       O_.repack_w4(wq.cuda(), bits),
                       sc.cuda(), bi.cuda(), gs, bits)
    wd = ref.dequantize(wq, sc, bi, gs, bits).float()
    want = (x.cpu().float() @ wd.T)
    _close(y, want.to(torch.bfloat16), atol=3e-2 * max(1.0, want.abs().max().item()))


def test_strided_view_ops_match_contiguous():
    """Fused-projection split views feed rope/glu/rms_norm in place —
    results must equal running on contiguous copies (the bindings used
    to materialize a .contiguous() copy per call; ~5/layer on llama)."""
    from mlx_sharding_amd import ops as O_
    ext = O_.hip_ext()
    torch.manual_seed(0)
    B, T, W = 2, 3, 256
    fused = torch.randn(B, T, W, dtype=torch.bfloat16, device="cuda")
    # rope on a strided [B,T,nH,D] slice (like q from a fused qkv)
    q = fused[..., :128].view(B, T, 4, 32)
    cos = torch.rand(T, 16, device="cuda")
    sin = torch.rand(T, 16, device="cuda")
    got = ext.apply_rope(q, cos, sin, False)
    want = ext.apply_rope(q.contiguous(), cos, sin, False)
    assert torch.equal(got, want)
    # inner slice with head stride != D (nope|rope split)
    qh = fused[..., :192].view(B, T, 4, 48)
    qpe = qh[..., 32:]  # D=16 -> cos/sin [T, 8]
    got = ext.apply_rope(qpe, cos[:, :8], sin[:, :8], True)
    want = ext.apply_rope(qpe.contiguous(), cos[:, :8], sin[:, :8], True)
    assert torch.equal(got, want)
    # glu on gate|up split views
    g, u = fused[..., :128], fused[..., 128:]
    got = ext.glu(g, u, False)
    want = ext.glu(g.contiguous(), u.contiguous(), False)
    assert torch.equal(got, want)
    # rms_norm on a column-slice view
    w = torch.randn(128, dtype=torch.bfloat16, device="cuda")
    got = ext.rms_norm(g, w, 1e-5, 0.0)
    want = ext.rms_norm(g.contiguous(), w, 1e-5, 0.0)
    assert torch.equal(got, want)
    # rope_append_kv from k|v split views
    kv = torch.randn(B, T, 2 * 2 * 32, dtype=torch.bfloat16, device="cuda")
    k = kv[..., :64].view(B, T, 2, 32)
    v = kv[..., 64:].view(B, T, 2, 32)
    kc1 = torch.zeros(B, 2, 8, 32, dtype=torch.bfloat16, device="cuda")
    vc1 = torch.zeros_like(kc1)
    ext.rope_append_kv(k, v, cos, sin, kc1, vc1, pos0=2)
    kc2 = torch.zeros_like(kc1)
    vc2 = torch.zeros_like(kc1)
    ext.rope_append_kv(k.contiguous(), v.contiguous(), cos, sin, kc2, vc2,
                       pos0=2)
    assert torch.equal(kc1, kc2) and torch.equal(vc1, vc2)


def test_w4f16_moe_kernels_large_H():
    """Exercise the w4f16 MoE kernels' BATCHED main loop (needs
    H/32 >= 2 full 8-slice batches — the tiny-H tests only hit the
    tail loop, which let an invalid batch remap slip through once)."""
    from mlx_sharding_amd import ops as O_
    ext = O_.hip_ext()
    torch.manual_seed(2)
    E, H, I, N, K, gs = 4, 640, 512, 24, 2, 64
    trip = {}
    for name, (o, i) in {"g": (I, H), "u": (I, H), "d": (H, I)}.items():
        ws, ss, bs = [], [], []
        for e in range(E):
            w = torch.randn(o, i, dtype=torch.bfloat16) * 0.05
            wq, sc, bi = ref.quantize(w, gs, 4)
            ws.append(wq); ss.append(sc); bs.append(bi)
        trip[name] = (torch.stack(ws).cuda(), torch.stack(ss).cuda(),
                      torch.stack(bs).cuda())
    x = torch.randn(N, H, dtype=torch.bfloat16, device="cuda")
    wts = torch.rand(N, K, dtype=torch.bfloat16, device="cuda")
    idx = torch.randint(0, E, (N, K), device="cuda")
    out = O_.grouped_expert_mlp_quant(x, trip["g"], trip["u"], trip["d"],
                                      wts, idx, gs, 4)
    cpu = lambda t: tuple(z.cpu() for z in t)
    want = O_.grouped_expert_mlp_quant(x.cpu(), cpu(trip["g"]), cpu(trip["u"]),
                                       cpu(trip["d"]), wts.cpu(), idx.cpu(),
                                       gs, 4)
    _close(out, want, atol=6e-2)


@pytest.mark.parametrize("M,O,H", [(64, 256, 512), (33, 192, 264),
                                   (17, 64, 2048), (64, 128, 288)])
def test_dense_gemm64(M, O, H):
    """LDS-tiled M<=64 dense GEMM vs torch matmul (deep-k decode path)."""
    from mlx_sharding_amd import ops as O_
    torch.manual_seed(1)
    x = torch.randn(M, H, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(O, H, dtype=torch.bfloat16, device="cuda") * 0.05
    y = O_.hip_ext().dense_gemm64(x, w)
    want = (x.float() @ w.float().T)
    _close(y, want.to(torch.bfloat16),
           atol=4e-2 * max(1.0, want.abs().max().item()))


@pytest.mark.gpu
def test_gemm_m64_kseg_matches_torch():
    """K-segmented coalesced-A dense GEMM vs fp32 torch on its target
    regime and edge cases (tail N, M<64, ksegs sweep)."""
    from mlx_sharding_amd import ops
    ext = ops.hip_ext()
    torch.manual_seed(0)
    for (M, N, K, ks) in [(64, 8192, 28672, 4), (64, 8256, 4096, 2),
                          (17, 8192, 16384, 8), (1, 8192, 16384, 1)]:
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) * 0.1
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.02
        ref = x.float() @ w.t().float()
        got = ext.gemm_m64_kseg(x, w, ks).float()
        rel = (got - ref).abs().max().item() / ref.abs().max().item()
        assert rel < 2e-2, f"{(M, N, K, ks)}: rel err {rel}"


@pytest.mark.gpu
def test_linear_dispatch_uses_kseg_on_target_shape():
    """ops.linear must route the 70B down-proj shape through the custom
    kernel (and match torch)."""
    from mlx_sharding_amd import ops
    x = torch.randn(64, 28672, device="cuda", dtype=torch.bfloat16) * 0.1
    w = torch.randn(8192, 28672, device="cuda", dtype=torch.bfloat16) * 0.02
    y = ops.linear(x, w)
    ref = torch.nn.functional.linear(x, w)
    cos = torch.nn.functional.cosine_similarity(
        y.float().flatten(), ref.float().flatten(), dim=0).item()
    assert cos > 0.999


@pytest.mark.gpu
def test_gemm_kseg_under_graph_capture():
    """The kseg launcher (memset + kernel + convert) must capture and
    replay correctly in a hipGraph — the B=1 serving path would capture
    it for 70B-class models."""
    from mlx_sharding_amd import ops
    x = torch.randn(8, 16384, device="cuda", dtype=torch.bfloat16) * 0.1
    w = torch.randn(8192, 16384, device="cuda", dtype=torch.bfloat16) * 0.02
    ref = ops.linear(x, w).float()

    g = torch.cuda.CUDAGraph()
    out = None
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(2):  # warmup allocs outside capture
            out = ops.linear(x, w)
    torch.cuda.current_stream().wait_stream(s)
    with torch.cuda.graph(g):
        out = ops.linear(x, w)
    x.copy_(x * 1.0)  # same values; replay must recompute into `out`
    g.replay()
    torch.cuda.synchronize()
    cos = torch.nn.functional.cosine_similarity(
        out.float().flatten(), ref.flatten(), dim=0).item()
    assert cos > 0.999
    # replay twice more (accumulation bugs would double results)
    g.replay()
    g.replay()
    torch.cuda.synchronize()
    rel = (out.float() - ref).abs().max().item() / ref.abs().max().item()
    assert rel < 2e-2


@pytest.mark.gpu
def test_gemm_m64_kseg_w4_matches_reference():
    """Packed-weight kseg GEMM vs dequant+torch fp32 (infrastructure
    kernel — faster than hipBLASLt-bf16 on K-long shapes at 4x less
    DRAM; see docs/PERFORMANCE.md)."""
    from mlx_sharding_amd import ops
    from mlx_sharding_amd.ops import reference as ref
    ext = ops.hip_ext()
    torch.manual_seed(1)
    for (M, N, K, gs, ks) in [(64, 8192, 16384, 64, 4),
                              (13, 4096, 4096, 32, 2),
                              (64, 4160, 8192, 128, 1)]:
        x = torch.randn(M, K, dtype=torch.bfloat16) * 0.1
        w = torch.randn(N, K, dtype=torch.bfloat16) * 0.02
        wq, sc, bi = ref.quantize(w, gs, 4)
        dq = ref.dequantize(wq, sc, bi, gs, 4).float()
        refo = (x.float() @ dq.t()).cuda()
        rp = ops.repack_w4(wq.cuda(), 4)
        got = ext.gemm_m64_kseg_w4(x.to(torch.float16).cuda(), rp,
                                   sc.cuda(), bi.cuda(), gs, ks).float()
        rel = (got - refo).abs().max().item() / refo.abs().max().item()
        assert rel < 3e-2, f"{(M, N, K, gs, ks)}: rel {rel}"
