"""GPU pipeline readiness tests: RCCL (nccl backend) init at world=1,
two-stage model partition chained on a single device, and the graphed
single-stream serving path with cross-generation graph reuse.

These pre-validate the multi-GPU path's building blocks on the 1-GPU
box; the 8-GPU scaling run itself belongs to the
round-end driver.
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_nccl_world1_init():
    """RCCL communicator init + collectives on hardware at world=1 —
    the same init path bench.py takes under torchrun."""
    import torch.distributed as dist
    if dist.is_initialized():  # another test initialized a group
        dist.destroy_process_group()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29539")
    os.environ["RANK"] = "0"
    os.environ["WORLD_SIZE"] = "1"
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        t = torch.ones(8, device="cuda")
        dist.all_reduce(t)
        dist.barrier()
        torch.cuda.synchronize()
        assert t.sum().item() == 8
    finally:
        dist.destroy_process_group()
        os.environ.pop("RANK", None)
        os.environ.pop("WORLD_SIZE", None)


def test_two_stage_partition_single_device():
    """Stage 0 + stage 1 chained on ONE GPU emit the same logits as the
    unsharded model — the per-stage forward the PP=2..8 runs execute,
    minus the wire."""
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.models.fuse import fuse_model
    from mlx_sharding_amd.parallel.rccl import build_stage_model, split_layers
    from mlx_sharding_amd.utils.presets import get_preset

    cfg = get_preset("debug-deepseek")
    dev = torch.device("cuda", 0)
    full = build_stage_model(cfg, 0, 1, dev, seed=3)
    sd = {k: v for k, v in full.state_dict().items()
          if "rope_inv_freq" not in k}
    cls = get_model_class(cfg.model_type)
    stages = []
    for r, (s, e) in enumerate(split_layers(cfg.num_hidden_layers, 2)):
        with torch.device(dev):
            m = cls(cfg, cfg.shard(s, e))
        m.load_weights(sd)
        m.eval()
        fuse_model(m)
        stages.append(m)
    torch.manual_seed(11)
    ids = torch.randint(0, cfg.vocab_size, (2, 9), device=dev)
    with torch.no_grad():
        ref_out = full(ids, full.make_cache(batch_size=2))
        h = stages[0](ids, stages[0].make_cache(batch_size=2))
        out = stages[1](h, stages[1].make_cache(batch_size=2))
    torch.cuda.synchronize()
    a = out[:, -1, :].float()
    b = ref_out[:, -1, :].float()
    assert torch.equal(a.argmax(-1), b.argmax(-1))
    assert (a - b).abs().max().item() < 0.05 * b.abs().max().item()


def test_serving_graph_reuse_matches_eager():
    """RcclPipeline's graphed B=1 decode: two back-to-back generations
    (capture, then re-arm over the same cache buffers) produce the same
    greedy tokens as the eager path."""
    from mlx_sharding_amd.parallel.engine import SamplingParams
    from mlx_sharding_amd.parallel.rccl import PipelineWorker, build_stage_model
    from mlx_sharding_amd.parallel.rccl_serve import RcclPipeline
    from mlx_sharding_amd.utils.presets import get_preset

    cfg = get_preset("debug-deepseek")
    dev = torch.device("cuda", 0)
    model = build_stage_model(cfg, 0, 1, dev, seed=5)
    worker = PipelineWorker(model, 0, 1, dev)
    pipe = RcclPipeline(worker)
    ids = torch.tensor([[5, 9, 2, 17, 3, 8]], dtype=torch.long)

    def gen_tokens(n):
        toks = []
        for tid, _ in pipe.generate_step(ids, SamplingParams(max_tokens=n)):
            toks.append(tid)
            if len(toks) >= n:
                break
        return toks

    t1 = gen_tokens(6)
    assert pipe._graph is not None, "graph path did not engage"
    g1 = pipe._graph
    t2 = gen_tokens(6)
    assert pipe._graph is g1, "graph was rebuilt instead of re-armed"
    assert t1 == t2, "graphed generations not reproducible"

    os.environ["MLXS_AMD_SERVE_GRAPH"] = "0"
    try:
        pipe2 = RcclPipeline(PipelineWorker(model, 0, 1, dev))
        t_eager = []
        for tid, _ in pipe2.generate_step(ids, SamplingParams(max_tokens=6)):
            t_eager.append(tid)
            if len(t_eager) >= 6:
                break
    finally:
        os.environ.pop("MLXS_AMD_SERVE_GRAPH", None)
    assert t1 == t_eager, f"graphed {t1} != eager {t_eager}"


def test_serving_graph_long_prompt_falls_back():
    """A generation that cannot fit the graph capacity must fall back
    to eager decode and still produce tokens."""
    from mlx_sharding_amd.parallel.engine import SamplingParams
    from mlx_sharding_amd.parallel.rccl import PipelineWorker, build_stage_model
    from mlx_sharding_amd.parallel.rccl_serve import RcclPipeline
    from mlx_sharding_amd.utils.presets import get_preset

    cfg = get_preset("debug-deepseek")
    dev = torch.device("cuda", 0)
    model = build_stage_model(cfg, 0, 1, dev, seed=5)
    os.environ["MLXS_AMD_SERVE_CAPACITY"] = "32"
    try:
        pipe = RcclPipeline(PipelineWorker(model, 0, 1, dev))
        ids = torch.randint(0, cfg.vocab_size, (1, 30))
        toks = []
        for tid, _ in pipe.generate_step(ids, SamplingParams(max_tokens=4)):
            toks.append(tid)
            if len(toks) >= 4:
                break
        assert len(toks) == 4
        assert pipe._graph is None
    finally:
        os.environ.pop("MLXS_AMD_SERVE_CAPACITY", None)


def test_grpc_two_stage_chain_on_gpu(tmp_path):
    """The reference's core workflow — shard server + driver chained
    over localhost gRPC — with BOTH stages running HIP kernels on the
    GPU (the CPU suite covers the same flow on torch reference ops)."""
    import json

    from safetensors.torch import save_file

    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.parallel.engine import SamplingParams, generate_step
    from mlx_sharding_amd.parallel.grpc_transport import StageClient
    from mlx_sharding_amd.server.shard_server import serve
    from mlx_sharding_amd.utils.loading import load_model
    from mlx_sharding_amd.utils.presets import get_preset

    cfg = get_preset("debug-llama")
    cls = get_model_class("llama")
    torch.manual_seed(13)
    m = cls(cfg, cfg.shard(0, cfg.num_hidden_layers))
    for p in m.parameters():
        p.data = p.data.float().normal_(0, 0.05).to(p.dtype)
    d = tmp_path / "ckpt"
    d.mkdir()
    save_file({k: v for k, v in m.state_dict().items()
               if "rope_inv_freq" not in k}, str(d / "model.safetensors"))
    with open(d / "config.json", "w") as f:
        json.dump(cfg.raw, f)

    server, _ = serve(str(d), 2, cfg.num_hidden_layers, port=0,
                      device="cuda", wait=False)
    try:
        client = StageClient(f"127.0.0.1:{server._mlxs_port}")
        m0, _ = load_model(d, 0, 2, device="cuda")
        ids = torch.randint(0, cfg.vocab_size, (1, 6),
                            generator=torch.Generator().manual_seed(3)).cuda()
        toks = []
        for tid, _ in generate_step(ids, m0, m0.make_cache(), [client],
                                    SamplingParams()):
            toks.append(tid)
            if len(toks) >= 5:
                break
        # single-process full model on GPU must agree (greedy)
        mf, _ = load_model(d, device="cuda")
        ref_toks = []
        for tid, _ in generate_step(ids, mf, mf.make_cache(), [],
                                    SamplingParams()):
            ref_toks.append(tid)
            if len(ref_toks) >= 5:
                break
        assert toks == ref_toks
        client.close()
    finally:
        server.stop(0)


@pytest.mark.parametrize("raw", [
    # head_dim 32 (the silent-skip shape), GQA
    dict(model_type="llama", hidden_size=256, num_hidden_layers=2,
         intermediate_size=512, num_attention_heads=8,
         num_key_value_heads=2, vocab_size=512),
    # head_dim 64, MHA
    dict(model_type="llama", hidden_size=512, num_hidden_layers=2,
         intermediate_size=768, num_attention_heads=8,
         num_key_value_heads=8, vocab_size=512),
    # head_dim 96 (no 96-dim prefill template: composed path)
    dict(model_type="llama", hidden_size=384, num_hidden_layers=2,
         intermediate_size=512, num_attention_heads=4,
         num_key_value_heads=2, vocab_size=512),
    # qwen2 remap: QKV biases through the fused-QKV path, and the
    # real model's GQA ratio 7 (28q/4kv -> G=7 decode case)
    dict(model_type="qwen2", hidden_size=448, num_hidden_layers=2,
         intermediate_size=512, num_attention_heads=14,
         num_key_value_heads=2, vocab_size=512),
    # gemma2 with 64-dim heads + softcap + window
    dict(model_type="gemma2", hidden_size=256, num_hidden_layers=2,
         intermediate_size=512, num_attention_heads=4,
         num_key_value_heads=2, head_dim=64, vocab_size=512,
         query_pre_attn_scalar=64, attn_logit_softcapping=50.0,
         final_logit_softcapping=30.0, sliding_window=16),
    # deepseek MLA with non-default dims
    dict(model_type="deepseek_v2", hidden_size=256, num_hidden_layers=2,
         intermediate_size=512, moe_intermediate_size=128,
         num_attention_heads=8, vocab_size=512, q_lora_rank=None,
         kv_lora_rank=128, qk_nope_head_dim=64, qk_rope_head_dim=32,
         v_head_dim=64, n_routed_experts=16, num_experts_per_tok=4,
         n_shared_experts=1, first_k_dense_replace=1, moe_layer_freq=1),
])
def test_native_vs_eager_shape_sweep(raw):
    """Teacher-forced prefill+decode A/B: the HIP dispatch must track
    MLXS_AMD_FORCE_TORCH eager across head-dim/shape variants — the
    generalized form of the smoke() numerics check (a silent kernel
    dispatch hole once returned uninitialized attention output for
    head_dim 32)."""
    from mlx_sharding_amd.config import ModelConfig
    from mlx_sharding_amd.parallel.rccl import build_stage_model

    raw = dict(raw, rms_norm_eps=1e-5, rope_theta=10000.0)
    cfg = ModelConfig.from_dict(raw)
    dev = torch.device("cuda", 0)
    model = build_stage_model(cfg, 0, 1, dev, seed=17)
    torch.manual_seed(4)
    ids = torch.randint(0, cfg.vocab_size, (2, 9), device=dev)

    def chain(force):
        os.environ.pop("MLXS_AMD_FORCE_TORCH", None)
        if force:
            os.environ["MLXS_AMD_FORCE_TORCH"] = "1"
        cache = model.make_cache(batch_size=2)
        outs = []
        with torch.no_grad():
            h = model(ids, cache)
            outs.append(h[:, -1, :].float())
            t = h[:, -1, :].argmax(-1, keepdim=True)
            for _ in range(2):
                h = model(t, cache)
                outs.append(h[:, -1, :].float())
        return outs

    try:
        a = chain(False)
        b = chain(True)
    finally:
        os.environ.pop("MLXS_AMD_FORCE_TORCH", None)
    for i, (x, y) in enumerate(zip(a, b)):
        cos = torch.nn.functional.cosine_similarity(
            x.flatten(), y.flatten(), dim=0).item()
        assert cos > 0.999, f"step {i}: native/eager diverge (cos={cos})"


def test_stage_timer_hip_events():
    """StageTimer's GPU path: HIP event pairs resolve lazily and report
    plausible wall times for a real kernel (utils/metrics.py)."""
    from mlx_sharding_amd.utils.metrics import StageTimer

    a = torch.randn(2048, 2048, device="cuda", dtype=torch.bfloat16)
    t = StageTimer()
    for _ in range(5):
        with t.measure(a.device):
            b = a @ a
    torch.cuda.synchronize()
    s = t.summary()
    assert s["count"] == 5
    assert 0.001 < s["mean_ms"] < 1000.0
    assert s["p50_ms"] <= s["p95_ms"] + 1e-6
    assert b.shape == (2048, 2048)


def test_shard_worker_times_forwards():
    """ShardWorker.forward is bracketed by the stage timer on GPU."""
    from mlx_sharding_amd.config import ModelConfig
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.server.shard_server import ShardWorker

    cfg = ModelConfig.from_dict({
        "model_type": "llama", "hidden_size": 64, "num_hidden_layers": 2,
        "intermediate_size": 128, "num_attention_heads": 4,
        "num_key_value_heads": 2, "vocab_size": 64, "rms_norm_eps": 1e-5,
        "rope_theta": 10000.0})
    m = get_model_class("llama")(cfg, cfg.shard(0, 2)).to("cuda").eval()
    w = ShardWorker(m)
    ids = torch.randint(0, 64, (1, 8), device="cuda")
    out = w.forward(ids)
    assert out.shape[1] == 8
    assert w.timer.summary()["count"] == 1


def test_quantize_on_load_generates_native(tmp_path):
    """Dense checkpoint -> quantize_weights at load -> native w4f16
    kernels on GPU: greedy tokens must track the dense model closely
    (quantize-on-load was previously CPU-tested only)."""
    import json

    from safetensors.torch import save_file

    from mlx_sharding_amd.config import ModelConfig
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.utils.loading import load_model

    cfg_raw = {"model_type": "llama", "hidden_size": 256,
               "num_hidden_layers": 2, "intermediate_size": 512,
               "num_attention_heads": 4, "num_key_value_heads": 2,
               "vocab_size": 128, "rms_norm_eps": 1e-5,
               "rope_theta": 10000.0}
    ckpt = tmp_path / "ckpt"
    ckpt.mkdir()
    with open(ckpt / "config.json", "w") as f:
        json.dump(cfg_raw, f)
    cfg = ModelConfig.from_dict(cfg_raw)
    torch.manual_seed(7)
    m = get_model_class("llama")(cfg, cfg.shard(0, 2))
    for p in m.parameters():
        p.data = p.data.float().normal_(0, 0.05).to(p.dtype)
    sd = {k: v for k, v in m.state_dict().items() if "rope_inv_freq" not in k}
    save_file(sd, str(ckpt / "model.safetensors"))

    dense, _ = load_model(ckpt, device="cuda")
    quant, qcfg = load_model(ckpt, device="cuda", quantize=(4, 64))
    assert qcfg.quantization is not None

    ids = torch.randint(0, 128, (1, 16), device="cuda")
    with torch.no_grad():
        ld = dense(ids, dense.make_cache(batch_size=1))[:, -1].float()
        lq = quant(ids, quant.make_cache(batch_size=1))[:, -1].float()
    cos = torch.nn.functional.cosine_similarity(ld, lq, dim=-1).item()
    assert cos > 0.98, f"int4-on-load logits diverged: cos={cos}"


def test_chunked_prefill_matches_full_on_gpu():
    """generate_step(prefill_chunk=8) must emit the same greedy tokens
    as a single full prefill — exercises attention with a non-zero
    cache offset at T>1 through the native kernels."""
    from mlx_sharding_amd.config import ModelConfig
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.parallel.engine import SamplingParams, generate_step

    cfg = ModelConfig.from_dict({
        "model_type": "llama", "hidden_size": 256, "num_hidden_layers": 2,
        "intermediate_size": 512, "num_attention_heads": 4,
        "num_key_value_heads": 2, "vocab_size": 128, "rms_norm_eps": 1e-5,
        "rope_theta": 10000.0})
    torch.manual_seed(11)
    m = get_model_class("llama")(cfg, cfg.shard(0, 2)).to("cuda").eval()
    ids = torch.randint(0, 128, (1, 29), device="cuda")  # odd length

    def first_tokens(chunk):
        g = generate_step(ids, m, m.make_cache(batch_size=1),
                          params=SamplingParams(temperature=0.0),
                          prefill_chunk=chunk)
        return [next(g)[0] for _ in range(6)]

    assert first_tokens(0) == first_tokens(8)


def test_api_concurrent_requests_on_gpu(tmp_path):
    """Two concurrent HTTP generations against one GPU-resident model:
    per-request caches + default-stream serialization must keep greedy
    outputs identical (threaded-server upgrade)."""
    import concurrent.futures as cf
    import http.client
    import json
    import threading

    from safetensors.torch import save_file
    from tokenizers import Tokenizer
    from tokenizers.models import WordLevel
    from tokenizers.pre_tokenizers import Whitespace

    from mlx_sharding_amd.config import ModelConfig
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.server import openai_api

    ckpt = tmp_path / "ckpt"
    ckpt.mkdir()
    vocab = {"<unk>": 0, "<eos>": 1, "hello": 2, "world": 3}
    vocab.update({f"t{i}": 4 + i for i in range(60)})
    tok = Tokenizer(WordLevel(vocab, unk_token="<unk>"))
    tok.pre_tokenizer = Whitespace()
    tok.save(str(ckpt / "tokenizer.json"))
    with open(ckpt / "tokenizer_config.json", "w") as f:
        json.dump({"tokenizer_class": "PreTrainedTokenizerFast",
                   "eos_token": "<eos>", "unk_token": "<unk>"}, f)
    cfg_raw = {"model_type": "llama", "hidden_size": 256,
               "num_hidden_layers": 2, "intermediate_size": 512,
               "num_attention_heads": 4, "num_key_value_heads": 2,
               "vocab_size": 64, "rms_norm_eps": 1e-5,
               "rope_theta": 10000.0}
    with open(ckpt / "config.json", "w") as f:
        json.dump(cfg_raw, f)
    cfg = ModelConfig.from_dict(cfg_raw)
    torch.manual_seed(3)
    m = get_model_class("llama")(cfg, cfg.shard(0, 2))
    for p in m.parameters():
        p.data = p.data.float().normal_(0, 0.05).to(p.dtype)
    sd = {k: v for k, v in m.state_dict().items() if "rope_inv_freq" not in k}
    save_file(sd, str(ckpt / "model.safetensors"))

    class Args:
        model = str(ckpt)
        llm_shard_addresses = ""
        start_layer = None
        end_layer = None

    provider = openai_api.ModelProvider(Args())
    assert next(provider.model.parameters()).is_cuda
    server = openai_api.run("127.0.0.1", 0, provider)
    threading.Thread(target=server.serve_forever, daemon=True).start()
    port = server.server_address[1]
    try:
        def one(_):
            conn = http.client.HTTPConnection("127.0.0.1", port, timeout=120)
            conn.request("POST", "/v1/completions",
                         json.dumps({"prompt": "hello world",
                                     "max_tokens": 8, "temperature": 0}),
                         {"Content-Type": "application/json"})
            resp = conn.getresponse()
            body = json.loads(resp.read())
            conn.close()
            assert resp.status == 200
            return body["choices"][0]["text"]

        with cf.ThreadPoolExecutor(3) as ex:
            texts = list(ex.map(one, range(3)))
        assert all(t == texts[0] for t in texts)
    finally:
        server.shutdown()


def test_prefix_cache_trim_parity_on_gpu():
    """KV trim + prefill_from on GPU: continuing from a reused prefix
    must emit the same greedy tokens as a fresh full prefill (the
    serving prefix cache's core invariant, at kernel level)."""
    from mlx_sharding_amd.config import ModelConfig
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.parallel.engine import SamplingParams, generate_step

    cfg = ModelConfig.from_dict({
        "model_type": "llama", "hidden_size": 256, "num_hidden_layers": 2,
        "intermediate_size": 512, "num_attention_heads": 4,
        "num_key_value_heads": 2, "vocab_size": 128, "rms_norm_eps": 1e-5,
        "rope_theta": 10000.0})
    torch.manual_seed(5)
    m = get_model_class("llama")(cfg, cfg.shard(0, 2)).to("cuda").eval()
    full = torch.randint(0, 128, (1, 24), device="cuda")

    def gen(ids, cache, prefill_from=0, n=5):
        g = generate_step(ids, m, cache, params=SamplingParams(),
                          prefill_from=prefill_from)
        return [next(g)[0] for _ in range(n)]

    ref = gen(full, m.make_cache(batch_size=1))

    # prior generation over the first 16 tokens fills a cache past 16;
    # trim back to 16 and continue with only the tail prefilled
    cache = m.make_cache(batch_size=1)
    gen(full[:, :16], cache, n=3)           # cache offset now 19
    for c in cache:
        c.trim(16)
    assert gen(full, cache, prefill_from=16) == ref
