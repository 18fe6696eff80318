"""Multi-process pipeline tests over gloo (world_size 2, CPU) — the same
code path bench.py / the RCCL engine run on GPUs (backend swap only)."""

import json
import os

import pytest
import torch
import torch.multiprocessing as mp

from mlx_sharding_amd.parallel.rccl import split_layers


def test_split_layers():
    assert split_layers(27, 2) == [(0, 14), (14, 27)]
    assert split_layers(8, 4) == [(0, 2), (2, 4), (4, 6), (6, 8)]
    assert split_layers(5, 2) == [(0, 3), (3, 5)]
    spans = split_layers(80, 8)
    assert spans[0][0] == 0 and spans[-1][1] == 80
    assert all(a[1] == b[0] for a, b in zip(spans, spans[1:]))


def _worker(rank, world, ckpt_dir, port, out_file, prompt, n_decode):
    import torch.distributed as dist

    from mlx_sharding_amd.parallel.rccl import PipelineWorker, split_layers
    from mlx_sharding_amd.utils.loading import load_model
    from mlx_sharding_amd.config import ModelConfig

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    cfg = ModelConfig.load(ckpt_dir)
    s, e = split_layers(cfg.num_hidden_layers, world)[rank]
    model, _ = load_model(ckpt_dir, s, e)
    worker = PipelineWorker(model, rank, world, torch.device("cpu"))

    ids = [torch.tensor([prompt], dtype=torch.long)]
    toks = worker.prefill(ids, 1, 1, len(prompt))
    seq = []
    if rank == 0:
        seq.append(int(toks[0].item()))
    for _ in range(n_decode):
        toks = worker.decode_step(toks, 1, 1)
        if rank == 0:
            seq.append(int(toks[0].item()))
    if rank == 0:
        with open(out_file, "w") as f:
            json.dump(seq, f)
    dist.barrier()
    dist.destroy_process_group()


@pytest.fixture
def tiny_checkpoint(tmp_path, tiny_llama_config):
    from safetensors.torch import save_file
    from mlx_sharding_amd.models import get_model_class
    from conftest import init_model
    cfg = tiny_llama_config
    cls = get_model_class("llama")
    m = init_model(cls, cfg, cfg.shard(0, cfg.num_hidden_layers), seed=11)
    sd = {k: v.clone() for k, v in m.state_dict().items()
          if "rope_inv_freq" not in k}
    d = tmp_path / "ckpt"
    d.mkdir()
    save_file(sd, str(d / "model.safetensors"))
    with open(d / "config.json", "w") as f:
        json.dump(cfg.raw, f)
    return d


def test_gloo_pp2_matches_single_process(tiny_checkpoint, tmp_path):
    from mlx_sharding_amd.parallel.engine import SamplingParams, generate_step
    from mlx_sharding_amd.utils.loading import load_model

    prompt = [5, 9, 2, 17, 3]
    n_decode = 6

    # single-process greedy reference
    mf, _ = load_model(tiny_checkpoint)
    cache = mf.make_cache()
    ref_seq = []
    ids = torch.tensor([prompt], dtype=torch.long)
    for tid, _ in generate_step(ids, mf, cache, []):
        ref_seq.append(tid)
        if len(ref_seq) >= n_decode + 1:
            break

    out_file = tmp_path / "out.json"
    mp.spawn(_worker, args=(2, str(tiny_checkpoint), 29531, str(out_file),
                            prompt, n_decode),
             nprocs=2, join=True)
    got = json.loads(out_file.read_text())
    assert got == ref_seq


def test_gloo_pp2_microbatched(tiny_checkpoint, tmp_path):
    """micro-batched decode (2 µbatches of 2 seqs) matches batched single-proc."""
    out_file = tmp_path / "mb.json"
    mp.spawn(_mb_worker, args=(2, str(tiny_checkpoint), 29532, str(out_file)),
             nprocs=2, join=True)
    got = json.loads(out_file.read_text())

    from mlx_sharding_amd.utils.loading import load_model
    mf, _ = load_model(tiny_checkpoint)
    torch.manual_seed(5)
    ids = torch.randint(0, 128, (4, 6))
    cache = mf.make_cache(batch_size=4)
    with torch.no_grad():
        h = mf(ids, cache)
        t = h[:, -1, :].float().argmax(-1)
        seqs = [t.tolist()]
        for _ in range(3):
            h = mf(t[:, None], cache)
            t = h[:, -1, :].float().argmax(-1)
            seqs.append(t.tolist())
    assert got == seqs


def _mb_worker(rank, world, ckpt_dir, port, out_file):
    import torch.distributed as dist

    from mlx_sharding_amd.parallel.rccl import PipelineWorker, split_layers
    from mlx_sharding_amd.utils.loading import load_model
    from mlx_sharding_amd.config import ModelConfig

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    cfg = ModelConfig.load(ckpt_dir)
    s, e = split_layers(cfg.num_hidden_layers, world)[rank]
    model, _ = load_model(ckpt_dir, s, e)
    worker = PipelineWorker(model, rank, world, torch.device("cpu"))

    torch.manual_seed(5)
    all_ids = torch.randint(0, 128, (4, 6))
    ids = [all_ids[0:2], all_ids[2:4]]  # 2 µbatches × 2 seqs
    toks = worker.prefill(ids, 2, 2, 6)
    seqs = []
    if rank == 0:
        seqs.append(torch.cat(toks).tolist())
    for _ in range(3):
        toks = worker.decode_step(toks, 2, 2)
        if rank == 0:
            seqs.append(torch.cat(toks).tolist())
    if rank == 0:
        with open(out_file, "w") as f:
            json.dump(seqs, f)
    dist.barrier()
    dist.destroy_process_group()


def _serve_worker(rank, world, ckpt_dir, port, out_file):
    import torch.distributed as dist
    from mlx_sharding_amd.config import ModelConfig
    from mlx_sharding_amd.parallel.engine import SamplingParams
    from mlx_sharding_amd.parallel.rccl import PipelineWorker, split_layers
    from mlx_sharding_amd.parallel.rccl_serve import RcclPipeline
    from mlx_sharding_amd.utils.loading import load_model

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    cfg = ModelConfig.load(ckpt_dir)
    s, e = split_layers(cfg.num_hidden_layers, world)[rank]
    model, _ = load_model(ckpt_dir, s, e)
    worker = PipelineWorker(model, rank, world, torch.device("cpu"))
    pipe = RcclPipeline(worker)
    if rank == 0:
        toks = []
        ids = torch.tensor([[5, 9, 2, 17]], dtype=torch.long)
        for tid, _ in pipe.generate_step(ids, SamplingParams()):
            toks.append(tid)
            if len(toks) >= 5:
                break
        # a second generation reuses the pipeline (fresh caches)
        toks2 = []
        for tid, _ in pipe.generate_step(ids, SamplingParams()):
            toks2.append(tid)
            if len(toks2) >= 5:
                break
        pipe.shutdown()
        with open(out_file, "w") as f:
            json.dump([toks, toks2], f)
    else:
        pipe.worker_loop()
    dist.barrier()
    dist.destroy_process_group()


def test_rccl_serve_pipeline_matches_single(tiny_checkpoint, tmp_path):
    from mlx_sharding_amd.parallel.engine import SamplingParams, generate_step
    from mlx_sharding_amd.utils.loading import load_model

    out_file = tmp_path / "serve.json"
    mp.spawn(_serve_worker, args=(2, str(tiny_checkpoint), 29533,
                                  str(out_file)), nprocs=2, join=True)
    got, got2 = json.loads(out_file.read_text())
    assert got == got2, "pipeline not reusable across generations"

    mf, _ = load_model(tiny_checkpoint)
    ids = torch.tensor([[5, 9, 2, 17]], dtype=torch.long)
    ref_toks = []
    for tid, _ in generate_step(ids, mf, mf.make_cache(), []):
        ref_toks.append(tid)
        if len(ref_toks) >= 5:
            break
    assert got == ref_toks


def test_rccl_serve_cli_end_to_end(tiny_checkpoint, tmp_path):
    """Full `mlx-sharding-rccl-serve` stack: torchrun 2 ranks (gloo/CPU),
    OpenAI request against rank 0's HTTP server."""
    import http.client
    import subprocess
    import sys
    import time
    from pathlib import Path

    # tokenizer files for the checkpoint
    from tokenizers import Tokenizer
    from tokenizers.models import WordLevel
    from tokenizers.pre_tokenizers import Whitespace
    vocab = {"<unk>": 0, "<eos>": 1}
    vocab.update({f"w{i}": 2 + i for i in range(126)})
    tk = Tokenizer(WordLevel(vocab, unk_token="<unk>"))
    tk.pre_tokenizer = Whitespace()
    tk.save(str(tiny_checkpoint / "tokenizer.json"))
    with open(tiny_checkpoint / "tokenizer_config.json", "w") as f:
        json.dump({"tokenizer_class": "PreTrainedTokenizerFast",
                   "eos_token": "<eos>", "unk_token": "<unk>"}, f)

    port = 18931
    proc = subprocess.Popen(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29541", "-m", "mlx_sharding_amd.cli.rccl_serve",
         "--model", str(tiny_checkpoint), "--port", str(port)],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        cwd=str(tmp_path),
        env={**os.environ,
             "PYTHONPATH": str(Path(__file__).parent.parent)})
    try:
        deadline = time.time() + 120
        up = False
        while time.time() < deadline:
            try:
                conn = http.client.HTTPConnection("127.0.0.1", port, timeout=5)
                conn.request("GET", "/")
                resp = conn.getresponse()
                resp.read()
                conn.close()
                up = True
                break
            except OSError:
                if proc.poll() is not None:
                    out = proc.stdout.read()
                    raise AssertionError(f"server died:\n{out[-2000:]}")
                time.sleep(1.0)
        assert up, "server did not come up"
        conn = http.client.HTTPConnection("127.0.0.1", port, timeout=120)
        conn.request("POST", "/v1/completions",
                     json.dumps({"prompt": "w1 w2", "max_tokens": 3,
                                 "temperature": 0}),
                     {"Content-Type": "application/json"})
        resp = conn.getresponse()
        body = json.loads(resp.read())
        conn.close()
        assert resp.status == 200
        assert body["usage"]["completion_tokens"] >= 1
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=20)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait(timeout=20)


@pytest.fixture
def tiny_quant_deepseek_checkpoint(tmp_path):
    """Coherently-quantized (w4a16) deepseek checkpoint: PP x quant x
    MoE has no other multi-process coverage, and saving/loading the
    packed uint32 + scales/biases layout round-trips the quant
    compatibility surface."""
    from safetensors.torch import save_file

    from conftest import init_model
    from mlx_sharding_amd.config import QuantConfig
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.models.base import Linear
    from mlx_sharding_amd.models.deepseek_v2 import _StackedLinear
    from mlx_sharding_amd.ops import reference as ref
    from mlx_sharding_amd.utils.presets import get_preset

    cfg = get_preset("debug-deepseek")
    cfg.raw["quantization"] = {"group_size": 32, "bits": 4}
    qc = QuantConfig(32, 4)
    cls = get_model_class("deepseek_v2")
    m = init_model(cls, cfg, cfg.shard(0, cfg.num_hidden_layers), seed=21,
                   quant_for=lambda p: qc)
    torch.manual_seed(9)
    for mod in m.modules():
        if isinstance(mod, Linear) and mod.quant is not None:
            w = torch.randn(mod.out_features, mod.in_features) * 0.05
            wq, sc, bi = ref.quantize(w.bfloat16(), 32, 4)
            mod.weight.data, mod.scales.data, mod.biases.data = wq, sc, bi
        elif isinstance(mod, _StackedLinear) and mod.quantc is not None:
            E, O = mod.weight.shape[0], mod.weight.shape[1]
            IN = mod.scales.shape[2] * 32
            ws, ss, bs = [], [], []
            for e in range(E):
                w = torch.randn(O, IN) * 0.05
                wq, sc, bi = ref.quantize(w.bfloat16(), 32, 4)
                ws.append(wq)
                ss.append(sc)
                bs.append(bi)
            mod.weight.data = torch.stack(ws)
            mod.scales.data = torch.stack(ss)
            mod.biases.data = torch.stack(bs)
    sd = {k: v.clone() for k, v in m.state_dict().items()
          if "rope_inv_freq" not in k}
    d = tmp_path / "qckpt"
    d.mkdir()
    save_file(sd, str(d / "model.safetensors"))
    with open(d / "config.json", "w") as f:
        json.dump(cfg.raw, f)
    return d


def test_gloo_pp2_quant_deepseek(tiny_quant_deepseek_checkpoint, tmp_path):
    """Quantized MoE deepseek over a 2-process pipeline emits the same
    greedy tokens as the single-process model."""
    from mlx_sharding_amd.parallel.engine import generate_step
    from mlx_sharding_amd.utils.loading import load_model

    prompt = [7, 3, 11, 2]
    n_decode = 4
    mf, _ = load_model(tiny_quant_deepseek_checkpoint)
    cache = mf.make_cache()
    ref_seq = []
    ids = torch.tensor([prompt], dtype=torch.long)
    for tid, _ in generate_step(ids, mf, cache, []):
        ref_seq.append(tid)
        if len(ref_seq) >= n_decode + 1:
            break

    out_file = tmp_path / "qout.json"
    mp.spawn(_worker, args=(2, str(tiny_quant_deepseek_checkpoint), 29537,
                            str(out_file), prompt, n_decode),
             nprocs=2, join=True)
    got = json.loads(out_file.read_text())
    assert got == ref_seq


def test_gloo_pp2_four_microbatches(tiny_checkpoint, tmp_path):
    """4 µbatches through 2 stages — deeper round-robin than the 2-µbatch
    test, exercising the posted-ahead irecv / isend overlap window."""
    out_file = tmp_path / "mb4.json"
    mp.spawn(_mb4_worker, args=(2, str(tiny_checkpoint), 29538, str(out_file)),
             nprocs=2, join=True)
    got = json.loads(out_file.read_text())

    from mlx_sharding_amd.utils.loading import load_model
    mf, _ = load_model(tiny_checkpoint)
    torch.manual_seed(6)
    ids = torch.randint(0, 128, (8, 5))
    cache = mf.make_cache(batch_size=8)
    with torch.no_grad():
        h = mf(ids, cache)
        t = h[:, -1, :].float().argmax(-1)
        seqs = [t.tolist()]
        for _ in range(4):
            h = mf(t[:, None], cache)
            t = h[:, -1, :].float().argmax(-1)
            seqs.append(t.tolist())
    assert got == seqs


def _mb4_worker(rank, world, ckpt_dir, port, out_file):
    import torch.distributed as dist

    from mlx_sharding_amd.config import ModelConfig
    from mlx_sharding_amd.parallel.rccl import PipelineWorker, split_layers
    from mlx_sharding_amd.utils.loading import load_model

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    cfg = ModelConfig.load(ckpt_dir)
    s, e = split_layers(cfg.num_hidden_layers, world)[rank]
    model, _ = load_model(ckpt_dir, s, e)
    worker = PipelineWorker(model, rank, world, torch.device("cpu"))

    torch.manual_seed(6)
    all_ids = torch.randint(0, 128, (8, 5))
    ids = [all_ids[i * 2:(i + 1) * 2] for i in range(4)]  # 4 µbatches × 2
    toks = worker.prefill(ids, 2, 4, 5)
    seqs = []
    if rank == 0:
        seqs.append(torch.cat(toks).tolist())
    for _ in range(4):
        toks = worker.decode_step(toks, 2, 4)
        if rank == 0:
            seqs.append(torch.cat(toks).tolist())
    if rank == 0:
        with open(out_file, "w") as f:
            json.dump(seqs, f)
    dist.barrier()
    dist.destroy_process_group()


def test_grpc_chain_quant_second_stage(tiny_quant_deepseek_checkpoint):
    """A QUANTIZED shard server for a non-first stage must cast the
    incoming hidden state to the activation dtype, not to its first
    parameter's dtype (which is a packed uint32 weight)."""
    from mlx_sharding_amd.parallel.engine import SamplingParams, generate_step
    from mlx_sharding_amd.parallel.grpc_transport import StageClient
    from mlx_sharding_amd.server.shard_server import serve
    from mlx_sharding_amd.utils.loading import load_model

    d = tiny_quant_deepseek_checkpoint
    server, _ = serve(str(d), 4, 8, port=0, wait=False)
    try:
        client = StageClient(f"127.0.0.1:{server._mlxs_port}")
        m0, _ = load_model(d, 0, 4)
        mf, _ = load_model(d)
        ids = torch.tensor([[7, 3, 11, 2]])
        def toks(model, remotes):
            out = []
            for tid, _ in generate_step(ids, model, model.make_cache(),
                                        remotes, SamplingParams()):
                out.append(tid)
                if len(out) >= 5:
                    break
            return out
        assert toks(m0, [client]) == toks(mf, [])
        client.close()
    finally:
        server.stop(0)
