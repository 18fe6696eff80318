"""Measure single-stream (B=1) API-style serving decode rate on GPU —
the VERDICT-9 acceptance metric (>= ~150 tok/s): graphed decode +
on-device sampling + one-step lookahead in RcclPipeline.generate_step."""
import sys
import time
import pathlib

sys.path.insert(0, str(pathlib.Path(__file__).parent.parent))
import torch

from mlx_sharding_amd.parallel.engine import SamplingParams
from mlx_sharding_amd.parallel.rccl import PipelineWorker, build_stage_model
from mlx_sharding_amd.parallel.rccl_serve import RcclPipeline
from mlx_sharding_amd.utils.presets import get_preset

name = sys.argv[1] if len(sys.argv) > 1 else "deepseek-v2-lite"
n_tok = int(sys.argv[2]) if len(sys.argv) > 2 else 200
config = get_preset(name, quant="deepseek" in name)
qc = config.quantization
dev = torch.device("cuda", 0)
model = build_stage_model(config, 0, 1, dev,
                          quant_for=(lambda p: qc) if qc else None)
pipe = RcclPipeline(PipelineWorker(model, 0, 1, dev))
torch.manual_seed(0)
ids = torch.randint(0, config.vocab_size, (1, 128))

def gen(n):
    toks = []
    t_first = t0 = time.perf_counter()
    for tid, _ in pipe.generate_step(ids, SamplingParams(max_tokens=n)):
        if not toks:
            t_first = time.perf_counter()
        toks.append(tid)
        if len(toks) >= n:
            break
    t_end = time.perf_counter()
    return toks, t_first - t0, (len(toks) - 1) / (t_end - t_first)

gen(16)          # warmup generation 1 (captures the serving graph)
toks, ttft, tps = gen(n_tok)
print(f"{name}: single-stream decode {tps:.1f} tok/s, "
      f"TTFT {ttft*1000:.1f} ms, {len(toks)} tokens "
      f"(graph={'armed' if pipe._graph is not None else 'eager'})",
      flush=True)
