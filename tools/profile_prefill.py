"""Attribute prefill GPU time to torch ops (TTFT analysis)."""
import sys
import pathlib

sys.path.insert(0, str(pathlib.Path(__file__).parent.parent))
import torch

from mlx_sharding_amd.parallel.rccl import PipelineWorker, build_stage_model
from mlx_sharding_amd.utils.presets import get_preset

config = get_preset("deepseek-v2-lite", quant=True)
dev = torch.device("cuda", 0)
qc = config.quantization
model = build_stage_model(config, 0, 1, dev, quant_for=lambda p: qc)
worker = PipelineWorker(model, 0, 1, dev)

B, T = 64, 512
torch.manual_seed(0)
ids = [torch.randint(0, config.vocab_size, (B, T))]
# warm up twice (hipBLASLt heuristics, dequant caches)
worker.prefill(ids, B, 1, T)
worker.prefill(ids, B, 1, T)
torch.cuda.synchronize()

from torch.profiler import ProfilerActivity, profile

with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
    worker.prefill(ids, B, 1, T)
    torch.cuda.synchronize()

print(prof.key_averages().table(sort_by="cuda_time_total", row_limit=40,
                                max_name_column_width=60))
