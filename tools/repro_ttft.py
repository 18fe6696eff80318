import sys, time
sys.path.insert(0, "/root/repo")
import torch
from mlx_sharding_amd.parallel.rccl import PipelineWorker, build_stage_model
from mlx_sharding_amd.utils.presets import get_preset

cfg = get_preset("deepseek-v2-lite")
dev = torch.device("cuda", 0)
m = build_stage_model(cfg, 0, 1, dev)
w = PipelineWorker(m, 0, 1, dev)
ids = [torch.randint(0, cfg.vocab_size, (32, 512), device=dev)]
for i in range(3):
    torch.cuda.synchronize(); t0 = time.time()
    toks = w.prefill(ids, 32, 1, 512)
    torch.cuda.synchronize(); print(f"prefill {i}: {time.time()-t0:.3f}s")
# host-enqueue-only estimate: time without sync
t0 = time.time()
toks = w.prefill(ids, 32, 1, 512)
t_enq = time.time() - t0
torch.cuda.synchronize()
print(f"host enqueue time: {t_enq:.3f}s; total {time.time()-t0:.3f}s")
