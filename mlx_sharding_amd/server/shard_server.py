"""Shard server: holds one stage model and serves SendTensor/ResetCache.

Equivalent of /root/reference/shard/server/server.py — but session state
is explicit (a ShardWorker owns its cache; no module-level globals) and
forwards run under a lock so ResetCache cannot race an in-flight
forward (the reference's known race).
"""

from __future__ import annotations

import logging
from typing import List, Optional

import torch

from ..models.base import StageModel
from ..ops.kvcache import KVCache
from ..parallel.grpc_transport import serve_forward
from ..utils.loading import load_model
from ..utils.metrics import StageTimer

log = logging.getLogger(__name__)


class ShardWorker:
    def __init__(self, model: StageModel):
        self.model = model
        self.cache: Optional[List[KVCache]] = None
        self._device = next(model.parameters()).device
        self._dtype = model.fp_dtype  # NOT first-param dtype: that can
        #                               be a packed uint32 quant weight
        # per-stage HIP-event timing (SURVEY.md §5.1 — the reference has
        # only print statements, server.py:29-37); summary() is cheap and
        # logged every 1024 forwards at INFO
        self.timer = StageTimer()

    def reset(self):
        self.cache = None

    def forward(self, t: torch.Tensor) -> torch.Tensor:
        if t.is_floating_point():
            t = t.to(self._dtype)
        t = t.to(self._device)
        if self.cache is None or (self.cache and self.cache[0].batch_size != t.shape[0]):
            self.cache = self.model.make_cache(batch_size=t.shape[0])
        with torch.no_grad(), self.timer.measure(self._device):
            out = self.model(t, self.cache)
        if self.timer.total_calls % 1024 == 0:
            log.info("stage forward timing: %s", self.timer.summary())
        return out


def serve(model_path: str, start_layer: Optional[int] = None,
          end_layer: Optional[int] = None, port: int = 0,
          device: str = "cpu", dtype: Optional[torch.dtype] = None,
          wait: bool = True, quantize=None):
    model, config = load_model(model_path, start_layer, end_layer,
                               device=device, dtype=dtype, quantize=quantize)
    worker = ShardWorker(model)
    server = serve_forward(worker.forward, worker.reset, port=port)
    # startup line matches the reference so scripts that scrape the port
    # keep working (/root/reference/shard/server/server.py:90-92)
    print(f"Server started, listening on 0.0.0.0:{server._mlxs_port}",
          flush=True)
    if start_layer is not None or end_layer is not None:
        print(f"Model loaded with layers {start_layer or 0} to "
              f"{end_layer or 'end'}", flush=True)
    if wait:
        server.wait_for_termination()
    return server, worker
