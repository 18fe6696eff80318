"""RCCL serving mode: the OpenAI API driving an N-GPU pipeline.

One process per GPU (torchrun); rank 0 owns stage 0, the tokenizer and
the HTTP server; ranks 1..N-1 run a control loop.  Hidden-state hops go
stage→stage over RCCL/xGMI; the last stage ships ONLY the last-position
logits back to rank 0, which samples (full SamplingParams) — the
reference instead relayed every hop through the driver and shipped
[1, T, V] logits over gRPC.

Control plane: rank 0 broadcasts a small int64 op tensor before each
collective phase — the RCCL-adjacent equivalent of the reference's
ResetCache/SendTensor RPCs (§2.5 C3).
"""

from __future__ import annotations

import os
import threading
from typing import Generator, List, Tuple

import torch
import torch.distributed as dist

from .. import ops
from ..parallel.engine import SamplingParams
from .rccl import CapturedDecode, PipelineWorker

OP_SHUTDOWN = 0
OP_PREFILL = 1
OP_DECODE = 2


class RcclPipeline:
    """Rank-0 driver + worker control loop around a PipelineWorker."""

    def __init__(self, worker: PipelineWorker):
        self.worker = worker
        self.device = worker.device
        self.lock = threading.Lock()  # one generation at a time (global cache)
        # graph-captured single-stream decode (PP=1, B=1): captured once,
        # re-armed per generation over the SAME cache buffers
        self._graph = None
        self._graph_capacity = int(os.environ.get("MLXS_AMD_SERVE_CAPACITY",
                                                  "4096"))

    # -- control plane -----------------------------------------------------
    def _bcast(self, op: int, a: int = 0, b: int = 0):
        t = torch.tensor([op, a, b], dtype=torch.int64,
                         device="cpu")
        dist.broadcast(t, src=0)
        return t

    def worker_loop(self):
        """Ranks >0: serve control ops until shutdown."""
        w = self.worker
        while True:
            t = torch.zeros(3, dtype=torch.int64, device="cpu")
            dist.broadcast(t, src=0)
            op, a, b = int(t[0]), int(t[1]), int(t[2])
            if op == OP_SHUTDOWN:
                return
            if op == OP_PREFILL:
                w.prefill(None, a, 1, b, return_logits=True)
            elif op == OP_DECODE:
                w.decode_step_eager(None, a, 1, return_logits=True)

    def shutdown(self):
        if self.worker.world > 1 and self.worker.rank == 0:
            self._bcast(OP_SHUTDOWN)

    # -- graphed single-stream decode (PP=1, B=1) ---------------------------
    def _try_graph(self, B: int, T: int, params: SamplingParams) -> bool:
        """Arm (or build) the captured decode graph for this generation.
        Conditions: single stage, B=1, GPU, and the whole generation
        fits the fixed cache capacity.  The prefill that precedes this
        must have run with reuse_cache=True so the graph's captured
        cache buffers are still the live ones."""
        w = self.worker
        if (w.world > 1 or B != 1 or self.device.type != "cuda"
                or os.environ.get("MLXS_AMD_SERVE_GRAPH", "1") == "0"):
            return False
        need = T + (params.max_tokens or 512) + 8
        if need > self._graph_capacity:
            return False
        if self._graph is not None and self._graph.rearm():
            return True
        try:
            self._graph = CapturedDecode(w, 1, 1, self._graph_capacity,
                                         return_logits=True)
            return True
        except Exception as e:  # noqa: BLE001
            import sys
            print(f"serving graph capture failed, eager decode: {e}",
                  file=sys.stderr, flush=True)
            self._graph = None
            for caches in w.caches:
                for c in caches:
                    c.graph_pos = None
            return False

    # -- rank-0 generation --------------------------------------------------
    def generate_step(self, prompt_ids: torch.Tensor,
                      params: SamplingParams
                      ) -> Generator[Tuple[int, torch.Tensor], None, None]:
        """Mirror of engine.generate_step over the RCCL pipeline.

        Single-stream serving (PP=1, B=1, GPU) replays a hipGraph per
        decode step — the launch-bound regime where graphs measured
        ~5.5 ms/step vs ~13 eager (docs/PERFORMANCE.md) — and samples
        on-device with a one-step lookahead enqueue, so the per-token
        host sync overlaps the next replay."""
        w = self.worker
        assert w.is_first, "generate_step runs on rank 0"
        with self.lock, torch.no_grad():
            B, T = prompt_ids.shape
            if w.world > 1:
                self._bcast(OP_PREFILL, B, T)
            res = w.prefill([prompt_ids.to(self.device)], B, 1, T,
                            return_logits=True,
                            reuse_cache=self._graph is not None)
            use_graph = self._try_graph(B, T, params)
            gen = None
            if params.seed is not None:
                gen = torch.Generator(device="cpu").manual_seed(params.seed)
            rep_context: List[int] = prompt_ids[0].tolist()
            logits_dev = res[0].float()
            while True:
                logits = logits_dev  # [B, V]
                if params.logit_bias:
                    idx = torch.tensor(list(params.logit_bias.keys()),
                                       device=logits.device)
                    vals = torch.tensor(list(params.logit_bias.values()),
                                        device=logits.device, dtype=logits.dtype)
                    logits[0, idx] += vals
                if params.repetition_penalty and params.repetition_penalty != 1.0:
                    window = rep_context[-params.repetition_context_size:] \
                        if params.repetition_context_size else rep_context
                    ctx = torch.tensor(window, device=logits.device,
                                       dtype=torch.long)
                    logits = ops.apply_repetition_penalty(
                        logits, ctx, params.repetition_penalty)
                logprobs = logits - torch.logsumexp(logits, -1, keepdim=True)
                if use_graph:
                    # sample on-device, enqueue the next replay, THEN
                    # sync for the yield — launch latency hidden.
                    # (seeded sampling uses the CPU generator for
                    # reproducibility across devices)
                    if gen is not None:
                        tok = ops.sample(logits.cpu(), params.temperature,
                                         params.top_p, gen).to(logits.device)
                    else:
                        tok = ops.sample(logits, params.temperature,
                                         params.top_p, None)
                    self._graph.x_in[0].copy_(tok.reshape(B, 1))
                    self._graph.graphs[0].replay()
                    nxt_logits = self._graph.tok_out[0].clone()
                    tid = int(tok.item())
                    rep_context.append(tid)
                    yield tid, logprobs[0]
                    logits_dev = nxt_logits
                    continue
                tok = ops.sample(logits.cpu(), params.temperature,
                                 params.top_p, gen)
                tid = int(tok.item())
                rep_context.append(tid)
                yield tid, logprobs[0]
                if w.world > 1:
                    self._bcast(OP_DECODE, B)
                nxt = torch.full((B,), tid, dtype=torch.int64,
                                 device=self.device)
                res = w.decode_step_eager([nxt], B, 1, return_logits=True)
                logits_dev = res[0].float()


class RcclModelProvider:
    """ModelProvider-compatible provider backed by the RCCL pipeline."""

    def __init__(self, cli_args, pipeline: RcclPipeline, tokenizer):
        self.args = cli_args
        self.pipeline = pipeline
        self.tokenizer = tokenizer
        self.model = pipeline.worker.model
        self.remotes = []

    def load(self, model_path: str):
        # hot-swap is not supported across a live pipeline; serve the
        # pipeline's model for any requested name
        return self.model, self.tokenizer

    def generate(self, prompt_ids: List[int], params: SamplingParams):
        ids = torch.tensor([prompt_ids], dtype=torch.long)
        return self.pipeline.generate_step(ids, params)
