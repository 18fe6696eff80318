"""RCCL pipeline: one process per GPU, stage hops over xGMI.

This replaces the reference's driver-relayed gRPC hops
(/root/reference/generate.py:72-78 — 2·N network crossings per token)
with direct stage_i→stage_{i+1} P2P send/recv through torch.distributed
(backend "nccl" IS RCCL on ROCm; "gloo" runs the same code on CPU for
tests).  The last stage samples on-GPU and ships ONE token id per
sequence back to stage 0 — never the [1, T, V] logits the reference
ships.

Decode is micro-batched: M micro-batches round-robin through the
stages, so stage s works on micro-batch m while stage s+1 works on m-1
— the inference analog of 1F1B, hiding the pipeline bubble whenever
the request batch ≥ the stage count.
"""

from __future__ import annotations

import os
import time
from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

from ..config import ModelConfig
from ..models import get_model_class
from ..models.base import StageModel
from ..ops.kvcache import KVCache


def split_layers(n_layers: int, n_stages: int) -> List[Tuple[int, int]]:
    """Balanced contiguous split; earlier stages get the remainder."""
    base = n_layers // n_stages
    rem = n_layers % n_stages
    out = []
    s = 0
    for i in range(n_stages):
        e = s + base + (1 if i < rem else 0)
        out.append((s, e))
        s = e
    return out


@dataclass
class PipelineConfig:
    hidden_size: int
    dtype: torch.dtype = torch.bfloat16
    micro_batches: int = 0  # 0 → min(batch, 2*world) chosen at run time


class PipelineWorker:
    """One pipeline stage = one process = one GPU (or CPU for tests)."""

    def __init__(self, model: StageModel, rank: int, world: int,
                 device: torch.device, dtype: torch.dtype = torch.bfloat16,
                 group: Optional[dist.ProcessGroup] = None):
        self.model = model
        self.rank = rank
        self.world = world
        self.device = device
        self.dtype = dtype
        self.group = group
        self.hidden = model.config.hidden_size
        self.caches: List[List[KVCache]] = []  # per micro-batch
        self._graph: Optional["CapturedDecode"] = None

    # -- helpers ----------------------------------------------------------
    @property
    def is_first(self) -> bool:
        return self.rank == 0

    @property
    def is_last(self) -> bool:
        return self.rank == self.world - 1

    @property
    def prev(self) -> int:
        return self.rank - 1

    @property
    def next(self) -> int:
        return self.rank + 1

    def reset(self, n_micro: int, micro_batch: int,
              reuse: bool = False) -> bool:
        """Fresh caches for a new generation.  ``reuse=True`` keeps the
        existing cache BUFFERS (offsets rewound to 0) when the layout
        matches — required for reusing a captured decode graph across
        generations, whose static tensors alias those buffers.  Returns
        True when the buffers were kept."""
        if (reuse and len(self.caches) == n_micro and self.caches
                and self.caches[0][0].batch_size == micro_batch
                and self.caches[0][0].keys_buffer() is not None):
            for caches in self.caches:
                for c in caches:
                    c.reset()
            return True
        self.caches = [self.model.make_cache(batch_size=micro_batch)
                       for _ in range(n_micro)]
        return False

    def _send(self, t: torch.Tensor, dst: int):
        dist.send(t.contiguous(), dst=dst, group=self.group)

    def _recv(self, shape, dtype) -> torch.Tensor:
        t = torch.empty(*shape, dtype=dtype, device=self.device)
        dist.recv(t, src=self.prev, group=self.group)
        return t

    # -- phases -----------------------------------------------------------
    def _ship_result(self, h, micro: int, return_logits: bool):
        """Last-stage result: greedy token ids [mb] (default) or the
        last-position logits [mb, V] for driver-side sampling (never the
        reference's [1, T, V] full-logit ship)."""
        if return_logits:
            return h[:, -1, :].to(self.dtype)
        return h[:, -1, :].float().argmax(-1)

    def _recv_result(self, micro: int, return_logits: bool) -> torch.Tensor:
        if return_logits:
            t = torch.empty(micro, self.model.config.vocab_size,
                            dtype=self.dtype, device=self.device)
        else:
            t = torch.empty(micro, dtype=torch.int64, device=self.device)
        dist.recv(t, src=self.world - 1, group=self.group)
        return t

    def prefill(self, ids_or_len, micro: int, n_micro: int, seq_len: int,
                return_logits: bool = False, reuse_cache: bool = False):
        """Run the prompt through this stage for every micro-batch.

        ids_or_len: on the first stage, list of [mb, T] id tensors; other
        stages only need (mb, T) shapes to size their recvs.

        Prefill hops are large ([mb, T, H] ≈ MBs), so recvs for ALL
        micro-batches are posted up front (irecv) and sends go out async
        (isend): stage s computes micro-batch m while m+1's hidden state
        is in flight — compute/comm overlap on the stage's streams.
        """
        self.reset(n_micro, micro, reuse=reuse_cache)
        out = []
        recv_bufs, recv_reqs, send_reqs, send_keep = [], [], [], []
        if not self.is_first:
            for _ in range(n_micro):
                buf = torch.empty(micro, seq_len, self.hidden,
                                  dtype=self.dtype, device=self.device)
                recv_bufs.append(buf)
                recv_reqs.append(dist.irecv(buf, src=self.prev,
                                            group=self.group))
        for m in range(n_micro):
            if self.is_first:
                x = ids_or_len[m].to(self.device)
            else:
                recv_reqs[m].wait()
                x = recv_bufs[m]
            with torch.no_grad():
                h = self.model(x, self.caches[m])
            if not self.is_last:
                hc = h.to(self.dtype).contiguous()
                send_keep.append(hc)  # keep alive until the isend completes
                send_reqs.append(dist.isend(hc, dst=self.next,
                                            group=self.group))
            else:
                out.append(self._ship_result(h, micro, return_logits))
        if self.is_last and not self.is_first:
            for m in range(n_micro):
                self._send(out[m].contiguous(), 0)
            for r in send_reqs:
                r.wait()
            return None
        if self.is_first and not self.is_last:
            res = [self._recv_result(micro, return_logits)
                   for _ in range(n_micro)]
            for r in send_reqs:
                r.wait()
            return res
        for r in send_reqs:
            r.wait()
        if self.is_first and self.is_last:
            return out
        return None

    def enable_graph_decode(self, tokens: Optional[List[torch.Tensor]],
                            micro: int, n_micro: int, capacity: int) -> bool:
        """Switch decode to hipGraph replay (GPU only).  ``tokens`` seeds
        the self-feeding input chain on the single-stage layout.  Falls
        back to eager decode (returning False) if capture fails — a
        failed capture must never take down a multi-rank run."""
        try:
            self._graph = CapturedDecode(self, micro, n_micro, capacity)
            if self.is_first and self.is_last and tokens is not None:
                self._graph.seed(tokens)
            return True
        except Exception as e:  # noqa: BLE001
            import sys
            print(f"[rank {self.rank}] hipGraph capture failed, using eager "
                  f"decode: {e}", file=sys.stderr, flush=True)
            self._graph = None
            for caches in self.caches:
                for c in caches:
                    c.graph_pos = None
            return False

    def decode_step(self, tokens: Optional[List[torch.Tensor]], micro: int,
                    n_micro: int) -> Optional[List[torch.Tensor]]:
        """One decode step over all micro-batches; returns next tokens on
        stage 0 (or the single stage)."""
        if self._graph is not None:
            return self._graph.decode_step(tokens)
        return self.decode_step_eager(tokens, micro, n_micro)

    def _post_result_recvs(self, micro: int, n_micro: int,
                           return_logits: bool):
        """First stage: post irecvs for the last stage's per-µbatch
        results up front so they land as they are produced."""
        bufs, reqs = [], []
        for _ in range(n_micro):
            if return_logits:
                t = torch.empty(micro, self.model.config.vocab_size,
                                dtype=self.dtype, device=self.device)
            else:
                t = torch.empty(micro, dtype=torch.int64, device=self.device)
            bufs.append(t)
            reqs.append(dist.irecv(t, src=self.world - 1, group=self.group))
        return bufs, reqs

    def decode_step_eager(self, tokens, micro: int, n_micro: int,
                          return_logits: bool = False):
        """One decode step, µbatches round-robin through the stages.

        Comms are overlapped with compute: every µbatch's incoming
        hidden-state recv is posted up front (irecv) and outgoing hops
        go out async (isend), so stage s computes µbatch m while m+1's
        activation is in flight and m-1's is on its way downstream —
        no per-hop serialization (the inference-side 1F1B analog)."""
        timing = os.environ.get("MLXS_STAGE_TIMING", "0") == "1"
        if timing and self.device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter() if timing else 0.0
        out: List[torch.Tensor] = []
        recv_bufs, recv_reqs = [], []
        send_reqs, send_keep = [], []
        res_bufs = res_reqs = None
        if not self.is_first:
            for _ in range(n_micro):
                buf = torch.empty(micro, 1, self.hidden, dtype=self.dtype,
                                  device=self.device)
                recv_bufs.append(buf)
                recv_reqs.append(dist.irecv(buf, src=self.prev,
                                            group=self.group))
        elif not self.is_last:
            res_bufs, res_reqs = self._post_result_recvs(micro, n_micro,
                                                         return_logits)
        for m in range(n_micro):
            if self.is_first:
                x = tokens[m].reshape(micro, 1)
            else:
                recv_reqs[m].wait()
                x = recv_bufs[m]
            with torch.no_grad():
                h = self.model(x, self.caches[m])
            if not self.is_last:
                hc = h.to(self.dtype).contiguous()
                send_keep.append(hc)  # alive until the isend completes
                send_reqs.append(dist.isend(hc, dst=self.next,
                                            group=self.group))
            else:
                out.append(self._ship_result(h, micro, return_logits))
        if self.is_last and not self.is_first:
            for m in range(n_micro):
                send_keep.append(out[m].contiguous())
                send_reqs.append(dist.isend(send_keep[-1], dst=0,
                                            group=self.group))
            for r in send_reqs:
                r.wait()
            return None
        if timing:
            if self.device.type == "cuda":
                torch.cuda.synchronize()
            print(f"[stage {self.rank}] decode step "
                  f"{(time.perf_counter() - t0) * 1000:.2f} ms "
                  f"({n_micro}x{micro})", flush=True)
        if self.is_first and not self.is_last:
            for r in res_reqs:
                r.wait()
            for r in send_reqs:
                r.wait()
            return res_bufs
        for r in send_reqs:
            r.wait()
        return out if self.is_first else None


def init_distributed(backend: Optional[str] = None) -> Tuple[int, int, torch.device]:
    """Init from torchrun env (RANK/WORLD_SIZE/LOCAL_RANK)."""
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local = int(os.environ.get("LOCAL_RANK", str(rank)))
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29512")
        dist.init_process_group(backend=backend, rank=rank, world_size=world)
    if backend == "nccl":
        torch.cuda.set_device(local)
        device = torch.device("cuda", local)
    else:
        device = torch.device("cpu")
    return rank, world, device


def build_stage_model(config: ModelConfig, rank: int, world: int,
                      device: torch.device, seed: int = 0,
                      quant_for=None) -> StageModel:
    """Random-init one stage (synthetic-weights path for bench/tests)."""
    ranges = split_layers(config.num_hidden_layers, world)
    s, e = ranges[rank]
    cls = get_model_class(config.model_type)
    torch.manual_seed(seed)
    with torch.device(device):  # allocate directly on-device (big models)
        model = cls(config, config.shard(s, e), quant_for=quant_for)
    with torch.no_grad():
        for name, p in model.named_parameters():
            if p.is_floating_point():
                if name.endswith(".scales"):
                    # dequant w = s*q + b with q ~ U[0, 2^bits): match the
                    # dense init's std (0.02) so synthetic quant models
                    # route/activate like dense ones — N(0, 0.02) scales
                    # give ~9x-hot weights, heavy-tailed router logits and
                    # degenerate expert load (3x prefill cap blow-up)
                    p.data.normal_(0, 0.0025)
                elif name.endswith(".biases"):
                    p.data.normal_(0, 0.005)
                else:
                    p.data.normal_(0, 0.02)
            elif p.dtype in (getattr(torch, "uint32", None), torch.int32):
                p.data.random_(0, 2 ** 31 - 1)
            p.requires_grad_(False)
    model.eval()
    from ..models.fuse import fuse_model
    fuse_model(model)
    return model


class CapturedDecode:
    """hipGraph-captured decode step for one stage's micro-batches.

    Captures the whole per-µbatch stage forward (norms, GEMMs, fused MoE
    gating + grouped experts, flash-decode attention) in ONE hipGraph
    and replays it per decode step — the launch-bound inner loop the
    CDNA4 guide says to capture.  The KV position lives on device
    (KVCache.graph_pos); the attention kernel reads S = pos+1 on device
    and pos.add_(1) is part of the graph, so replays need zero host
    work.  Comms (RCCL send/recv) stay eager between replays, writing
    into the graph's static input buffers.
    """

    def __init__(self, worker: "PipelineWorker", micro: int, n_micro: int,
                 capacity: int, return_logits: bool = False):
        assert torch.cuda.is_available(), "graph capture needs a GPU"
        self.worker = worker
        self.micro = micro
        self.n_micro = n_micro
        self.return_logits = return_logits
        self.capacity = capacity
        w = worker
        dev = w.device
        self.graphs = []
        self.x_in = []      # static input (tokens or hidden)
        self.h_out = []     # static output hidden (non-last stages)
        self.tok_out = []   # static output tokens/logits (last stage)
        self.pos = []

        for m in range(n_micro):
            caches = w.caches[m]
            off = caches[0].offset
            for c in caches:
                c.ensure_capacity(capacity, micro)
            pos = torch.tensor([off], dtype=torch.int32, device=dev)
            for c in caches:
                c.graph_pos = pos
            if w.is_first:
                x = torch.zeros(micro, 1, dtype=torch.int64, device=dev)
            else:
                x = torch.zeros(micro, 1, w.hidden, dtype=w.dtype, device=dev)

            def step_fn(x=x, caches=caches, pos=pos):
                with torch.no_grad():
                    h = w.model(x, caches)
                out_h = None
                out_t = None
                if w.is_last:
                    if return_logits:
                        # serving: ship last-position logits; sampling
                        # happens outside the graph
                        out_t = h[:, -1, :].float()
                    else:
                        out_t = h[:, -1, :].float().argmax(-1)
                        if w.is_first:
                            x.copy_(out_t.reshape(micro, 1))  # self-feed
                else:
                    out_h = h.to(w.dtype)
                pos.add_(1)
                return out_h, out_t

            # warmup on a side stream (hipblaslt workspaces etc.); the two
            # garbage steps write cache slots off/off+1 — rewind pos and the
            # real steps overwrite them before they are ever attended to.
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):
                    step_fn()
            torch.cuda.current_stream().wait_stream(s)
            pos.fill_(off)
            torch.cuda.synchronize()

            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                out_h, out_t = step_fn()
            self.graphs.append(g)
            self.x_in.append(x)
            self.h_out.append(out_h)
            self.tok_out.append(out_t)
            self.pos.append(pos)
        # identity of the captured cache buffers, for rearm() validity
        self._cap_bufs = [tuple(c.keys_buffer() for c in w.caches[m])
                          for m in range(n_micro)]

    def seed(self, tokens: List[torch.Tensor]):
        for m in range(self.n_micro):
            self.x_in[m].copy_(tokens[m].reshape(self.micro, 1))

    def rearm(self) -> bool:
        """Re-attach this graph to the worker's caches for a NEW
        generation: the caches must be the same buffer objects the graph
        captured (PipelineWorker.reset(reuse=True)); positions restart
        from the fresh prefill offset.  Returns False if the cache
        objects changed (graph invalid — rebuild)."""
        w = self.worker
        for m in range(self.n_micro):
            caches = w.caches[m]
            for c, cap_buf in zip(caches, self._cap_bufs[m]):
                if c.keys_buffer() is not cap_buf:
                    return False  # buffers were reallocated
            self.pos[m].fill_(caches[0].offset)
            for c in caches:
                c.graph_pos = self.pos[m]
        return True

    def decode_step(self, tokens: Optional[List[torch.Tensor]]):
        """Replay per-µbatch graphs with overlapped comms: recvs into
        the static input buffers are posted for ALL µbatches up front
        and outgoing hops are isends, so replay(m) runs while m+1's
        activation is still in flight (same 1F1B analog as the eager
        path; the graphs themselves stay comm-free)."""
        w = self.worker
        out_tokens: List[torch.Tensor] = []
        recv_reqs = []
        send_reqs, send_keep = [], []
        res_bufs = res_reqs = None
        if not w.is_first:
            recv_reqs = [dist.irecv(self.x_in[m], src=w.prev, group=w.group)
                         for m in range(self.n_micro)]
        elif not w.is_last:
            res_bufs, res_reqs = w._post_result_recvs(self.micro,
                                                      self.n_micro, False)
        for m in range(self.n_micro):
            if w.is_first and not w.is_last:
                self.x_in[m].copy_(tokens[m].reshape(self.micro, 1))
            if not w.is_first:
                recv_reqs[m].wait()
            self.graphs[m].replay()
            if not w.is_last:
                send_reqs.append(dist.isend(self.h_out[m], dst=w.next,
                                            group=w.group))
            else:
                out_tokens.append(self.tok_out[m])
        if w.is_last and not w.is_first:
            for m in range(self.n_micro):
                send_keep.append(out_tokens[m].to(torch.int64))
                send_reqs.append(dist.isend(send_keep[-1], dst=0,
                                            group=w.group))
            for r in send_reqs:
                r.wait()
            return None
        if w.is_first and not w.is_last:
            for r in res_reqs:
                r.wait()
            for r in send_reqs:
                r.wait()
            return res_bufs
        # middle stages: the next step's replay(m) overwrites h_out[m],
        # so the isend must be complete (stream-ordered under NCCL,
        # host-blocking under gloo) before returning
        for r in send_reqs:
            r.wait()
        return out_tokens if w.is_first else None
