"""Driver-side generation loop over a chain of pipeline stages.

Mirrors the reference's driver loops (greedy CLI at
/root/reference/generate.py:52-88; full-sampling closure at
/root/reference/shard/utils.py:111-188): local first shard forward →
chain remote stages → sample on the last logits.  Remote stages are
abstract (`StageHandle`): gRPC clients (CPU plumbing / cross-node) or
the RCCL in-node pipeline (parallel/rccl.py).
"""

from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Generator, Iterable, List, Optional, Protocol, Tuple

import torch

from .. import ops
from ..models.base import StageModel
from ..ops.kvcache import KVCache


class StageHandle(Protocol):
    def send_tensor(self, t: torch.Tensor, wire_fp16: bool = False,
                    device: str = "cpu") -> torch.Tensor: ...

    def reset_cache(self) -> None: ...


@dataclass
class SamplingParams:
    temperature: float = 0.0
    top_p: float = 1.0
    repetition_penalty: Optional[float] = None
    repetition_context_size: int = 20
    logit_bias: Optional[dict] = None
    seed: Optional[int] = None
    # upper bound on tokens this generation will draw (sizing hint for
    # graph-captured serving decode; the caller still enforces its own
    # stop conditions)
    max_tokens: Optional[int] = None


def generate_step(
    prompt_ids: torch.Tensor,
    model: StageModel,
    cache: List[KVCache],
    remotes: Iterable[StageHandle] = (),
    params: Optional[SamplingParams] = None,
    wire_fp16: bool = False,
    prefill_chunk: int = 0,
    prefill_from: int = 0,
) -> Generator[Tuple[int, torch.Tensor], None, None]:
    """Yield (token_id, logprobs[V]) forever; caller decides when to stop.

    Resets every remote stage's cache at generator start (the
    reference's per-request lifecycle, utils.py:122-124); the local
    ``cache`` must be fresh.

    ``prefill_chunk`` > 0 feeds the prompt in chunks of that many
    tokens (bounded activation memory for long prompts; each chunk is
    one more SendTensor to the remote stages, whose caches accumulate
    — the on-GPU chunked-prefill equivalence is kernel-tested).

    ``prefill_from`` > 0 skips prefilling the first N prompt tokens:
    the caller guarantees ``cache`` already holds their K/V (prefix
    caching — the caches were trimmed to exactly N).  Only valid
    without remotes (remote stages have no trim control channel);
    0 < prefill_from < prompt length.
    """
    params = params or SamplingParams()
    remotes = list(remotes)
    if prefill_from:
        if remotes:
            raise ValueError("prefill_from requires a purely local chain")
        if not 0 < prefill_from < prompt_ids.shape[1]:
            raise ValueError(f"prefill_from {prefill_from} outside "
                             f"(0, {prompt_ids.shape[1]})")
        if any(c.offset != prefill_from for c in cache):
            raise ValueError("cache offsets do not match prefill_from")
    for r in remotes:
        r.reset_cache()
    device = prompt_ids.device
    gen = None
    if params.seed is not None:
        gen = torch.Generator(device="cpu").manual_seed(params.seed)

    rep_context: List[int] = prompt_ids[0].tolist()

    def _forward(y):
        h = model(y, cache)
        for r in remotes:
            h = r.send_tensor(h, wire_fp16=wire_fp16, device=str(device))
        return h

    with torch.no_grad():
        fresh = prompt_ids[:, prefill_from:] if prefill_from else prompt_ids
        if prefill_chunk and fresh.shape[1] > prefill_chunk:
            for c0 in range(0, fresh.shape[1], prefill_chunk):
                h = _forward(fresh[:, c0: c0 + prefill_chunk])
        else:
            h = _forward(fresh)
        while True:
            logits = h[:, -1, :].float()
            if params.logit_bias:
                idx = torch.tensor(list(params.logit_bias.keys()), device=device)
                vals = torch.tensor(list(params.logit_bias.values()),
                                    device=device, dtype=logits.dtype)
                logits[0, idx] += vals
            if params.repetition_penalty and params.repetition_penalty != 1.0:
                window = rep_context[-params.repetition_context_size:] \
                    if params.repetition_context_size else rep_context
                ctx = torch.tensor(window, device=device, dtype=torch.long)
                logits = ops.apply_repetition_penalty(
                    logits, ctx, params.repetition_penalty)
            logprobs = logits - torch.logsumexp(logits, dim=-1, keepdim=True)
            tok = ops.sample(logits, params.temperature, params.top_p, gen)
            # one-step lookahead (the reference's mx.async_eval,
            # generate.py:82-88): enqueue the NEXT forward before the
            # host sync on .item(), hiding its launch latency behind
            # the caller's per-token work.  The final lookahead step is
            # wasted compute, exactly as in the reference.
            h = _forward(tok.reshape(-1, 1))
            tid = int(tok.item())
            rep_context.append(tid)
            yield tid, logprobs[0]


@dataclass
class GenerationStats:
    prompt_tokens: int = 0
    generation_tokens: int = 0
    prompt_tps: float = 0.0
    generation_tps: float = 0.0
    ttft_s: float = 0.0


def stream_generate(
    prompt_ids: torch.Tensor,
    model: StageModel,
    remotes: Iterable[StageHandle] = (),
    max_tokens: int = 256,
    params: Optional[SamplingParams] = None,
    eos_token_ids: Iterable[int] = (),
    wire_fp16: bool = False,
    stats: Optional[GenerationStats] = None,
) -> Generator[Tuple[int, torch.Tensor], None, None]:
    """Token stream with EOS stop + the reference's TPS accounting
    (prompt TPS = prompt_tokens / time-to-first-token; gen TPS =
    (n-1)/decode_time — /root/reference/generate.py:103-122)."""
    eos = set(int(e) for e in eos_token_ids)
    cache = model.make_cache(batch_size=prompt_ids.shape[0])
    t0 = time.perf_counter()
    n = 0
    t_first = t0
    for tid, logprobs in generate_step(prompt_ids, model, cache, remotes,
                                       params, wire_fp16):
        if n == 0:
            t_first = time.perf_counter()
        n += 1
        if tid in eos:
            break
        yield tid, logprobs
        if n >= max_tokens:
            break
    t_end = time.perf_counter()
    if stats is not None:
        stats.prompt_tokens = int(prompt_ids.shape[1])
        stats.generation_tokens = n
        stats.ttft_s = t_first - t0
        stats.prompt_tps = stats.prompt_tokens / max(t_first - t0, 1e-9)
        stats.generation_tps = (n - 1) / max(t_end - t_first, 1e-9) if n > 1 else 0.0


class LocalChain:
    """In-process stage chain (testing + PP=1): behaves like a remote."""

    def __init__(self, model: StageModel, batch_size: int = 1):
        self.model = model
        self.batch_size = batch_size
        self.cache = None

    def send_tensor(self, t: torch.Tensor, wire_fp16: bool = False,
                    device: str = "cpu") -> torch.Tensor:
        if wire_fp16 and t.dtype == torch.bfloat16:
            t = t.to(torch.float16)
        x = t.to(self.model.fp_dtype) if t.is_floating_point() else t
        if self.cache is None or (self.cache and self.cache[0].batch_size != x.shape[0]):
            self.cache = self.model.make_cache(batch_size=x.shape[0])
        with torch.no_grad():
            return self.model(x, self.cache)

    def reset_cache(self):
        self.cache = None
