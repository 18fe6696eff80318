"""Observability: per-stage timing + machine-scrapeable serving metrics.

The reference has NO metrics subsystem — only ad-hoc tokens-per-sec
prints (/root/reference/generate.py:115-122) and server-side `print`
(/root/reference/shard/server/server.py:29-37); SURVEY.md §5.1/§5.5
calls out per-stage HIP-event timing and machine-scrapeable metrics as
the build's upgrade.  This module provides both without any external
dependency:

- ``StageTimer``: brackets each stage forward with HIP events
  (``torch.cuda.Event(enable_timing=True)``) when the tensors live on a
  GPU, falling back to ``time.perf_counter`` on CPU.  Event pairs are
  resolved lazily (``elapsed_time`` syncs only the recorded event, not
  the device), so timing never serializes the decode loop.
- ``Counter`` / ``Histogram`` / ``Registry``: minimal Prometheus-style
  primitives rendering the text exposition format (``render()``), served
  by the API server at ``GET /metrics`` alongside ``GET /health``.
"""

from __future__ import annotations

import threading
import time
from bisect import bisect_left
from collections import deque
from typing import Deque, Dict, List, Optional, Tuple

import torch


# ---------------------------------------------------------------------------
# Stage timing (HIP events on GPU, perf_counter on CPU)
# ---------------------------------------------------------------------------

class StageTimer:
    """Per-forward wall time of one pipeline stage.

    Usage::

        with timer.measure(device):
            out = model(x, cache)

    GPU path records a HIP event pair on the current stream and resolves
    the pair the NEXT time stats are requested — no sync inside the hot
    loop.  ``summary()`` returns count/mean/p50/p95 in milliseconds.
    """

    def __init__(self, window: int = 512):
        self._samples: Deque[float] = deque(maxlen=window)
        self._pending: Deque[Tuple[torch.cuda.Event, torch.cuda.Event]] = deque()
        self._lock = threading.Lock()
        self.total_calls = 0

    class _Ctx:
        def __init__(self, timer: "StageTimer", use_gpu: bool):
            self.timer = timer
            self.use_gpu = use_gpu

        def __enter__(self):
            if self.use_gpu:
                self.e0 = torch.cuda.Event(enable_timing=True)
                self.e0.record()
            else:
                self.t0 = time.perf_counter()
            return self

        def __exit__(self, *exc):
            with self.timer._lock:
                self.timer.total_calls += 1
                if self.use_gpu:
                    e1 = torch.cuda.Event(enable_timing=True)
                    e1.record()
                    self.timer._pending.append((self.e0, e1))
                else:
                    self.timer._samples.append(
                        (time.perf_counter() - self.t0) * 1e3)
            return False

    def measure(self, device=None) -> "StageTimer._Ctx":
        use_gpu = torch.cuda.is_available() and (
            device is None or str(device).startswith("cuda"))
        return StageTimer._Ctx(self, use_gpu)

    def _drain(self):
        # resolve completed event pairs; stop at the first unfinished one
        while self._pending:
            e0, e1 = self._pending[0]
            if not e1.query():
                break
            self._pending.popleft()
            self._samples.append(e0.elapsed_time(e1))

    def summary(self) -> Dict[str, float]:
        with self._lock:
            if self._pending:
                # sync only the newest recorded event, then drain all
                self._pending[-1][1].synchronize()
            self._drain()
            xs = sorted(self._samples)
        if not xs:
            return {"count": 0, "mean_ms": 0.0, "p50_ms": 0.0, "p95_ms": 0.0}
        return {
            "count": len(xs),
            "mean_ms": sum(xs) / len(xs),
            "p50_ms": xs[len(xs) // 2],
            "p95_ms": xs[min(len(xs) - 1, int(len(xs) * 0.95))],
        }


# ---------------------------------------------------------------------------
# Prometheus-style registry (text exposition format, no dependency)
# ---------------------------------------------------------------------------

class Counter:
    def __init__(self, name: str, help_: str):
        self.name, self.help = name, help_
        self._v = 0.0
        self._lock = threading.Lock()

    def inc(self, by: float = 1.0):
        with self._lock:
            self._v += by

    @property
    def value(self) -> float:
        return self._v

    def render(self) -> str:
        return (f"# HELP {self.name} {self.help}\n"
                f"# TYPE {self.name} counter\n"
                f"{self.name} {self._v:g}\n")


class Histogram:
    """Fixed-bucket histogram (Prometheus cumulative-bucket semantics)."""

    DEFAULT_BUCKETS = (1, 2.5, 5, 10, 25, 50, 100, 250, 500, 1000, 2500,
                       5000, 10000)

    def __init__(self, name: str, help_: str,
                 buckets: Optional[Tuple[float, ...]] = None):
        self.name, self.help = name, help_
        self.buckets: List[float] = sorted(buckets or self.DEFAULT_BUCKETS)
        self._counts = [0] * (len(self.buckets) + 1)  # +inf bucket
        self._sum = 0.0
        self._n = 0
        self._lock = threading.Lock()

    def observe(self, v: float):
        with self._lock:
            # first bucket with v <= le, else the +inf slot at the end
            self._counts[bisect_left(self.buckets, v)] += 1
            self._sum += v
            self._n += 1

    @property
    def count(self) -> int:
        return self._n

    @property
    def sum(self) -> float:
        return self._sum

    def render(self) -> str:
        out = [f"# HELP {self.name} {self.help}",
               f"# TYPE {self.name} histogram"]
        cum = 0
        for b, c in zip(self.buckets, self._counts):
            cum += c
            out.append(f'{self.name}_bucket{{le="{b:g}"}} {cum}')
        cum += self._counts[-1]
        out.append(f'{self.name}_bucket{{le="+Inf"}} {cum}')
        out.append(f"{self.name}_sum {self._sum:g}")
        out.append(f"{self.name}_count {self._n}")
        return "\n".join(out) + "\n"


class Registry:
    def __init__(self):
        self._metrics: Dict[str, object] = {}
        self._lock = threading.Lock()

    def counter(self, name: str, help_: str = "") -> Counter:
        with self._lock:
            m = self._metrics.get(name)
            if m is None:
                m = self._metrics[name] = Counter(name, help_)
            return m  # type: ignore[return-value]

    def histogram(self, name: str, help_: str = "",
                  buckets: Optional[Tuple[float, ...]] = None) -> Histogram:
        with self._lock:
            m = self._metrics.get(name)
            if m is None:
                m = self._metrics[name] = Histogram(name, help_, buckets)
            return m  # type: ignore[return-value]

    def render(self) -> str:
        with self._lock:
            return "".join(m.render() for m in self._metrics.values())


# Default registry used by the API server (tests may construct their own).
REGISTRY = Registry()


def serving_metrics(reg: Optional[Registry] = None):
    """The API server's standard metric set (created on first use)."""
    reg = reg or REGISTRY
    return {
        "requests": reg.counter(
            "mlxs_requests_total", "completed /v1 completion requests"),
        "errors": reg.counter(
            "mlxs_request_errors_total", "requests that returned an error"),
        "prompt_tokens": reg.counter(
            "mlxs_prompt_tokens_total", "prompt tokens processed"),
        "gen_tokens": reg.counter(
            "mlxs_generation_tokens_total", "tokens generated"),
        "ttft_ms": reg.histogram(
            "mlxs_ttft_ms", "time to first token (ms)"),
        "decode_tps": reg.histogram(
            "mlxs_decode_tokens_per_s", "per-request decode throughput",
            buckets=(1, 5, 10, 25, 50, 100, 150, 200, 300, 500, 1000,
                     5000, 20000)),
    }
