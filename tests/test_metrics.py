"""Observability subsystem (utils/metrics.py) — SURVEY.md §5.1/§5.5.

The reference has no metrics at all; these tests pin the build's
Prometheus-text rendering, histogram bucket semantics, and the CPU path
of the HIP-event stage timer."""

import time

from mlx_sharding_amd.utils.metrics import (Counter, Histogram, Registry,
                                            StageTimer, serving_metrics)


def test_counter_render():
    c = Counter("x_total", "help text")
    c.inc()
    c.inc(2.5)
    assert c.value == 3.5
    out = c.render()
    assert "# TYPE x_total counter" in out
    assert "x_total 3.5" in out


def test_histogram_buckets_cumulative():
    h = Histogram("lat_ms", "latency", buckets=(10, 100, 1000))
    for v in (5, 50, 50, 500, 5000):
        h.observe(v)
    out = h.render()
    # cumulative: le=10 -> 1, le=100 -> 3, le=1000 -> 4, +Inf -> 5
    assert 'lat_ms_bucket{le="10"} 1' in out
    assert 'lat_ms_bucket{le="100"} 3' in out
    assert 'lat_ms_bucket{le="1000"} 4' in out
    assert 'lat_ms_bucket{le="+Inf"} 5' in out
    assert "lat_ms_count 5" in out
    assert h.sum == 5605


def test_histogram_boundary_is_le():
    h = Histogram("b", "", buckets=(10, 100))
    h.observe(10)  # exactly on a boundary counts in that bucket (le)
    assert 'b_bucket{le="10"} 1' in h.render()


def test_registry_idempotent_and_render():
    r = Registry()
    c1 = r.counter("a_total", "a")
    c2 = r.counter("a_total")
    assert c1 is c2
    r.histogram("h_ms")
    out = r.render()
    assert "a_total 0" in out and "h_ms_count 0" in out


def test_serving_metrics_set():
    r = Registry()
    m = serving_metrics(r)
    m["requests"].inc()
    m["ttft_ms"].observe(42.0)
    out = r.render()
    assert "mlxs_requests_total 1" in out
    assert "mlxs_ttft_ms_count 1" in out


def test_stage_timer_cpu_path():
    t = StageTimer()
    for _ in range(3):
        with t.measure("cpu"):
            time.sleep(0.002)
    s = t.summary()
    assert s["count"] == 3
    assert s["mean_ms"] >= 1.5
    assert s["p50_ms"] <= s["p95_ms"] + 1e-9
    assert t.total_calls == 3


def test_stage_timer_empty_summary():
    assert StageTimer().summary()["count"] == 0
