"""Prometheus metrics for the serving engine.

The reference has no metrics backend (SURVEY.md §5 "no structured
logger, no metrics backend") — production serving on MI355X wants one:
`runbook metrics serve` (cli.py) exposes the engine's counters plus KV
pool occupancy on a /metrics HTTP endpoint via prometheus_client.

Collection is pull-based and reads the engine's stats dict and KV pool
sizes directly — no instrumentation on the hot path.
"""
from __future__ import annotations

from typing import Any, Optional


class EngineCollector:
    """Custom collector: samples engine state at scrape time."""

    def __init__(self, engine: Any) -> None:
        self.engine = engine

    def collect(self):  # pragma: no cover - exercised via registry.collect()
        from prometheus_client.core import CounterMetricFamily, GaugeMetricFamily

        eng = self.engine
        s = eng.stats

        def counter(name: str, doc: str, value: float):
            c = CounterMetricFamily(f"runbook_engine_{name}", doc)
            c.add_metric([], float(value))
            return c

        yield counter("requests_total", "LLM requests submitted", s.get("requests", 0))
        yield counter("steps_total", "engine iterations", s.get("steps", 0))
        yield counter("decode_tokens_total", "tokens decoded", s.get("decode_tokens", 0))
        yield counter("prefill_tokens_total", "tokens prefilled", s.get("prefill_tokens", 0))
        yield counter("chunk_tokens_total", "tokens through the chunk path",
                      s.get("chunk_tokens", 0))
        yield counter("cached_prefix_tokens_total",
                      "prompt tokens served from the prefix pool",
                      s.get("cached_prefix_tokens", 0))
        yield counter("step_errors_total", "isolated step failures",
                      s.get("step_errors", 0))
        yield counter("decode_seconds_total", "decode+chunk wall seconds",
                      s.get("decode_time", 0.0))
        yield counter("prefill_seconds_total", "prefill wall seconds",
                      s.get("prefill_time", 0.0))

        g = GaugeMetricFamily("runbook_engine_requests_in_flight",
                              "waiting + running requests")
        with eng._lock:
            g.add_metric([], float(len(eng.waiting) + len(eng.running)))
        yield g

        lat = eng.latency_stats() if hasattr(eng, "latency_stats") else {}
        if lat.get("samples"):
            lg = GaugeMetricFamily("runbook_engine_latency_seconds",
                                   "request latency percentiles (recent window)",
                                   labels=["kind", "quantile"])
            for key, (kind, q) in (("ttft_p50_s", ("ttft", "0.5")),
                                   ("ttft_p95_s", ("ttft", "0.95")),
                                   ("e2e_p50_s", ("e2e", "0.5")),
                                   ("e2e_p95_s", ("e2e", "0.95"))):
                if key in lat:
                    lg.add_metric([kind, q], float(lat[key]))
            yield lg

        kv = eng.model.kv
        pool = GaugeMetricFamily("runbook_engine_kv_blocks", "KV pool blocks",
                                 labels=["state"])
        live = sum(len(t) for t in kv.block_tables.values())
        pool.add_metric(["live"], float(live))
        pool.add_metric(["free"], float(len(kv._free)))
        pool.add_metric(["retired_prefix"], float(len(kv.pool_lru)))
        yield pool


def make_registry(engine: Any):
    """A fresh CollectorRegistry with the engine collector attached."""
    from prometheus_client import CollectorRegistry

    reg = CollectorRegistry()
    reg.register(EngineCollector(engine))
    return reg


def render_metrics(engine: Any) -> bytes:
    """One text-format scrape (used by tests and the CLI endpoint)."""
    from prometheus_client import generate_latest

    return generate_latest(make_registry(engine))


def serve_metrics(engine: Any, port: int = 9464,
                  addr: str = "127.0.0.1") -> Optional[object]:
    """Start the /metrics HTTP endpoint; returns the WSGI server handle
    (its .server_port is useful with port=0)."""
    from prometheus_client import start_http_server

    server, _thread = start_http_server(port, addr=addr,
                                        registry=make_registry(engine))
    return server
