"""Prometheus metrics exporter for the serving engine (beyond the
reference, which has no metrics backend — SURVEY §5)."""
from __future__ import annotations

import urllib.request

from runbookai_amd.engine.engine import LLMEngine
from runbookai_amd.engine.metrics import render_metrics, serve_metrics


def test_scrape_reflects_engine_state():
    eng = LLMEngine(model="tiny", device="cpu", background=False)
    try:
        eng.generate(eng.tokenizer.encode("hello"), max_new_tokens=4)
        text = render_metrics(eng).decode()
        assert "runbook_engine_requests_total 1.0" in text
        assert "runbook_engine_decode_tokens_total" in text
        assert 'runbook_engine_kv_blocks{state="free"}' in text
        assert 'runbook_engine_kv_blocks{state="live"} 0.0' in text
        assert "runbook_engine_requests_in_flight 0.0" in text
    finally:
        eng.shutdown()


def test_http_endpoint_serves_scrapes():
    eng = LLMEngine(model="tiny", device="cpu", background=False)
    server = None
    try:
        eng.generate(eng.tokenizer.encode("hi"), max_new_tokens=2)
        server = serve_metrics(eng, port=0)          # OS-assigned port
        port = server.server_port
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/metrics", timeout=10).read().decode()
        assert "runbook_engine_requests_total 1.0" in body
        assert "runbook_engine_prefill_tokens_total" in body
    finally:
        if server is not None:
            server.shutdown()
        eng.shutdown()


def test_cli_help_lists_metrics():
    from click.testing import CliRunner

    from runbookai_amd.cli import cli

    r = CliRunner().invoke(cli, ["metrics", "serve", "--help"])
    assert r.exit_code == 0
    assert "/metrics" in r.output or "Prometheus" in r.output


class TestLatencyStats:
    def test_percentiles_after_requests(self):
        from runbookai_amd.engine.engine import LLMEngine

        eng = LLMEngine(model="tiny", device="cpu", background=False, kv_blocks=128)
        try:
            for _ in range(4):
                eng.generate([1, 2, 3], max_new_tokens=4)
            lat = eng.latency_stats()
            assert lat["samples"] == 4
            assert lat["e2e_p50_s"] > 0
            assert lat["e2e_p95_s"] >= lat["e2e_p50_s"]
            assert "ttft_p50_s" in lat
        finally:
            eng.shutdown()

    def test_empty_ring(self):
        from runbookai_amd.engine.engine import LLMEngine

        eng = LLMEngine(model="tiny", device="cpu", background=False, kv_blocks=128)
        try:
            assert eng.latency_stats() == {"samples": 0}
        finally:
            eng.shutdown()

    def test_latency_in_scrape(self):
        from runbookai_amd.engine.engine import LLMEngine
        from runbookai_amd.engine.metrics import render_metrics

        eng = LLMEngine(model="tiny", device="cpu", background=False, kv_blocks=128)
        try:
            eng.generate([1, 2], max_new_tokens=2)
            text = render_metrics(eng).decode()
            assert "runbook_engine_latency_seconds" in text
            assert 'kind="e2e"' in text
        finally:
            eng.shutdown()
