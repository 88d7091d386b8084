"""Log pattern analyzer: error dictionaries + hypothesis seeds.

Parity with reference src/agent/log-analyzer.ts (625 LoC): ERROR_PATTERNS
dictionary (oom, timeout, conn-refused, 5xx, deadlock, ...) (L14-186);
line parsing + timestamp extraction (L230-272); analyze_patterns (L274-325),
service-mention extraction (L327-369), hypothesis generation from patterns
(L415-430), LLM merge (L541-583), time/level filters (L584-621).
"""
from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Any, Optional

from .llm_parser import PROMPTS, parse_log_analysis


@dataclass
class ErrorPattern:
    name: str
    regex: re.Pattern
    severity: str
    hypothesis: str


# Reference log-analyzer.ts:14-186.
ERROR_PATTERNS: list[ErrorPattern] = [
    ErrorPattern("oom", re.compile(r"out of memory|oom[- ]?kill|memory limit exceeded|java\.lang\.OutOfMemoryError", re.I),
                 "critical", "A service is exhausting memory (OOM kills / heap exhaustion)"),
    ErrorPattern("timeout", re.compile(r"\btime[d ]?out\b|deadline exceeded|ETIMEDOUT", re.I),
                 "error", "Requests are timing out — downstream latency or saturation"),
    ErrorPattern("connection_refused", re.compile(r"connection refused|ECONNREFUSED|connect: connection reset", re.I),
                 "error", "A dependency is refusing connections (down or port closed)"),
    ErrorPattern("connection_pool", re.compile(r"connection pool (?:exhausted|timeout)|too many connections|pool is full", re.I),
                 "critical", "Connection pool exhaustion against a shared dependency"),
    ErrorPattern("http_5xx", re.compile(r"\b5\d\d\b.{0,40}(?:error|status)|status[= ]5\d\d|internal server error", re.I),
                 "error", "Upstream service returning 5xx errors"),
    ErrorPattern("deadlock", re.compile(r"deadlock|lock wait timeout", re.I),
                 "critical", "Database deadlocks / lock contention"),
    ErrorPattern("dns", re.compile(r"dns|name resolution|ENOTFOUND|no such host", re.I),
                 "error", "DNS resolution failures"),
    ErrorPattern("disk", re.compile(r"no space left on device|disk full|ENOSPC", re.I),
                 "critical", "Disk exhaustion"),
    ErrorPattern("throttle", re.compile(r"throttl|rate limit|429|TooManyRequests", re.I),
                 "warning", "Rate limiting / throttling by a dependency"),
    ErrorPattern("auth", re.compile(r"unauthorized|forbidden|access denied|401|403", re.I),
                 "warning", "Authentication/authorization failures (expired creds/config)"),
]

_TS_RES = [
    re.compile(r"\d{4}-\d{2}-\d{2}[T ]\d{2}:\d{2}:\d{2}(?:\.\d+)?(?:Z|[+-]\d{2}:?\d{2})?"),
    re.compile(r"\d{2}:\d{2}:\d{2}(?:\.\d+)?"),
]
_LEVEL_RE = re.compile(r"\b(TRACE|DEBUG|INFO|WARN(?:ING)?|ERROR|FATAL|CRITICAL)\b", re.I)
_SERVICE_RE = re.compile(r"\b([a-z][a-z0-9]*(?:-[a-z0-9]+)+)\b")


@dataclass
class ParsedLine:
    raw: str
    timestamp: Optional[str] = None
    level: Optional[str] = None


@dataclass
class PatternHit:
    name: str
    severity: str
    count: int
    samples: list[str] = field(default_factory=list)
    hypothesis: str = ""


class LogAnalyzer:
    def __init__(self, llm: Any = None) -> None:
        self.llm = llm

    # -- parsing (reference L230-272) ----------------------------------------

    def parse_line(self, line: str) -> ParsedLine:
        ts = None
        for rx in _TS_RES:
            m = rx.search(line)
            if m:
                ts = m.group(0)
                break
        lm = _LEVEL_RE.search(line)
        level = lm.group(1).upper() if lm else None
        if level == "WARNING":
            level = "WARN"
        return ParsedLine(raw=line, timestamp=ts, level=level)

    # -- pattern analysis (reference L274-325) -------------------------------

    def analyze_patterns(self, lines: list[str], max_samples: int = 3) -> list[PatternHit]:
        hits: dict[str, PatternHit] = {}
        for line in lines:
            for pat in ERROR_PATTERNS:
                if pat.regex.search(line):
                    hit = hits.setdefault(
                        pat.name,
                        PatternHit(name=pat.name, severity=pat.severity, count=0, hypothesis=pat.hypothesis),
                    )
                    hit.count += 1
                    if len(hit.samples) < max_samples:
                        hit.samples.append(line.strip()[:200])
        order = {"critical": 0, "error": 1, "warning": 2, "info": 3}
        return sorted(hits.values(), key=lambda h: (order.get(h.severity, 3), -h.count))

    # -- service mentions (reference L327-369) -------------------------------

    def extract_services(self, lines: list[str]) -> list[str]:
        counts: dict[str, int] = {}
        for line in lines:
            for m in _SERVICE_RE.finditer(line):
                name = m.group(1)
                if len(name) >= 4:
                    counts[name] = counts.get(name, 0) + 1
        return [s for s, _ in sorted(counts.items(), key=lambda kv: -kv[1])[:8]]

    # -- hypotheses from patterns (reference L415-430) -----------------------

    def hypotheses_from_patterns(self, hits: list[PatternHit]) -> list[str]:
        return [h.hypothesis for h in hits if h.severity in ("critical", "error")][:5]

    # -- filters (reference L584-621) ----------------------------------------

    def filter_by_level(self, lines: list[str], min_level: str = "WARN") -> list[str]:
        order = ["TRACE", "DEBUG", "INFO", "WARN", "ERROR", "FATAL", "CRITICAL"]
        try:
            threshold = order.index(min_level.upper())
        except ValueError:
            threshold = 3
        out = []
        for line in lines:
            parsed = self.parse_line(line)
            if parsed.level and order.index(parsed.level) >= threshold:
                out.append(line)
        return out

    # -- full analysis with optional LLM merge (reference L541-583) ----------

    def analyze(self, lines: list[str]) -> dict[str, Any]:
        hits = self.analyze_patterns(lines)
        services = self.extract_services(lines)
        result: dict[str, Any] = {
            "summary": f"{len(lines)} lines; {sum(h.count for h in hits)} error-pattern matches "
                       f"across {len(hits)} patterns",
            "patterns": [
                {"pattern": h.name, "count": h.count, "severity": h.severity, "sample": h.samples[0] if h.samples else ""}
                for h in hits
            ],
            "services": services,
            "suggestedHypotheses": self.hypotheses_from_patterns(hits),
        }
        if self.llm is not None and lines:
            try:
                prompt = PROMPTS["analyzeLogs"].format(
                    logs="\n".join(lines[:80]),
                    patterns="\n".join(f"{h.name} x{h.count} ({h.severity})" for h in hits),
                )
                llm_result = parse_log_analysis(self.llm.complete(prompt))
                if llm_result.get("summary"):
                    result["summary"] = llm_result["summary"]
                known = {p["pattern"] for p in result["patterns"]}
                for p in llm_result.get("patterns", []):
                    if p.get("pattern") and p["pattern"] not in known:
                        result["patterns"].append(p)
                for s in llm_result.get("services", []):
                    if s not in result["services"]:
                        result["services"].append(s)
            except Exception:  # noqa: BLE001 — LLM merge is best-effort
                pass
        return result
