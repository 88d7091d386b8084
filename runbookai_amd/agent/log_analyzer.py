"""Log pattern analyzer: error dictionaries + hypothesis seeds.

Parity with reference src/agent/log-analyzer.ts (625 LoC): ERROR_PATTERNS
dictionary with categories (L14-186); line parsing — ISO / syslog / unix
timestamps, level, [source] extraction (L230-272); analyze_patterns with
first/last-seen tracking (L274-325); service-mention extraction — known
list, service= pairs, [source] (L327-369); time range + level counts
(L371-413); hypothesis generation from patterns (L415-430); summary
(L432-480); LLM log sampling + prompt (L482-539); full analysis + LLM
merge (L541-583); time/level filters and search (L584-621).
"""
from __future__ import annotations

import re
from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Any, Optional, Union

from .llm_parser import PROMPTS, parse_log_analysis


@dataclass
class ErrorPattern:
    name: str
    regex: re.Pattern
    severity: str
    category: str
    hypothesis: str


# Reference log-analyzer.ts:14-186 — keyed dictionary of named patterns,
# each carrying a category and a ready-made hypothesis seed.
ERROR_PATTERNS: dict[str, ErrorPattern] = {
    p.name: p
    for p in [
        ErrorPattern("oom", re.compile(r"out of memory|oom[- ]?kill|memory limit exceeded|java\.lang\.OutOfMemoryError|OutOfMemoryError|heap out of memory", re.I),
                     "critical", "resources", "A service is exhausting memory (OOM kills / heap exhaustion)"),
        ErrorPattern("timeout", re.compile(r"\btimed?[ -]?out\b|deadline exceeded|ETIMEDOUT", re.I),
                     "error", "connectivity", "Requests are timing out — downstream latency or saturation"),
        ErrorPattern("connection_refused", re.compile(r"connection refused|ECONNREFUSED|connect: connection reset", re.I),
                     "error", "connectivity", "A dependency is refusing connections (down or port closed)"),
        ErrorPattern("connection_pool", re.compile(r"connection pool (?:exhausted|timeout)|too many connections|pool is full", re.I),
                     "critical", "database", "Connection pool exhaustion against a shared dependency"),
        ErrorPattern("http_5xx", re.compile(r"\b5\d\d\b.{0,40}(?:error|status)|status[= ]5\d\d|internal server error", re.I),
                     "error", "service", "Upstream service returning 5xx errors"),
        ErrorPattern("deadlock", re.compile(r"deadlock|lock wait timeout|query timeout", re.I),
                     "critical", "database", "Database deadlocks / lock contention"),
        ErrorPattern("dns", re.compile(r"dns|name resolution|ENOTFOUND|no such host", re.I),
                     "error", "connectivity", "DNS resolution failures"),
        ErrorPattern("disk", re.compile(r"no space left on device|disk full|ENOSPC", re.I),
                     "critical", "resources", "Disk exhaustion"),
        ErrorPattern("throttle", re.compile(r"throttl|rate limit|429|TooManyRequests", re.I),
                     "warning", "service", "Rate limiting / throttling by a dependency"),
        ErrorPattern("auth", re.compile(r"unauthorized|forbidden|access denied|401|403", re.I),
                     "warning", "security", "Authentication/authorization failures (expired creds/config)"),
        ErrorPattern("ssl", re.compile(r"ssl|tls|certificate (?:expired|verify|invalid)|x509|handshake fail", re.I),
                     "error", "security", "TLS/certificate problems (expired or mismatched certs)"),
        ErrorPattern("crash", re.compile(r"segfault|segmentation fault|panic:|core dumped|fatal error|process exited|killed signal", re.I),
                     "critical", "process", "A process is crashing (segfault/panic/kill)"),
        ErrorPattern("kubernetes", re.compile(r"CrashLoopBackOff|ImagePullBackOff|evicted|liveness probe|readiness probe|OOMKilled|FailedScheduling", re.I),
                     "error", "kubernetes", "Kubernetes pod instability (restarts, evictions, probe failures)"),
    ]
}

_ISO_RE = re.compile(r"(\d{4}-\d{2}-\d{2})[T ](\d{2}:\d{2}:\d{2})(\.\d+)?(Z|[+-]\d{2}:?\d{2})?")
_SYSLOG_RE = re.compile(r"\b(Jan|Feb|Mar|Apr|May|Jun|Jul|Aug|Sep|Oct|Nov|Dec)\s+(\d{1,2})\s+(\d{2}:\d{2}:\d{2})\b")
_UNIX_RE = re.compile(r"(?<!\d)(\d{13}|\d{10})(?!\d)")
_LEVEL_RE = re.compile(r"\b(TRACE|DEBUG|INFO|WARN(?:ING)?|ERROR|FATAL|CRITICAL)\b", re.I)
_SOURCE_RE = re.compile(r"[\[<]([A-Za-z][\w.-]*)[\]>]")
_SERVICE_RE = re.compile(r"\b([a-z][a-z0-9]*(?:-[a-z0-9]+)+)\b")
_SERVICE_KV_RE = re.compile(r"\bservice[=:]\"?([\w-]+)\"?", re.I)
_MONTHS = {m: i + 1 for i, m in enumerate(
    ["Jan", "Feb", "Mar", "Apr", "May", "Jun", "Jul", "Aug", "Sep", "Oct", "Nov", "Dec"])}
_LEVEL_ORDER = ["TRACE", "DEBUG", "INFO", "WARN", "ERROR", "FATAL", "CRITICAL"]


def parse_timestamp(line: str) -> Optional[datetime]:
    """ISO-8601, syslog ('Jan 15 10:30:45') or bare unix s/ms epoch.

    Reference log-analyzer.ts:230-258.
    """
    m = _ISO_RE.search(line)
    if m:
        frac = m.group(3) or ""
        tz = m.group(4) or ""
        iso = f"{m.group(1)}T{m.group(2)}{frac}{tz.replace('Z', '+00:00')}"
        try:
            dt = datetime.fromisoformat(iso)
            return dt if dt.tzinfo else dt.replace(tzinfo=timezone.utc)
        except ValueError:
            pass
    m = _SYSLOG_RE.search(line)
    if m:
        hh, mm, ss = m.group(3).split(":")
        try:
            return datetime(datetime.now(timezone.utc).year, _MONTHS[m.group(1)],
                            int(m.group(2)), int(hh), int(mm), int(ss),
                            tzinfo=timezone.utc)
        except ValueError:  # "Jan 99 99:99:99" matches the regex shape
            return None
    m = _UNIX_RE.search(line)
    if m:
        val = int(m.group(1))
        if len(m.group(1)) == 13:
            val /= 1000.0
        try:
            return datetime.fromtimestamp(val, tz=timezone.utc)
        except (ValueError, OSError, OverflowError):
            return None
    return None


@dataclass
class ParsedLine:
    raw: str
    timestamp: Optional[datetime] = None
    level: Optional[str] = None
    source: Optional[str] = None
    message: str = ""


@dataclass
class PatternHit:
    name: str
    severity: str
    category: str
    count: int
    samples: list[str] = field(default_factory=list)
    hypothesis: str = ""
    first_seen: Optional[datetime] = None
    last_seen: Optional[datetime] = None


class LogAnalyzer:
    def __init__(self, llm: Any = None) -> None:
        self.llm = llm

    # -- parsing (reference L230-272) ----------------------------------------

    def parse_line(self, line: str) -> ParsedLine:
        ts = parse_timestamp(line)
        lm = _LEVEL_RE.search(line)
        level = lm.group(1).upper() if lm else None
        if level == "WARNING":
            level = "WARN"
        sm = _SOURCE_RE.search(line)
        source = sm.group(1) if sm and sm.group(1).upper() not in _LEVEL_ORDER else None
        msg = line
        if lm:
            msg = line[lm.end():].strip() or line
        return ParsedLine(raw=line, timestamp=ts, level=level, source=source, message=msg)

    # -- pattern analysis (reference L274-325) -------------------------------

    def analyze_patterns(self, lines: list[str], max_samples: int = 3) -> list[PatternHit]:
        hits: dict[str, PatternHit] = {}
        for line in lines:
            ts = None
            ts_done = False
            for pat in ERROR_PATTERNS.values():
                if pat.regex.search(line):
                    hit = hits.setdefault(
                        pat.name,
                        PatternHit(name=pat.name, severity=pat.severity,
                                   category=pat.category, count=0, hypothesis=pat.hypothesis),
                    )
                    hit.count += 1
                    if len(hit.samples) < max_samples:
                        hit.samples.append(line.strip()[:200])
                    if not ts_done:
                        ts, ts_done = parse_timestamp(line), True
                    if ts is not None:
                        if hit.first_seen is None or ts < hit.first_seen:
                            hit.first_seen = ts
                        if hit.last_seen is None or ts > hit.last_seen:
                            hit.last_seen = ts
        order = {"critical": 0, "error": 1, "warning": 2, "info": 3}
        return sorted(hits.values(), key=lambda h: (order.get(h.severity, 3), -h.count))

    # -- service mentions (reference L327-369) -------------------------------

    def extract_service_counts(self, lines: list[str],
                               known_services: Optional[list[str]] = None) -> dict[str, int]:
        """Counts per service from known-name mentions, service=NAME pairs,
        [source] prefixes, and dashed-name heuristics."""
        counts: dict[str, int] = {}

        def bump(name: str) -> None:
            counts[name] = counts.get(name, 0) + 1

        known = set(known_services or [])
        for line in lines:
            seen: set[str] = set()
            for svc in known:
                if svc in line:
                    seen.add(svc)
            for m in _SERVICE_KV_RE.finditer(line):
                seen.add(m.group(1))
            sm = _SOURCE_RE.search(line)
            if sm and sm.group(1).upper() not in _LEVEL_ORDER:
                seen.add(sm.group(1))
            if not known:
                for m in _SERVICE_RE.finditer(line):
                    if len(m.group(1)) >= 4:
                        seen.add(m.group(1))
            for name in seen:
                bump(name)
        return counts

    def extract_services(self, lines: list[str],
                         known_services: Optional[list[str]] = None) -> list[str]:
        counts = self.extract_service_counts(lines, known_services)
        return [s for s, _ in sorted(counts.items(), key=lambda kv: -kv[1])[:8]]

    # -- time range + level counts (reference L371-413) ----------------------

    def time_range(self, lines: list[str]) -> Optional[tuple[datetime, datetime]]:
        stamps = [ts for ts in (parse_timestamp(l) for l in lines) if ts is not None]
        if not stamps:
            return None
        return min(stamps), max(stamps)

    def count_by_level(self, lines: list[str]) -> dict[str, int]:
        errors = warnings = 0
        for line in lines:
            lvl = self.parse_line(line).level
            if lvl in ("ERROR", "FATAL", "CRITICAL"):
                errors += 1
            elif lvl == "WARN":
                warnings += 1
        return {"errors": errors, "warnings": warnings}

    # -- hypotheses from patterns (reference L415-430) -----------------------

    def hypotheses_from_patterns(self, hits: list[PatternHit]) -> list[str]:
        out: list[str] = []
        for h in hits:
            if h.severity in ("critical", "error") and h.hypothesis not in out:
                out.append(h.hypothesis)
        return out[:5]

    # -- summary (reference L432-480) ----------------------------------------

    def summarize(self, total: int, counts: dict[str, int], hits: list[PatternHit],
                  services: dict[str, int],
                  rng: Optional[tuple[datetime, datetime]]) -> str:
        parts = [f"Analyzed {total} log lines: {counts['errors']} errors, "
                 f"{counts['warnings']} warnings."]
        if rng:
            parts.append(f"Time range {rng[0].isoformat()} → {rng[1].isoformat()}.")
        if hits:
            tops = ", ".join(f"{h.name} x{h.count} ({h.severity})" for h in hits[:5])
            parts.append(f"Top patterns: {tops}.")
        if services:
            svc = ", ".join(s for s, _ in sorted(services.items(), key=lambda kv: -kv[1])[:5])
            parts.append(f"Services mentioned: {svc}.")
        return " ".join(parts)

    # -- LLM formatting (reference L482-539) ---------------------------------

    def format_logs_for_llm(self, lines: list[str], max_lines: int = 200) -> str:
        if len(lines) <= max_lines:
            return "\n".join(lines)
        head = max_lines * 2 // 3
        tail = max_lines - head
        omitted = len(lines) - head - tail
        return "\n".join(lines[:head] + [f"... {omitted} lines omitted ..."] + lines[-tail:])

    def analysis_prompt(self, lines: list[str], hits: list[PatternHit],
                        max_lines: int = 80) -> str:
        return PROMPTS["analyzeLogs"].format(
            logs=self.format_logs_for_llm(lines, max_lines),
            patterns="\n".join(f"{h.name} x{h.count} ({h.severity})" for h in hits),
        )

    # -- filters + search (reference L584-621) -------------------------------

    def filter_by_time(self, lines: list[str], start: datetime, end: datetime) -> list[str]:
        """Keeps lines inside [start, end] and lines with no parseable
        timestamp (reference keeps untimestamped lines)."""
        out = []
        for line in lines:
            ts = parse_timestamp(line)
            if ts is None or start <= ts <= end:
                out.append(line)
        return out

    def filter_by_level(self, lines: list[str], min_level: str = "WARN",
                        keep_unleveled: bool = True) -> list[str]:
        """Keeps lines at/above min_level; lines with no detectable level are
        kept by default (reference semantics)."""
        try:
            threshold = _LEVEL_ORDER.index(min_level.upper())
        except ValueError:
            threshold = 3
        out = []
        for line in lines:
            lvl = self.parse_line(line).level
            if lvl is None:
                if keep_unleveled:
                    out.append(line)
            elif _LEVEL_ORDER.index(lvl) >= threshold:
                out.append(line)
        return out

    def search(self, lines: list[str], query: Union[str, re.Pattern]) -> list[str]:
        if isinstance(query, str):
            q = query.lower()
            return [l for l in lines if q in l.lower()]
        return [l for l in lines if query.search(l)]

    # -- full analysis with optional LLM merge (reference L541-583) ----------

    def analyze(self, lines: list[str],
                known_services: Optional[list[str]] = None) -> dict[str, Any]:
        hits = self.analyze_patterns(lines)
        svc_counts = self.extract_service_counts(lines, known_services)
        counts = self.count_by_level(lines)
        rng = self.time_range(lines)
        services = [s for s, _ in sorted(svc_counts.items(), key=lambda kv: -kv[1])[:8]]
        result: dict[str, Any] = {
            "totalLines": len(lines),
            "errorCount": counts["errors"],
            "warningCount": counts["warnings"],
            "summary": self.summarize(len(lines), counts, hits, svc_counts, rng),
            "patterns": [
                {"pattern": h.name, "count": h.count, "severity": h.severity,
                 "category": h.category,
                 "firstSeen": h.first_seen.isoformat() if h.first_seen else None,
                 "lastSeen": h.last_seen.isoformat() if h.last_seen else None,
                 "sample": h.samples[0] if h.samples else ""}
                for h in hits
            ],
            "services": services,
            "serviceMentions": svc_counts,
            "timeRange": {"start": rng[0].isoformat(), "end": rng[1].isoformat()} if rng else None,
            "suggestedHypotheses": self.hypotheses_from_patterns(hits),
        }
        if self.llm is not None and lines:
            try:
                llm_result = parse_log_analysis(self.llm.complete(self.analysis_prompt(lines, hits)))
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
