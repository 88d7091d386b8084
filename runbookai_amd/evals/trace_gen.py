"""Synthetic investigation-trace generation for policy training.

An ORACLE client answers the real InvestigationOrchestrator's prompts
correctly (deterministically, from the case's expected root cause), while
every (prompt, response) pair is recorded. The recorded pairs are exactly
the serving distribution: the same prompt builders (agent/llm_parser
PROMPTS), the same simulated telemetry (providers/simulation), the same
knowledge-base snippets — so a model trained on them serves through
LLMEngine(checkpoint=...) with no distribution shift.

Case variety comes from ARCHETYPES x sampled service names/symptoms: each
generated case builds its scenario with SimScenario.from_fixture, whose
telemetry embeds the expected keywords — the learnable signal is
"telemetry content -> root cause", which transfers to the held-out eval
fixtures built the same way.

The reference has no counterpart (its investigation quality is bought
from hosted frontier models); this module is what makes BASELINE.md's
accuracy axis measurable on local hardware.
"""
from __future__ import annotations

import random
from dataclasses import dataclass, field
from typing import Any, Optional

from ..agent.llm_parser import split_schema_tag

# (root-cause phrase template, keyword pool, symptom pool, service pool)
ARCHETYPES: list[dict[str, Any]] = [
    {"name": "conn-pool-exhaustion",
     "keywords": ["connection", "pool", "exhausted"],
     "extra": [["redis"], ["postgres"], ["mysql"], ["database"]],
     "symptoms": ["latency spike", "timeouts acquiring connections",
                  "p99 latency over SLO"],
     "services": [["checkout-api", "cart-service"], ["orders-api", "billing"],
                  ["session-service", "auth-api"]]},
    {"name": "gateway-5xx",
     "keywords": ["upstream", "timeout", "5xx"],
     "extra": [["gateway"], ["nginx"], ["envoy"]],
     "symptoms": ["HTTP 502 rate climbing", "upstream connect errors",
                  "gateway error budget burn"],
     "services": [["api-gateway", "user-service"], ["edge-proxy", "search-api"]]},
    {"name": "disk-pressure",
     "keywords": ["disk", "space", "retention"],
     "extra": [["kafka"], ["broker"], ["full"]],
     "symptoms": ["producer latency rising", "consumer lag growing",
                  "broker disk usage over 95%"],
     "services": [["kafka-broker", "order-events"], ["event-bus", "analytics-sink"]]},
    {"name": "cert-expiry",
     "keywords": ["certificate", "expired", "tls"],
     "extra": [["handshake"], ["x509"]],
     "symptoms": ["TLS handshake failures", "unknown certificate errors",
                  "payment callbacks failing"],
     "services": [["payments-api", "bank-connector"], ["webhook-dispatcher", "partner-api"]]},
    {"name": "oom-crashloop",
     "keywords": ["memory", "oom", "killed"],
     "extra": [["container"], ["heap"], ["limit"]],
     "symptoms": ["pods restarting", "OOMKilled events", "rss near limit"],
     "services": [["recommendation-svc", "feed-api"], ["ml-inference", "ranker"]]},
    {"name": "deploy-regression",
     "keywords": ["deployment", "regression", "rollback"],
     "extra": [["release"], ["canary"], ["bad config"]],
     "symptoms": ["error rate up after deploy", "new version crashlooping",
                  "canary failing health checks"],
     "services": [["catalog-api", "inventory"], ["pricing-svc", "promo-engine"]]},
    {"name": "db-cpu-saturation",
     "keywords": ["cpu", "saturation", "slow queries"],
     "extra": [["database"], ["index"], ["full scan"]],
     "symptoms": ["query latency growing", "db cpu pegged at 100%",
                  "application timeouts"],
     "services": [["orders-db", "orders-api"], ["reporting-db", "dashboard-api"]]},
    {"name": "dns-resolution",
     "keywords": ["dns", "resolution", "failing"],
     "extra": [["ndots"], ["coredns"], ["nameserver"]],
     "symptoms": ["intermittent name resolution errors", "SERVFAIL responses",
                  "connection setup failures"],
     "services": [["service-mesh", "discovery"], ["coredns", "cluster-api"]]},
    {"name": "rate-limit",
     "keywords": ["rate", "limit", "throttled"],
     "extra": [["quota"], ["429"], ["burst"]],
     "symptoms": ["HTTP 429 responses", "third-party quota exceeded",
                  "request queue growing"],
     "services": [["notification-svc", "email-provider"], ["sync-worker", "crm-api"]]},
    {"name": "cache-stampede",
     "keywords": ["cache", "miss", "stampede"],
     "extra": [["ttl"], ["thundering herd"], ["eviction"]],
     "symptoms": ["cache hit rate collapsed", "origin load spiking",
                  "latency p95 rising"],
     "services": [["content-cache", "cms-api"], ["edge-cache", "media-service"]]},
    {"name": "queue-backlog",
     "keywords": ["queue", "backlog", "consumer"],
     "extra": [["lag"], ["stuck"], ["dead letter"]],
     "symptoms": ["message age growing", "consumers stalled",
                  "processing delayed"],
     "services": [["job-queue", "worker-pool"], ["task-broker", "batch-runner"]]},
    {"name": "gateway-deploy",
     "keywords": ["deploy", "gateway", "timeout", "upstream"],
     "extra": [["5xx"], ["regression"], ["rollback"]],
     "symptoms": ["5xx spike minutes after a deploy", "upstream timeouts after release",
                  "gateway errors following rollout"],
     "services": [["api-gateway", "user-service"], ["edge-router", "profile-api"]]},
    {"name": "network-partition",
     "keywords": ["network", "partition", "unreachable"],
     "extra": [["packet loss"], ["link"], ["az"]],
     "symptoms": ["cross-zone call failures", "elevated packet loss",
                  "health checks flapping"],
     "services": [["zone-b-fleet", "replicator"], ["cluster-peer", "gossip-mesh"]]},
    {"name": "egress-port-exhaustion",
     "keywords": ["nat", "port", "exhausted", "egress"],
     "extra": [["snat"], ["keep-alive"], ["outbound"]],
     "symptoms": ["external calls timing out", "in-VPC traffic healthy",
                  "port allocation errors on the NAT gateway"],
     "services": [["payment-service", "fraud-scorer"],
                  ["webhook-sender", "partner-sync"]]},
    {"name": "clock-skew",
     "keywords": ["clock", "skew", "offset", "time"],
     "extra": [["chronyd"], ["ntp"], ["iat"]],
     "symptoms": ["token validation failures", "401 rate climbing",
                  "signatures rejected as not yet valid"],
     "services": [["auth-service", "session-api"], ["token-issuer", "api-edge"]]},
    {"name": "memory-leak-gradual",
     "keywords": ["memory", "leak", "growing", "rss"],
     "extra": [["heap"], ["unbounded cache"], ["native"]],
     "symptoms": ["rss climbing for hours", "gc pause times rising",
                  "restarts temporarily fix latency"],
     "services": [["stream-processor", "metrics-ingest"],
                  ["graph-api", "cache-warmer"]]},
]

NATURAL_TELEMETRY: dict[str, dict] = {
 "egress-port-exhaustion": {
  "logs": [
   "POST https://api.external.example/v2/send: dial tcp: connect: connection timed out",
   "outbound call failed; retried 3 times ({extra0} ports at ceiling)",
   "in-VPC dependencies responding normally; only egress affected",
   "ErrorPortAllocation reported by the NAT gateway"
  ],
  "alarm": "NAT gateway port allocation errors > 0 affecting {svc}",
  "metric": "{svc}.external_error_rate"
 },
 "clock-skew": {
  "logs": [
   "jwt validation failed: token used before issued (iat in the future)",
   "nbf check failed rejecting request from {svc}",
   "{extra0} unit inactive on node; system clock offset +40s and drifting",
   "tokens minted on the other node pool validate fine"
  ],
  "alarm": "node clock offset over 1s affecting {svc}",
  "metric": "{svc}.jwt_rejections"
 },
 "memory-leak-gradual": {
  "logs": [
   "rss grew 3.1GB over 6h with flat traffic on {svc}",
   "gc: full collection took 2100ms, freed 1% of heap",
   "{extra0} retained size growing monotonically between deploys",
   "container approaching memory limit; no OOM kill yet"
  ],
  "alarm": "memory usage slope positive for 6h on {svc}",
  "metric": "{svc}.rss_bytes"
 },
 "conn-pool-exhaustion": {
  "logs": [
   "could not acquire connection from pool after 5000ms ({extra0} pool size 50, in use 50)",
   "{extra0}: max clients reached, rejecting new connections",
   "timeout waiting for a free connection slot; pool utilization at 100%",
   "upstream call aborted: connection pool saturated"
  ],
  "alarm": "connection wait time p99 over threshold on {svc}",
  "metric": "{svc}.pool_wait_ms"
 },
 "gateway-5xx": {
  "logs": [
   "upstream unavailable: {svc1} timed out after 10s (gateway timeout)",
   "504 gateway timeout routing /v1/requests",
   "proxy error: no healthy upstream endpoints for {svc1}",
   "5xx rate over 5% across listener group"
  ],
  "alarm": "5xx rate > 5% for 10 minutes on {svc}",
  "metric": "{svc}.5xx_rate"
 },
 "disk-pressure": {
  "logs": [
   "disk usage on /var/lib/{extra0} at 97%, segments cannot be rolled",
   "No space left on device while appending to partition log",
   "retention enforcement behind: oldest segments not deleted",
   "producer request timed out; broker flush stalled on full volume"
  ],
  "alarm": "disk usage above 95% on {svc}",
  "metric": "{svc}.disk_used_pct"
 },
 "cert-expiry": {
  "logs": [
   "x509: certificate has expired or is not yet valid (notAfter in the past)",
   "TLS handshake error from peer: bad certificate",
   "ssl verification failed calling {svc1}: certificate expired",
   "remote rejected connection during handshake"
  ],
  "alarm": "TLS handshake failure rate rising on {svc}",
  "metric": "{svc}.tls_handshake_errors"
 },
 "oom-crashloop": {
  "logs": [
   "container killed: OOMKilled (memory limit 512Mi exceeded)",
   "java.lang.OutOfMemoryError: heap space exhausted",
   "pod restarted 7 times in 10m; last state: OOMKilled",
   "rss climbed to limit before kill signal"
  ],
  "alarm": "container restarts exceed threshold on {svc}",
  "metric": "{svc}.memory_rss_bytes"
 },
 "deploy-regression": {
  "logs": [
   "error rate tripled within 2m of rollout v{ver} on {svc}",
   "new release failing readiness probe; previous version was healthy",
   "canary analysis failed: comparison against baseline exceeded error budget",
   "panic in handler introduced by latest deploy"
  ],
  "alarm": "error budget burn after deployment on {svc}",
  "metric": "{svc}.error_rate"
 },
 "db-cpu-saturation": {
  "logs": [
   "slow query log (11.2s): SELECT ... Seq Scan on large table (index missing)",
   "database CPU pegged at 100%; active backends waiting on CPU",
   "query planner falling back to sequential scan after index drop",
   "statement timeout reached for 14% of queries; cpu saturation sustained"
  ],
  "alarm": "database CPU utilization sustained above 95% on {svc}",
  "metric": "{svc}.cpu_util"
 },
 "dns-resolution": {
  "logs": [
   "lookup {svc1} on 10.0.0.2:53: server misbehaving (SERVFAIL)",
   "dial tcp: lookup failed: no such host (intermittent)",
   "coredns: upstream nameserver timeout, retrying",
   "resolution latency spiking; ndots expansion generating extra queries"
  ],
  "alarm": "DNS error rate above threshold on {svc}",
  "metric": "{svc}.dns_errors"
 },
 "rate-limit": {
  "logs": [
   "HTTP 429 Too Many Requests from provider API (quota exceeded)",
   "request rejected: rate limit bucket empty, retry after 30s",
   "burst above contracted quota; requests being throttled",
   "backoff engaged after repeated 429 responses"
  ],
  "alarm": "throttled request rate rising on {svc}",
  "metric": "{svc}.throttled_requests"
 },
 "cache-stampede": {
  "logs": [
   "cache hit ratio fell from 98% to 11% after key expiry wave",
   "origin overwhelmed: concurrent regeneration of the same hot key",
   "TTL expiry storm: thousands of misses for identical entries",
   "eviction pressure high; working set exceeds cache capacity"
  ],
  "alarm": "cache hit ratio below threshold on {svc}",
  "metric": "{svc}.cache_hit_ratio"
 },
 "queue-backlog": {
  "logs": [
   "queue depth growing: 1.2M messages, oldest 42m",
   "consumer group stalled; no acknowledgements in 10m",
   "dead letter queue receiving poison messages repeatedly",
   "processing lag exceeds SLA for downstream jobs"
  ],
  "alarm": "message age above threshold on {svc}",
  "metric": "{svc}.queue_depth"
 },
 "gateway-deploy": {
  "logs": [
   "upstream unavailable: {svc1} timed out after 10s (gateway timeout)",
   "panic: nil pointer dereference in handler v{ver} (deployed minutes ago)",
   "504 gateway timeout routing /v1/users",
   "5xx rate jumped from 0.1% to 9% right after the {svc1} rollout"
  ],
  "alarm": "5xx rate > 5% for 10 minutes on {svc}",
  "metric": "{svc}.5xx_rate"
 },
 "network-partition": {
  "logs": [
   "peer unreachable: i/o timeout dialing 10.2.0.0/16 endpoints",
   "packet loss to zone-b above 30% on inter-az link",
   "gossip membership flapping; suspect marks rising",
   "replication halted: cannot reach quorum across zones"
  ],
  "alarm": "cross-zone connectivity degraded for {svc}",
  "metric": "{svc}.packet_loss_pct"
 }
}


_SVC_HEADS = ["auth", "user", "cart", "order", "billing", "search", "media",
              "sync", "edge", "feed", "profile", "payment", "invoice", "ship",
              "geo", "notify", "ledger", "catalog", "session", "ingest"]
_SVC_TAILS = ["api", "svc", "service", "worker", "gateway", "db", "cache",
              "broker", "proxy", "runner", "engine", "store"]


def _svc_name(rng: random.Random) -> str:
    return f"{rng.choice(_SVC_HEADS)}-{rng.choice(_SVC_TAILS)}"


# open-vocabulary keyword pool: teaches the policy to EXTRACT the salient
# failure keywords from telemetry (pure copying) instead of classifying
# into a closed archetype set — what transfers to failure modes no
# archetype covers (measured: the closed-set policy scored 0.30 on the
# converter-generated suites; see docs/ACCURACY.md)
_GENERIC_KW = [
    "cpu", "stress", "hog", "leak", "memory", "heap", "swap", "gc", "pause",
    "shard", "relocation", "storm", "rebalance", "failover", "split", "brain",
    "quorum", "checksum", "corruption", "overflow", "underrun", "socket",
    "descriptor", "exhaustion", "latency", "jitter", "saturation", "spill",
    "eviction", "fragmentation", "contention", "deadlock", "livelock",
    "starvation", "thrashing", "backpressure", "churn", "flapping", "drift",
    "skew", "expiry", "rotation", "revocation", "mismatch", "regression",
    "rollback", "throttle", "ratelimit", "timeout", "refused", "reset",
    "partition", "blackhole", "loss", "duplication", "reorder", "stale",
    "inconsistency", "divergence", "lag", "backlog", "overload",
]

_GENERIC_LOG_TMPL = [
    "{phrase} detected on {svc}",
    "alert: {svc} reporting {phrase} for 10m",
    "{svc}: repeated {phrase} events in the error log",
    "health probe failing on {svc}: {phrase}",
    "{phrase} observed; {svc1} downstream requests degrading",
    "incident signature matches {phrase} on {svc}",
]


def gen_generic_case(rng: random.Random, idx: int) -> dict[str, Any]:
    """Open-vocabulary case: 3-4 keywords sampled from the wide pool."""
    kws = rng.sample(_GENERIC_KW, rng.randint(3, 4))
    services = [_svc_name(rng) for _ in range(2)]
    while services[1] == services[0]:
        services[1] = _svc_name(rng)
    incident = f"PD-GEN-{idx:05d}"
    return {
        "id": f"generic-{idx}",
        "_arch": "_generic",
        "incidentId": incident,
        "query": (f"Investigate incident {incident}: {services[0]} "
                  f"{kws[0]} {kws[1]} reported"),
        "context": "",
        "expected": {
            "rootCauseKeywords": kws,
            "affectedServices": services,
            "minimumConfidence": "medium",
        },
        "execute": {"maxIterations": 4},
    }


def gen_case(rng: random.Random, idx: int) -> dict[str, Any]:
    """One fixture-style case dict drawn from the archetype pools. 70% of
    cases use freshly combined service names — the policy cannot memorize
    an archetype->service mapping and must COPY names from the prompt.
    30% of cases are OPEN-VOCABULARY (gen_generic_case)."""
    if rng.random() < 0.30:
        return gen_generic_case(rng, idx)
    arch = rng.choice(ARCHETYPES)
    if rng.random() < 0.7:
        services = [_svc_name(rng) for _ in range(2)]
        while services[1] == services[0]:
            services[1] = _svc_name(rng)
    else:
        services = list(rng.choice(arch["services"]))
    extra = rng.choice(arch["extra"])
    keywords = list(arch["keywords"]) + extra
    symptom = rng.choice(arch["symptoms"])
    incident = f"PD-GEN-{idx:05d}"
    return {
        "id": f"{arch['name']}-{idx}",
        "_arch": arch["name"],
        "incidentId": incident,
        "query": (f"Investigate incident {incident}: {services[0]} "
                  f"{symptom}"),
        "context": f"{services[0]}: {' '.join(keywords)} detected",
        "expected": {
            "rootCauseKeywords": keywords,
            "affectedServices": services,
            "minimumConfidence": "medium",
        },
        "execute": {"maxIterations": 4},
    }


def _build_generic_scenario(case: dict[str, Any], rng: random.Random):
    from ..providers.simulation import SimScenario

    svc = case["expected"]["affectedServices"]
    kw = case["expected"]["rootCauseKeywords"]
    phrase = " ".join(kw)
    s = SimScenario(name=case["id"])
    s.incident = {"id": case["incidentId"], "title": case["query"],
                  "status": "triggered", "urgency": "high",
                  "service": svc[0], "createdAt": "2026-02-12T10:00:00Z"}
    s.services = [{"name": x, "status": "degraded" if i < 1 else "healthy",
                   "type": "ecs"} for i, x in enumerate(svc)]
    fmt = {"svc": svc[0], "svc1": svc[-1], "phrase": phrase}
    s.alarms = [{"name": f"{svc[0]}-alert", "state": "ALARM",
                 "reason": rng.choice(_GENERIC_LOG_TMPL).format(**fmt),
                 "service": svc[0]}]
    msgs = rng.sample(_GENERIC_LOG_TMPL, rng.randint(2, 3))
    for i, msg in enumerate(msgs):
        s.log_events.append({
            "timestamp": f"2026-02-12T10:{i:02d}:{(i * 17) % 60:02d}Z",
            "service": rng.choice(svc), "level": "ERROR",
            "message": msg.format(**fmt)})
    base = rng.uniform(30, 200)
    s.metrics = {f"{svc[0]}.error_rate":
                 [round(base * (1 + 0.5 * i), 1) for i in range(6)]}
    s.monitors = [{"name": f"{svc[0]} errors", "status": "Alert",
                   "query": f"avg:{svc[0]}.errors > 10"}]
    s.resources = {"ecs": [{"name": x, "desiredCount": 3, "runningCount": 3,
                            "taskDefinition": f"{x}:1"} for x in svc]}
    return s


def build_scenario(case: dict[str, Any], rng: random.Random):
    """Scenario with NATURAL telemetry phrasing (log lines written like the
    hand-built demo scenarios, not keyword echoes): the policy must learn
    telemetry -> root-cause extraction, which is what transfers to the
    held-out eval scenarios."""
    from ..providers.simulation import SimScenario

    arch_name = case.get("_arch")
    if arch_name == "_generic":
        return _build_generic_scenario(case, rng)
    tmpl = NATURAL_TELEMETRY.get(arch_name)
    svc = case["expected"]["affectedServices"]
    kw = case["expected"]["rootCauseKeywords"]
    s = SimScenario(name=case["id"])
    s.incident = {
        "id": case["incidentId"], "title": case["query"],
        "status": "triggered", "urgency": rng.choice(["high", "critical"]),
        "service": svc[0], "createdAt": "2026-02-12T10:00:00Z",
    }
    noise_svc = rng.choice(["metrics-agent", "log-shipper", "cron-runner",
                            "fluentd", "health-prober"])
    s.services = ([{"name": x, "status": "degraded" if i < 2 else "healthy",
                    "type": "ecs"} for i, x in enumerate(svc)]
                  + [{"name": noise_svc, "status": "healthy", "type": "ecs"}])
    fmt = {"svc": svc[0], "svc1": svc[-1], "extra0": kw[-1],
           "ver": f"{rng.randint(1,9)}.{rng.randint(0,20)}.{rng.randint(0,9)}"}
    if tmpl:
        s.alarms = [{"name": f"{svc[0]}-alert", "state": "ALARM",
                     "reason": tmpl["alarm"].format(**fmt), "service": svc[0]}]
        msgs = list(tmpl["logs"])
        rng.shuffle(msgs)
        for i, msg in enumerate(msgs[:rng.randint(2, 4)]):
            s.log_events.append({
                "timestamp": f"2026-02-12T10:{i:02d}:{(i * 13) % 60:02d}Z",
                "service": rng.choice(svc), "level": "ERROR",
                "message": msg.format(**fmt)})
        # noise line that must NOT become the root cause
        s.log_events.append({
            "timestamp": "2026-02-12T10:09:00Z", "service": noise_svc,
            "level": "INFO", "message": "scheduled healthcheck completed"})
        base = rng.uniform(40, 300)
        s.metrics = {tmpl["metric"].format(**fmt):
                     [round(base * (1 + 0.4 * i), 1) for i in range(6)]}
        s.monitors = [{"name": f"{svc[0]} {arch_name}", "status": "Alert",
                       "query": f"avg:{tmpl['metric'].format(**fmt)} > 10"}]
    else:
        return SimScenario.from_fixture(case)
    if arch_name == "deploy-regression":
        s.deployments = [{"service": svc[0], "version": f"v{fmt['ver']}",
                          "at": "2026-02-12T09:58:00Z",
                          "change": f"release v{fmt['ver']}"}]
    s.resources = {"ecs": [{"name": x, "desiredCount": 3, "runningCount": 3,
                            "taskDefinition": f"{x}:1"} for x in svc]}
    return s


@dataclass
class OracleClient:
    """Answers orchestrator prompts correctly from the case's expected
    data; optionally records every (prompt, response) pair."""

    case: dict[str, Any]
    rng: random.Random = field(default_factory=lambda: random.Random(0))
    records: Optional[list[dict[str, str]]] = None

    def _kw(self) -> list[str]:
        return self.case["expected"]["rootCauseKeywords"]

    def _svc(self) -> list[str]:
        return self.case["expected"]["affectedServices"]

    def _record(self, prompt: str, response: str) -> None:
        if self.records is not None:
            kind, body = split_schema_tag(prompt)
            self.records.append({"kind": kind or "", "body": body,
                                 "response": response})

    def complete(self, prompt: str) -> str:
        import json

        kind, _body = split_schema_tag(prompt)
        kw, svc = self._kw(), self._svc()
        phrase = " ".join(kw)
        if kind == "triage":
            out = {"summary": f"{svc[0]} degraded: {phrase}",
                   "symptoms": [f"{phrase} on {svc[0]}"],
                   "affectedServices": svc[:4],
                   "severity": "high"}
        elif kind == "generateHypotheses":
            decoy_arch = self.rng.choice(ARCHETYPES)
            out = {"hypotheses": [
                {"statement": f"{svc[0]} failing due to {phrase}",
                 "rationale": f"telemetry shows {phrase} on {svc[0]}",
                 "priority": 1, "affectedServices": svc[:3]},
                {"statement": f"possible {' '.join(decoy_arch['keywords'][:2])} on {svc[-1]}",
                 "rationale": "secondary signal, lower confidence",
                 "priority": 3},
            ]}
        elif kind == "evaluateEvidence":
            # confirm the hypothesis when it carries the true keywords
            hyp_match = all(k.lower() in prompt.lower() for k in kw[:2])
            out = {"action": "confirm" if hyp_match else "prune",
                   "confidence": 0.9 if hyp_match else 0.2,
                   "reasoning": (f"evidence shows {phrase}" if hyp_match
                                 else "no supporting signal"),
                   "evidence": [{"description": f"{phrase} observed on {svc[0]}",
                                 "supports": hyp_match}]}
        elif kind == "generateConclusion":
            out = {"rootCause": f"{phrase} on {' and '.join(svc[:2])}",
                   "confidence": "high",
                   "summary": f"Root cause of the incident is {phrase} on "
                              f"{svc[0]}; downstream impact on {', '.join(svc[1:2])}",
                   "affectedServices": svc[:4],
                   "evidence": [f"{phrase} in telemetry"]}
        elif kind == "generateRemediation":
            out = {"summary": f"Mitigate {phrase} on {svc[0]}",
                   "steps": [
                       {"description": f"Address {phrase} on {svc[0]}",
                        "risk": "medium", "requiresApproval": True},
                       {"description": "Verify recovery and close incident",
                        "risk": "low", "requiresApproval": False}]}
        elif kind == "analyzeLogs":
            out = {"summary": f"log patterns show {phrase}",
                   "patterns": [{"pattern": phrase, "severity": "error"}],
                   "services": svc[:3]}
        else:
            out = {"answer": f"{phrase} on {svc[0]}"}
        resp = json.dumps(out)
        self._record(prompt, resp)
        return resp

    def chat(self, system: str, user: str, tools=None):  # pragma: no cover
        from ..engine.client import ChatResponse

        return ChatResponse(content=self.complete(user))


def generate_traces(n_cases: int, seed: int = 0,
                    retriever=None) -> list[dict[str, str]]:
    """Run `n_cases` oracle-driven investigations through the REAL
    orchestrator against from_fixture scenarios; returns recorded
    (kind, body, response) dicts."""
    from ..agent.orchestrator import InvestigationOrchestrator
    from ..providers.simulation import set_thread_scenario
    from ..tools.registry import ToolRegistry

    if retriever is None:
        from ..knowledge.retriever.default import create_retriever

        retriever = create_retriever(in_memory=True)
        retriever.sync()
    rng = random.Random(seed)
    records: list[dict[str, str]] = []
    for i in range(n_cases):
        case = gen_case(rng, i)
        set_thread_scenario(build_scenario(case, rng))
        oracle = OracleClient(case, rng=random.Random(seed * 7919 + i),
                              records=records)
        registry = ToolRegistry(knowledge_retriever=retriever)
        orch = InvestigationOrchestrator(
            llm=oracle, tool_executor=registry, knowledge_retriever=retriever,
            max_iterations=int(case["execute"]["maxIterations"]))
        orch.investigate(case["query"], incident_id=case.get("incidentId"))
    set_thread_scenario(None)
    return records
