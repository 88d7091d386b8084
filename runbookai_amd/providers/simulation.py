"""Simulated incident environment backing every provider tool.

The reference's providers call live AWS/Datadog/PagerDuty/... HTTP APIs.
This environment has no egress, and BASELINE.json specifies measurement on
a "simulated incident set" — so the provider layer reads from a
SimScenario: a deterministic in-memory world (services, alarms, logs,
metrics, incidents, deployments, pods) that tools query exactly as they
would query the real APIs. Scenarios are either built-in (demo), generated
from eval fixtures, or loaded from YAML.

This mirrors the role of the reference's scripts/simulate/setup-incidents.sh
(which provisions REAL AWS failures for manual testing) as an in-process,
hermetic equivalent.
"""
from __future__ import annotations

import random
import threading
from dataclasses import dataclass, field
from typing import Any, Optional

_tlocal = threading.local()


@dataclass
class SimScenario:
    name: str = "default"
    incident: dict[str, Any] = field(default_factory=dict)
    services: list[dict[str, Any]] = field(default_factory=list)
    alarms: list[dict[str, Any]] = field(default_factory=list)
    log_events: list[dict[str, Any]] = field(default_factory=list)
    metrics: dict[str, list[float]] = field(default_factory=dict)
    deployments: list[dict[str, Any]] = field(default_factory=list)
    pods: list[dict[str, Any]] = field(default_factory=list)
    monitors: list[dict[str, Any]] = field(default_factory=list)
    resources: dict[str, list[dict[str, Any]]] = field(default_factory=dict)  # aws service -> items
    notes: list[dict[str, Any]] = field(default_factory=list)  # incident notes added by tools
    slack_messages: list[dict[str, Any]] = field(default_factory=list)
    mutations: list[dict[str, Any]] = field(default_factory=list)

    # ------------------------------------------------------------------ built-ins

    @classmethod
    def redis_exhaustion(cls) -> "SimScenario":
        """The flagship demo scenario (reference demo/demo-data.ts: scripted
        Redis-connection-exhaustion investigation)."""
        s = cls(name="redis-conn-exhaustion")
        s.incident = {
            "id": "PD-EXAMPLE-001",
            "title": "checkout-api latency spiked and redis timeouts increased",
            "status": "triggered",
            "urgency": "high",
            "service": "checkout-api",
            "createdAt": "2026-02-10T09:10:00Z",
        }
        s.services = [
            {"name": "checkout-api", "status": "degraded", "type": "ecs"},
            {"name": "cart-service", "status": "degraded", "type": "ecs"},
            {"name": "redis", "status": "saturated", "type": "elasticache"},
            {"name": "payment-service", "status": "healthy", "type": "ecs"},
        ]
        s.alarms = [
            {"name": "checkout-api-p99-latency", "state": "ALARM",
             "reason": "p99 > 2000ms for 15 minutes", "service": "checkout-api"},
            {"name": "redis-connected-clients", "state": "ALARM",
             "reason": "connected_clients > 950 (maxclients 1000)", "service": "redis"},
            {"name": "payment-success-rate", "state": "OK", "reason": "", "service": "payment-service"},
        ]
        s.log_events = [
            {"timestamp": "2026-02-10T09:12:03Z", "service": "checkout-api", "level": "ERROR",
             "message": "redis: connection pool exhausted (100/100 in use), waited 5000ms"},
            {"timestamp": "2026-02-10T09:12:09Z", "service": "cart-service", "level": "ERROR",
             "message": "dial tcp 10.0.3.12:6379: i/o timeout to redis"},
            {"timestamp": "2026-02-10T09:13:41Z", "service": "checkout-api", "level": "ERROR",
             "message": "redis: connection pool exhausted (100/100 in use), waited 5000ms"},
            {"timestamp": "2026-02-10T09:14:02Z", "service": "cart-service", "level": "WARN",
             "message": "retrying GET cart:u-8812 after timeout (attempt 3)"},
            {"timestamp": "2026-02-10T09:15:00Z", "service": "checkout-api", "level": "ERROR",
             "message": "checkout failed: upstream cart-service returned 503"},
        ]
        s.metrics = {
            "redis.net.clients": [420, 455, 512, 778, 943, 991, 998, 1000],
            "checkout-api.latency.p99": [180, 210, 240, 890, 1900, 2400, 2600, 2550],
            "cart-service.error_rate": [0.1, 0.2, 0.1, 4.5, 12.0, 18.2, 22.0, 19.8],
        }
        s.deployments = [
            {"service": "cart-service", "version": "v2026.02.10-1", "at": "2026-02-10T09:02:00Z",
             "change": "config: redis pool size 200 -> 100"},
        ]
        s.pods = [
            {"name": "checkout-api-7f9c", "namespace": "prod", "status": "Running", "restarts": 0,
             "cpu": "240m", "memory": "512Mi"},
            {"name": "cart-service-1b2d", "namespace": "prod", "status": "Running", "restarts": 3,
             "cpu": "180m", "memory": "420Mi"},
        ]
        s.monitors = [
            {"name": "checkout latency", "status": "Alert", "query": "avg:checkout.latency{*} > 2000"},
            {"name": "redis clients", "status": "Alert", "query": "avg:redis.net.clients{*} > 900"},
        ]
        s.resources = {
            "elasticache": [{"id": "redis-prod-001", "engine": "redis", "status": "available",
                             "nodes": 3, "maxclients": 1000}],
            "ecs": [{"name": "checkout-api", "desiredCount": 6, "runningCount": 6,
                     "taskDefinition": "checkout-api:118"},
                    {"name": "cart-service", "desiredCount": 4, "runningCount": 4,
                     "taskDefinition": "cart-service:201"}],
        }
        return s

    @classmethod
    def gateway_5xx(cls) -> "SimScenario":
        s = cls(name="api-gateway-5xx")
        s.incident = {
            "id": "PD-EXAMPLE-002",
            "title": "API gateway started returning 5xx after deploy",
            "status": "triggered", "urgency": "high", "service": "api-gateway",
            "createdAt": "2026-02-11T14:05:00Z",
        }
        s.services = [
            {"name": "api-gateway", "status": "degraded", "type": "ecs"},
            {"name": "user-service", "status": "unhealthy", "type": "ecs"},
        ]
        s.alarms = [
            {"name": "gateway-5xx-rate", "state": "ALARM",
             "reason": "5xx rate > 5% for 10 minutes", "service": "api-gateway"},
        ]
        s.log_events = [
            {"timestamp": "2026-02-11T14:06:10Z", "service": "api-gateway", "level": "ERROR",
             "message": "upstream unavailable: user-service timed out after 10s (gateway timeout)"},
            {"timestamp": "2026-02-11T14:06:30Z", "service": "user-service", "level": "ERROR",
             "message": "panic: nil pointer dereference in handler v3.4.0 (deploy 14:03)"},
            {"timestamp": "2026-02-11T14:07:00Z", "service": "api-gateway", "level": "ERROR",
             "message": "504 gateway timeout routing /v1/users"},
        ]
        s.metrics = {
            "api-gateway.5xx_rate": [0.1, 0.1, 0.2, 6.5, 9.1, 11.2],
            "user-service.availability": [100, 100, 100, 42, 18, 11],
        }
        s.deployments = [
            {"service": "user-service", "version": "v3.4.0", "at": "2026-02-11T14:03:00Z",
             "change": "release v3.4.0: new profile handler"},
        ]
        s.monitors = [{"name": "gateway 5xx", "status": "Alert", "query": "sum:gateway.5xx{*} > 100"}]
        s.resources = {"ecs": [{"name": "user-service", "desiredCount": 4, "runningCount": 2,
                                "taskDefinition": "user-service:77"}]}
        return s

    @classmethod
    def kafka_disk_pressure(cls) -> "SimScenario":
        """Broker disk filling up → producers time out; the cause is a
        retention misconfiguration, NOT a deploy (exercises non-deploy
        causality: the deploy history is clean)."""
        s = cls(name="kafka-disk-pressure")
        s.incident = {
            "id": "PD-EXAMPLE-003",
            "title": "order events delayed; kafka producers timing out",
            "status": "triggered", "urgency": "high", "service": "order-events",
            "createdAt": "2026-02-12T03:40:00Z",
        }
        s.services = [
            {"name": "order-events", "status": "degraded", "type": "ecs"},
            {"name": "kafka-broker-2", "status": "unhealthy", "type": "msk"},
            {"name": "analytics-ingest", "status": "degraded", "type": "ecs"},
        ]
        s.alarms = [
            {"name": "kafka-broker-2-disk-used", "state": "ALARM",
             "reason": "disk used > 95% on /var/kafka (volume kafka-data-2)",
             "service": "kafka-broker-2"},
            {"name": "order-events-producer-errors", "state": "ALARM",
             "reason": "producer error rate > 2% for 20 minutes", "service": "order-events"},
        ]
        s.log_events = [
            {"timestamp": "2026-02-12T03:41:12Z", "service": "kafka-broker-2", "level": "ERROR",
             "message": "No space left on device: /var/kafka/order-events-7/00000000.log"},
            {"timestamp": "2026-02-12T03:41:40Z", "service": "order-events", "level": "ERROR",
             "message": "KafkaTimeoutError: batch for order-events-7 expired after 30000ms"},
            {"timestamp": "2026-02-12T03:42:05Z", "service": "kafka-broker-2", "level": "WARN",
             "message": "log retention check skipped: retention.ms=-1 on topic order-events (unbounded)"},
            {"timestamp": "2026-02-12T03:44:20Z", "service": "analytics-ingest", "level": "WARN",
             "message": "consumer lag growing on order-events: 1.2M messages behind"},
        ]
        s.metrics = {
            "kafka-broker-2.disk.used_pct": [71, 78, 84, 89, 93, 96, 98, 99],
            "order-events.producer.error_rate": [0.0, 0.0, 0.1, 0.4, 1.8, 3.5, 5.2, 6.0],
            "analytics-ingest.consumer.lag": [1200, 1500, 9000, 120000, 480000, 1200000],
        }
        s.deployments = []   # clean deploy history — the cause is config drift
        s.pods = [
            {"name": "order-events-5d4f", "namespace": "prod", "status": "Running",
             "restarts": 0, "cpu": "310m", "memory": "800Mi"},
        ]
        s.monitors = [
            {"name": "kafka disk", "status": "Alert",
             "query": "max:kafka.disk.used_pct{broker:2} > 90"},
        ]
        s.resources = {
            "msk": [{"id": "kafka-prod", "brokers": 3, "status": "active",
                     "storagePerBrokerGiB": 512}],
            "ebs": [{"id": "kafka-data-2", "sizeGiB": 512, "usedPct": 99,
                     "attachedTo": "kafka-broker-2"}],
        }
        return s

    @classmethod
    def cert_expiry(cls) -> "SimScenario":
        """Expired TLS certificate on an internal endpoint → handshake
        failures cascading into upstream 503s; no alarm fires on the cert
        itself (exercises log-driven causality over alarm-driven)."""
        s = cls(name="tls-cert-expiry")
        s.incident = {
            "id": "PD-EXAMPLE-004",
            "title": "payments failing: upstream TLS errors to auth-service",
            "status": "triggered", "urgency": "high", "service": "payment-service",
            "createdAt": "2026-02-13T00:02:00Z",
        }
        s.services = [
            {"name": "payment-service", "status": "degraded", "type": "ecs"},
            {"name": "auth-service", "status": "degraded", "type": "ecs"},
        ]
        s.alarms = [
            {"name": "payment-success-rate", "state": "ALARM",
             "reason": "success rate < 90% for 10 minutes", "service": "payment-service"},
        ]
        s.log_events = [
            {"timestamp": "2026-02-13T00:01:02Z", "service": "payment-service", "level": "ERROR",
             "message": "x509: certificate has expired or is not yet valid: auth-internal.prod "
                        "notAfter=2026-02-13T00:00:00Z"},
            {"timestamp": "2026-02-13T00:01:30Z", "service": "payment-service", "level": "ERROR",
             "message": "TLS handshake error dialing auth-internal.prod:8443: bad certificate"},
            {"timestamp": "2026-02-13T00:02:10Z", "service": "auth-service", "level": "WARN",
             "message": "cert-rotation job auth-internal last succeeded 92 days ago (expected every 60d)"},
            {"timestamp": "2026-02-13T00:03:00Z", "service": "payment-service", "level": "ERROR",
             "message": "POST /charge returning 503: auth unavailable"},
        ]
        s.metrics = {
            "payment-service.success_rate": [99.9, 99.8, 99.9, 41.0, 8.2, 3.1],
            "auth-service.tls.handshake_errors": [0, 0, 0, 180, 1450, 2600],
        }
        s.deployments = []
        s.monitors = [
            {"name": "payment success", "status": "Alert",
             "query": "avg:payment.success_rate{*} < 90"},
        ]
        s.resources = {
            "acm": [{"id": "auth-internal.prod", "status": "EXPIRED",
                     "notAfter": "2026-02-13T00:00:00Z", "renewalEligibility": "ELIGIBLE"}],
        }
        return s

    @classmethod
    def oom_crashloop(cls) -> "SimScenario":
        """Recommendation service OOM-killed in a loop after a feature
        flag doubled the embedding cache; exercises k8s pod-state +
        memory-metric causality."""
        s = cls(name="oom-crashloop")
        s.incident = {
            "id": "PD-EXAMPLE-005",
            "title": "recommendation-svc unavailable: pods restarting repeatedly",
            "status": "triggered", "urgency": "high",
            "service": "recommendation-svc", "createdAt": "2026-02-14T08:10:00Z",
        }
        s.services = [
            {"name": "recommendation-svc", "status": "unhealthy", "type": "eks"},
            {"name": "feed-api", "status": "degraded", "type": "ecs"},
        ]
        s.alarms = [
            {"name": "recommendation-svc-restarts", "state": "ALARM",
             "reason": "container restarts > 5 in 10 minutes",
             "service": "recommendation-svc"},
        ]
        s.log_events = [
            {"timestamp": "2026-02-14T08:08:12Z", "service": "recommendation-svc",
             "level": "ERROR",
             "message": "Last State: Terminated, Reason: OOMKilled, exit code 137 "
                        "(memory limit 2Gi)"},
            {"timestamp": "2026-02-14T08:08:40Z", "service": "recommendation-svc",
             "level": "WARN",
             "message": "embedding cache resident size 1.9GiB after flag "
                        "rec_cache_v2=on (was 0.8GiB)"},
            {"timestamp": "2026-02-14T08:09:30Z", "service": "feed-api",
             "level": "ERROR",
             "message": "upstream recommendation-svc: connection refused (pod restarting)"},
        ]
        s.metrics = {
            "recommendation-svc.memory_rss_gib": [0.8, 0.9, 1.4, 1.9, 2.0, 2.0],
            "recommendation-svc.restart_count": [0, 0, 1, 3, 6, 9],
        }
        s.pods = [
            {"name": "recommendation-svc-7d9f-1", "phase": "CrashLoopBackOff",
             "restarts": 9, "reason": "OOMKilled"},
            {"name": "recommendation-svc-7d9f-2", "phase": "CrashLoopBackOff",
             "restarts": 8, "reason": "OOMKilled"},
        ]
        s.monitors = [{"name": "rec-svc availability", "status": "Alert",
                       "query": "avg:rec.availability{*} < 95"}]
        s.resources = {"eks": [{"name": "recommendation-svc", "desiredCount": 4,
                                "runningCount": 0, "taskDefinition": "rec:44"}]}
        return s

    @classmethod
    def dns_resolution(cls) -> "SimScenario":
        """Intermittent SERVFAIL from cluster DNS after a nameserver
        config push; exercises infrastructure-layer causality below the
        application services."""
        s = cls(name="dns-resolution")
        s.incident = {
            "id": "PD-EXAMPLE-006",
            "title": "intermittent connection failures across services: name "
                     "resolution errors",
            "status": "triggered", "urgency": "high",
            "service": "service-mesh", "createdAt": "2026-02-15T13:30:00Z",
        }
        s.services = [
            {"name": "service-mesh", "status": "degraded", "type": "eks"},
            {"name": "coredns", "status": "degraded", "type": "eks"},
            {"name": "orders-api", "status": "degraded", "type": "ecs"},
        ]
        s.alarms = [
            {"name": "mesh-5xx-rate", "state": "ALARM",
             "reason": "upstream resolution failures above 2%",
             "service": "service-mesh"},
        ]
        s.log_events = [
            {"timestamp": "2026-02-15T13:28:05Z", "service": "orders-api",
             "level": "ERROR",
             "message": "dial tcp: lookup payments-internal.prod.svc on "
                        "10.100.0.10:53: server misbehaving (SERVFAIL)"},
            {"timestamp": "2026-02-15T13:28:40Z", "service": "coredns",
             "level": "ERROR",
             "message": "plugin/forward: upstream 10.0.9.9:53 timeout "
                        "(configured 13:25 by dns-config push #812)"},
            {"timestamp": "2026-02-15T13:29:15Z", "service": "coredns",
             "level": "WARN",
             "message": "dns resolution latency p99 840ms; ndots:5 expansion "
                        "multiplying query volume"},
        ]
        s.metrics = {
            "coredns.servfail_rate": [0.0, 0.0, 0.1, 2.4, 4.8, 5.1],
            "coredns.forward_latency_ms": [2, 2, 3, 410, 790, 840],
        }
        s.deployments = [
            {"service": "coredns", "version": "config-812",
             "at": "2026-02-15T13:25:00Z",
             "change": "dns-config push #812: forward nameserver changed"},
        ]
        s.monitors = [{"name": "cluster dns health", "status": "Alert",
                       "query": "avg:coredns.servfail{*} > 1"}]
        s.resources = {"eks": [{"name": "coredns", "desiredCount": 2,
                                "runningCount": 2, "taskDefinition": "coredns:9"}]}
        return s

    @classmethod
    def queue_backlog(cls) -> "SimScenario":
        """Job queue backlog after the consumer group stalled on a poison
        message; exercises lag/age metrics + dead-letter causality."""
        s = cls(name="queue-backlog")
        s.incident = {
            "id": "PD-EXAMPLE-007",
            "title": "order processing delayed: job queue backlog growing",
            "status": "triggered", "urgency": "high",
            "service": "job-queue", "createdAt": "2026-02-16T19:05:00Z",
        }
        s.services = [
            {"name": "job-queue", "status": "degraded", "type": "sqs"},
            {"name": "worker-pool", "status": "degraded", "type": "ecs"},
        ]
        s.alarms = [
            {"name": "job-queue-age", "state": "ALARM",
             "reason": "oldest message age > 30 minutes", "service": "job-queue"},
        ]
        s.log_events = [
            {"timestamp": "2026-02-16T18:40:20Z", "service": "worker-pool",
             "level": "ERROR",
             "message": "consumer stalled: message 9f31 failed deserialization 12 "
                        "times, not acked (poison message)"},
            {"timestamp": "2026-02-16T18:55:00Z", "service": "job-queue",
             "level": "WARN",
             "message": "queue depth 1.2M and rising; consumer group lag growing"},
            {"timestamp": "2026-02-16T19:01:10Z", "service": "worker-pool",
             "level": "ERROR",
             "message": "backlog processing ETA exceeds SLA; dead letter queue "
                        "not configured for orders topic"},
        ]
        s.metrics = {
            "job-queue.depth": [12000, 15000, 180000, 560000, 910000, 1200000],
            "job-queue.oldest_age_min": [1, 1, 8, 19, 33, 42],
        }
        s.monitors = [{"name": "queue backlog", "status": "Alert",
                       "query": "max:queue.age{*} > 30m"}]
        s.resources = {"sqs": [{"name": "orders-jobs", "depth": 1200000,
                                "dlq": "absent"}]}
        return s

    @classmethod
    def db_cpu_saturation(cls) -> "SimScenario":
        """Reporting database CPU pinned by a full-scan query after an
        index was dropped in a migration; exercises slow-query causality."""
        s = cls(name="db-cpu-saturation")
        s.incident = {
            "id": "PD-EXAMPLE-008",
            "title": "dashboard timeouts: reporting database CPU saturated",
            "status": "triggered", "urgency": "high",
            "service": "reporting-db", "createdAt": "2026-02-17T09:45:00Z",
        }
        s.services = [
            {"name": "reporting-db", "status": "degraded", "type": "rds"},
            {"name": "dashboard-api", "status": "degraded", "type": "ecs"},
        ]
        s.alarms = [
            {"name": "reporting-db-cpu", "state": "ALARM",
             "reason": "CPUUtilization > 95% for 15 minutes",
             "service": "reporting-db"},
        ]
        s.log_events = [
            {"timestamp": "2026-02-17T09:40:00Z", "service": "reporting-db",
             "level": "WARN",
             "message": "slow query (11.8s): SELECT ... FROM order_events WHERE "
                        "tenant_id = $1 — Seq Scan on order_events "
                        "(idx_order_events_tenant missing)"},
            {"timestamp": "2026-02-17T09:41:30Z", "service": "reporting-db",
             "level": "WARN",
             "message": "cpu saturation: 97% utilization, 48 active backends "
                        "waiting on CPU"},
            {"timestamp": "2026-02-17T09:43:00Z", "service": "dashboard-api",
             "level": "ERROR",
             "message": "statement timeout after 10s querying reporting-db"},
        ]
        s.metrics = {
            "reporting-db.cpu_util": [22, 24, 61, 93, 97, 98],
            "dashboard-api.timeout_rate": [0, 0, 2, 11, 29, 35],
        }
        s.deployments = [
            {"service": "reporting-db", "version": "migration-0412",
             "at": "2026-02-17T09:30:00Z",
             "change": "migration 0412: rebuilt order_events (index drop)"},
        ]
        s.monitors = [{"name": "reporting db cpu", "status": "Alert",
                       "query": "avg:rds.cpu{id:reporting-db} > 90"}]
        s.resources = {"rds": [{"name": "reporting-db", "status": "available",
                                "class": "db.r6g.4xlarge"}]}
        return s

    @classmethod
    def nat_port_exhaustion(cls) -> "SimScenario":
        """Outbound SNAT port exhaustion on the NAT gateway after a deploy
        disabled HTTP keep-alives — external calls fail intermittently while
        everything in-VPC looks healthy."""
        s = cls(name="nat-port-exhaustion")
        s.incident = {
            "id": "PD-EXAMPLE-009",
            "title": "payment-service reporting intermittent gateway timeouts to the card processor",
            "status": "triggered",
            "urgency": "high",
            "service": "payment-service",
            "createdAt": "2026-03-02T14:05:00Z",
        }
        s.services = [
            {"name": "payment-service", "status": "degraded", "type": "ecs"},
            {"name": "fraud-scorer", "status": "degraded", "type": "ecs"},
            {"name": "checkout-api", "status": "healthy", "type": "ecs"},
            {"name": "nat-gateway-prod", "status": "saturated", "type": "network"},
        ]
        s.alarms = [
            {"name": "payment-external-errors", "state": "ALARM",
             "reason": "card-processor call failures > 8% for 10 minutes",
             "service": "payment-service"},
            {"name": "nat-gw-ErrorPortAllocation", "state": "ALARM",
             "reason": "ErrorPortAllocation > 0 (SNAT ports exhausted)",
             "service": "nat-gateway-prod"},
            {"name": "payment-cpu", "state": "OK", "reason": "", "service": "payment-service"},
        ]
        s.log_events = [
            {"timestamp": "2026-03-02T14:06:11Z", "service": "payment-service", "level": "ERROR",
             "message": "POST https://api.cardprocessor.example/v2/charge: dial tcp: connect: connection timed out"},
            {"timestamp": "2026-03-02T14:06:40Z", "service": "fraud-scorer", "level": "ERROR",
             "message": "GET https://ipinfo.external.example/lookup: i/o timeout after 10s"},
            {"timestamp": "2026-03-02T14:07:02Z", "service": "payment-service", "level": "WARN",
             "message": "retrying charge c-55281 (attempt 2/3) — previous attempt timed out"},
            {"timestamp": "2026-03-02T14:08:15Z", "service": "payment-service", "level": "ERROR",
             "message": "charge failed after 3 attempts: upstream unreachable (all external calls affected)"},
            {"timestamp": "2026-03-02T14:09:00Z", "service": "payment-service", "level": "INFO",
             "message": "in-VPC dependencies healthy: db 2ms, redis 1ms — only egress is failing"},
        ]
        s.metrics = {
            "nat.ErrorPortAllocation": [0, 0, 0, 12, 240, 890, 1450, 1600],
            "nat.ActiveConnectionCount": [18000, 19500, 21000, 44000, 61000, 64000, 64500, 64512],
            "payment-service.external_error_rate": [0.2, 0.3, 0.2, 3.1, 8.4, 11.9, 12.5, 12.2],
        }
        s.deployments = [
            {"service": "payment-service", "version": "v2026.03.02-4", "at": "2026-03-02T13:48:00Z",
             "change": "http client rewrite: per-request connections (keep-alive disabled)"},
        ]
        s.pods = [
            {"name": "payment-service-9d1f", "namespace": "prod", "status": "Running",
             "restarts": 0, "cpu": "310m", "memory": "640Mi"},
            {"name": "fraud-scorer-3c77", "namespace": "prod", "status": "Running",
             "restarts": 0, "cpu": "150m", "memory": "380Mi"},
        ]
        s.monitors = [
            {"name": "external call error rate", "status": "Alert",
             "query": "avg:payment.external.errors{*} > 5"},
            {"name": "nat port allocation errors", "status": "Alert",
             "query": "sum:aws.natgateway.error_port_allocation{*} > 0"},
        ]
        s.resources = {
            "ec2": [{"id": "nat-0a1b2c3d", "type": "natgateway", "status": "available",
                     "subnet": "private-a", "snatPortsInUse": 64512, "snatPortsMax": 64512}],
            "ecs": [{"name": "payment-service", "desiredCount": 8, "runningCount": 8,
                     "taskDefinition": "payment-service:77"}],
        }
        return s

    @classmethod
    def clock_skew_auth(cls) -> "SimScenario":
        """Auth failures from node clock skew: the time-sync daemon died on
        one node pool, JWTs mint with future timestamps and verifiers reject
        them — a 401 spike that is NOT a credential or deploy problem."""
        s = cls(name="clock-skew-auth")
        s.incident = {
            "id": "PD-EXAMPLE-010",
            "title": "auth-service 401 rate spiked; users intermittently logged out",
            "status": "triggered",
            "urgency": "high",
            "service": "auth-service",
            "createdAt": "2026-03-05T03:22:00Z",
        }
        s.services = [
            {"name": "auth-service", "status": "degraded", "type": "eks"},
            {"name": "session-api", "status": "degraded", "type": "eks"},
            {"name": "user-profile", "status": "healthy", "type": "eks"},
        ]
        s.alarms = [
            {"name": "auth-401-rate", "state": "ALARM",
             "reason": "401 responses > 6% of auth traffic for 20 minutes",
             "service": "auth-service"},
            {"name": "node-clock-sync", "state": "ALARM",
             "reason": "chronyd inactive on node pool spot-c; offset 47s and drifting",
             "service": "auth-service"},
        ]
        s.log_events = [
            {"timestamp": "2026-03-05T03:24:10Z", "service": "session-api", "level": "ERROR",
             "message": "jwt validation failed: token used before issued (iat is in the future)"},
            {"timestamp": "2026-03-05T03:24:31Z", "service": "auth-service", "level": "WARN",
             "message": "issued token with iat 2026-03-05T03:25:18Z from pod on node spot-c-14"},
            {"timestamp": "2026-03-05T03:25:02Z", "service": "session-api", "level": "ERROR",
             "message": "jwt validation failed: nbf (not before) check failed, rejecting request"},
            {"timestamp": "2026-03-05T03:26:40Z", "service": "session-api", "level": "INFO",
             "message": "tokens minted by pods on node pool on-demand-a validate fine"},
            {"timestamp": "2026-03-05T03:27:12Z", "service": "auth-service", "level": "ERROR",
             "message": "node spot-c-14: chronyd unit failed 41 minutes ago; system clock offset +47.2s"},
        ]
        s.metrics = {
            "auth-service.401_rate": [0.4, 0.5, 0.4, 2.1, 4.8, 6.3, 6.9, 6.6],
            "node.spot-c.clock_offset_s": [0.01, 0.02, 0.01, 8.5, 21.0, 34.5, 43.1, 47.2],
            "session-api.jwt_rejections": [2, 1, 3, 180, 560, 840, 960, 910],
        }
        s.deployments = []  # no deploys: the red herring everyone checks first
        s.pods = [
            {"name": "auth-service-5k2m", "namespace": "prod", "status": "Running",
             "restarts": 0, "cpu": "200m", "memory": "500Mi", "node": "spot-c-14"},
            {"name": "auth-service-8n4q", "namespace": "prod", "status": "Running",
             "restarts": 0, "cpu": "190m", "memory": "480Mi", "node": "on-demand-a-02"},
            {"name": "session-api-2j9x", "namespace": "prod", "status": "Running",
             "restarts": 0, "cpu": "160m", "memory": "350Mi", "node": "on-demand-a-03"},
        ]
        s.monitors = [
            {"name": "auth 401 spike", "status": "Alert",
             "query": "sum:auth.responses.401{*}.as_rate() > 0.06"},
            {"name": "node clock offset", "status": "Alert",
             "query": "max:node.clock.offset{pool:spot-c} > 1"},
        ]
        s.resources = {
            "eks": [{"name": "prod-cluster", "nodePools": [
                {"name": "spot-c", "nodes": 6, "issue": "chronyd inactive on 3 nodes"},
                {"name": "on-demand-a", "nodes": 8, "issue": None}]}],
        }
        return s

    @classmethod
    def from_fixture(cls, case: dict[str, Any]) -> "SimScenario":
        """Generate a scenario from an eval fixture case: the telemetry
        reflects the expected root cause so a competent agent can find it."""
        expected = case.get("expected", {})
        keywords = expected.get("rootCauseKeywords", ["error"])
        services = expected.get("affectedServices", ["service-a"])
        rng = random.Random(case.get("id", "seed"))
        s = cls(name=case.get("id", "generated"))
        s.incident = {
            "id": case.get("incidentId", "PD-GEN-001"),
            "title": case.get("query", "incident"),
            "status": "triggered", "urgency": "high",
            "service": services[0] if services else "unknown",
            "createdAt": "2026-02-12T10:00:00Z",
        }
        s.services = [{"name": svc, "status": "degraded" if i < 2 else "healthy", "type": "ecs"}
                      for i, svc in enumerate(services)]
        phrase = " ".join(keywords)
        for i, svc in enumerate(services[:3]):
            s.alarms.append({
                "name": f"{svc}-health", "state": "ALARM",
                "reason": f"{phrase} detected on {svc}", "service": svc,
            })
            for j in range(3):
                s.log_events.append({
                    "timestamp": f"2026-02-12T10:{i:02d}:{j * 7:02d}Z", "service": svc,
                    "level": "ERROR",
                    "message": f"{svc}: {phrase} (occurrence {j + 1})",
                })
        base = rng.uniform(50, 200)
        s.metrics = {f"{services[0]}.error_rate": [round(base * (1 + 0.5 * i), 1) for i in range(6)]}
        s.monitors = [{"name": f"{services[0]} errors", "status": "Alert",
                       "query": f"avg:{services[0]}.errors > 10"}]
        s.resources = {"ecs": [{"name": svc, "desiredCount": 3, "runningCount": 3,
                                "taskDefinition": f"{svc}:1"} for svc in services]}
        context = case.get("context", "")
        if context:
            s.log_events.append({"timestamp": "2026-02-12T10:05:00Z",
                                 "service": services[0] if services else "unknown",
                                 "level": "ERROR", "message": context})
        return s


_SCENARIOS = {
    "redis-conn-exhaustion": SimScenario.redis_exhaustion,
    "api-gateway-5xx": SimScenario.gateway_5xx,
    "kafka-disk-pressure": SimScenario.kafka_disk_pressure,
    "tls-cert-expiry": SimScenario.cert_expiry,
    "oom-crashloop": SimScenario.oom_crashloop,
    "dns-resolution": SimScenario.dns_resolution,
    "queue-backlog": SimScenario.queue_backlog,
    "db-cpu-saturation": SimScenario.db_cpu_saturation,
    "nat-port-exhaustion": SimScenario.nat_port_exhaustion,
    "clock-skew-auth": SimScenario.clock_skew_auth,
}

_current: Optional[SimScenario] = None


def set_scenario(scenario: Optional[SimScenario]) -> None:
    global _current
    _current = scenario


def set_thread_scenario(scenario: Optional[SimScenario]) -> None:
    """Thread-scoped override: concurrent investigations (the bench runs 32
    in flight) each pin their own scenario on their worker thread while the
    process-global default stays available to every other thread."""
    _tlocal.scenario = scenario


def get_scenario() -> SimScenario:
    s = getattr(_tlocal, "scenario", None)
    if s is not None:
        return s
    global _current
    if _current is None:
        _current = SimScenario.redis_exhaustion()
    return _current


def load_scenario(name: str) -> SimScenario:
    factory = _SCENARIOS.get(name)
    if factory is None:
        raise KeyError(f"unknown scenario '{name}' (have: {sorted(_SCENARIOS)})")
    return factory()
