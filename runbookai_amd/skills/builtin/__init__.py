"""The 8 built-in skills (reference src/skills/builtin/*.ts, registry.ts:17-24):
investigate-incident, deploy-service, scale-service, troubleshoot-service,
rollback-deployment, cost-analysis, investigate-cost-spike, security-audit.

Shape mirrors builtin/investigate-incident.ts:9-80 — declarative steps
chaining prior results via {{steps.<id>.result.*}} templates with per-step
onError policy.
"""
from __future__ import annotations

from ..types import SkillDefinition


def _skill(d: dict) -> SkillDefinition:
    return SkillDefinition.from_dict(d)


INVESTIGATE_INCIDENT = _skill({
    "id": "investigate-incident",
    "name": "Investigate incident",
    "description": "Seed an investigation from an incident: fetch incident, pull alarms/logs, search knowledge.",
    "parameters": {"incidentId": {"type": "string", "required": True}},
    "riskLevel": "low",
    "steps": [
        {"id": "fetch_incident", "action": "pagerduty_get_incident",
         "parameters": {"incidentId": "{{incidentId}}"}, "onError": "continue"},
        {"id": "alarms", "action": "cloudwatch_alarms", "parameters": {"state": "ALARM"},
         "onError": "continue"},
        {"id": "logs", "action": "cloudwatch_logs",
         "parameters": {"filter": "{{steps.fetch_incident.result.incident.title}}", "limit": 30},
         "onError": "continue"},
        {"id": "knowledge", "action": "search_knowledge",
         "parameters": {"query": "{{steps.fetch_incident.result.incident.title}}", "limit": 5},
         "onError": "continue"},
    ],
})

DEPLOY_SERVICE = _skill({
    "id": "deploy-service",
    "name": "Deploy service",
    "description": "Deploy a service version with a post-deploy health check.",
    "parameters": {"service": {"type": "string", "required": True},
                   "version": {"type": "string", "required": True}},
    "riskLevel": "high",
    "steps": [
        {"id": "predeploy_health", "action": "cloudwatch_alarms",
         "parameters": {"state": "ALARM", "service": "{{service}}"}, "onError": "abort"},
        {"id": "deploy", "action": "aws_mutate",
         "parameters": {"service": "ecs", "operation": "update-service",
                        "resource": "{{service}}", "version": "{{version}}"},
         "requiresApproval": True, "onError": "abort"},
        {"id": "postdeploy_health", "action": "cloudwatch_alarms",
         "parameters": {"state": "ALARM", "service": "{{service}}"}, "onError": "continue"},
    ],
})

SCALE_SERVICE = _skill({
    "id": "scale-service",
    "name": "Scale service",
    "description": "Scale an ECS service to a target count.",
    "parameters": {"service": {"type": "string", "required": True},
                   "desiredCount": {"type": "integer", "required": True}},
    "riskLevel": "medium",
    "steps": [
        {"id": "current", "action": "aws_query",
         "parameters": {"service": "ecs", "operation": "describe-services"}, "onError": "abort"},
        {"id": "scale", "action": "aws_mutate",
         "parameters": {"service": "ecs", "operation": "update-service",
                        "resource": "{{service}}", "desiredCount": "{{desiredCount}}"},
         "requiresApproval": True, "onError": "abort"},
    ],
})

TROUBLESHOOT_SERVICE = _skill({
    "id": "troubleshoot-service",
    "name": "Troubleshoot service",
    "description": "Gather logs/metrics/events for one service.",
    "parameters": {"service": {"type": "string", "required": True}},
    "riskLevel": "low",
    "steps": [
        {"id": "logs", "action": "cloudwatch_logs",
         "parameters": {"filter": "ERROR", "service": "{{service}}", "limit": 40},
         "onError": "continue"},
        {"id": "metrics", "action": "datadog",
         "parameters": {"action": "metrics", "query": "avg:{{service}}.error_rate{*}"},
         "onError": "continue"},
        {"id": "pods", "action": "kubernetes_query", "parameters": {"action": "pods"},
         "onError": "continue"},
        {"id": "events", "action": "kubernetes_query", "parameters": {"action": "events"},
         "onError": "continue"},
    ],
})

ROLLBACK_DEPLOYMENT = _skill({
    "id": "rollback-deployment",
    "name": "Rollback deployment",
    "description": "Roll a service back to its previous version.",
    "parameters": {"service": {"type": "string", "required": True}},
    "riskLevel": "high",
    "steps": [
        {"id": "deployments", "action": "kubernetes_query",
         "parameters": {"action": "deployments"}, "onError": "abort"},
        {"id": "rollback", "action": "aws_mutate",
         "parameters": {"service": "ecs", "operation": "rollback",
                        "resource": "{{service}}"},
         "requiresApproval": True, "onError": "abort"},
        {"id": "verify", "action": "cloudwatch_alarms",
         "parameters": {"state": "ALARM", "service": "{{service}}"}, "onError": "continue"},
    ],
})

COST_ANALYSIS = _skill({
    "id": "cost-analysis",
    "name": "Cost analysis",
    "description": "Inventory the main cost drivers across compute/storage/db.",
    "parameters": {},
    "riskLevel": "low",
    "steps": [
        {"id": "ec2", "action": "aws_query", "parameters": {"service": "ec2", "operation": "list"},
         "onError": "continue"},
        {"id": "rds", "action": "aws_query", "parameters": {"service": "rds", "operation": "list"},
         "onError": "continue"},
        {"id": "s3", "action": "aws_query", "parameters": {"service": "s3", "operation": "list"},
         "onError": "continue"},
        {"id": "summary", "action": "prompt",
         "prompt": "Summarize cost drivers from: ec2={{steps.ec2.result.count}} "
                   "rds={{steps.rds.result.count}} s3={{steps.s3.result.count}}",
         "onError": "continue"},
    ],
})

INVESTIGATE_COST_SPIKE = _skill({
    "id": "investigate-cost-spike",
    "name": "Investigate cost spike",
    "description": "Correlate a cost spike with scaling/deploy activity.",
    "parameters": {"service": {"type": "string"}},
    "riskLevel": "low",
    "steps": [
        {"id": "autoscaling", "action": "aws_query",
         "parameters": {"service": "autoscaling", "operation": "list"}, "onError": "continue"},
        {"id": "deploys", "action": "github_query", "parameters": {"action": "recent_commits"},
         "onError": "continue"},
        {"id": "analysis", "action": "prompt",
         "prompt": "Given autoscaling activity {{steps.autoscaling.result.count}} and recent "
                   "deploys, hypothesize the cost-spike cause for {{service}}.",
         "onError": "continue"},
    ],
})

SECURITY_AUDIT = _skill({
    "id": "security-audit",
    "name": "Security audit",
    "description": "Read-only inventory of IAM/KMS/secrets/WAF posture.",
    "parameters": {},
    "riskLevel": "low",
    "steps": [
        {"id": "iam", "action": "aws_query", "parameters": {"service": "iam", "operation": "list"},
         "onError": "continue"},
        {"id": "kms", "action": "aws_query", "parameters": {"service": "kms", "operation": "list"},
         "onError": "continue"},
        {"id": "secrets", "action": "aws_query",
         "parameters": {"service": "secretsmanager", "operation": "list"}, "onError": "continue"},
        {"id": "waf", "action": "aws_query", "parameters": {"service": "waf", "operation": "list"},
         "onError": "continue"},
    ],
})

BUILTIN_SKILLS = [
    INVESTIGATE_INCIDENT, DEPLOY_SERVICE, SCALE_SERVICE, TROUBLESHOOT_SERVICE,
    ROLLBACK_DEPLOYMENT, COST_ANALYSIS, INVESTIGATE_COST_SPIKE, SECURITY_AUDIT,
]
