"""Onboarding templates + quick setup.

Parity with reference src/config/onboarding.ts (399 LoC): templates
ecs-rds/serverless/enterprise, quick_setup (L251), config generation.
"""
from __future__ import annotations

import os
from typing import Any

import yaml

TEMPLATES: dict[str, dict[str, Any]] = {
    "ecs-rds": {
        "config": {
            "llm": {"provider": "local", "model": "llama3-8b", "tensorParallel": 1,
                    "dtype": "bf16"},
            "providers": {"aws": {"enabled": True, "region": "us-east-1"},
                          "kubernetes": {"enabled": False}},
            "incident": {"pagerduty": {"enabled": True}},
            "knowledge": {"sources": [{"kind": "filesystem", "path": ".runbook/runbooks"}]},
            "safety": {"requireApproval": True, "maxMutationsPerSession": 10},
            "agent": {"maxIterations": 10, "maxHypothesisDepth": 4},
        },
        "services": {
            "aws": {"accounts": [{"accountId": "default", "region": "us-east-1"}]},
            "services": [
                {"name": "api", "type": "ecs", "dependsOn": ["db", "cache"]},
                {"name": "db", "type": "rds"},
                {"name": "cache", "type": "elasticache"},
            ],
        },
    },
    "serverless": {
        "config": {
            "llm": {"provider": "local", "model": "llama3-8b", "tensorParallel": 1},
            "providers": {"aws": {"enabled": True, "region": "us-east-1"}},
            "incident": {"opsgenie": {"enabled": True}},
            "knowledge": {"sources": [{"kind": "filesystem", "path": ".runbook/runbooks"}]},
            "safety": {"requireApproval": True},
            "agent": {"maxIterations": 10},
        },
        "services": {
            "services": [
                {"name": "api-lambda", "type": "lambda", "dependsOn": ["dynamo", "queue"]},
                {"name": "dynamo", "type": "dynamodb"},
                {"name": "queue", "type": "sqs"},
            ],
        },
    },
    "enterprise": {
        "config": {
            "llm": {"provider": "local", "model": "llama3-70b", "tensorParallel": 8,
                    "dtype": "bf16"},
            "providers": {"aws": {"enabled": True}, "kubernetes": {"enabled": True},
                          "github": {"enabled": True}},
            "incident": {"pagerduty": {"enabled": True},
                         "slack": {"enabled": True, "channel": "#incidents"}},
            "knowledge": {"sources": [
                {"kind": "filesystem", "path": ".runbook/runbooks"},
                {"kind": "confluence", "options": {"exportDir": ".runbook/confluence-export"}},
            ]},
            "safety": {"requireApproval": True, "maxMutationsPerSession": 5,
                       "cooldownSeconds": 120},
            "agent": {"maxIterations": 15, "maxHypothesisDepth": 4},
        },
        "services": {"services": []},
    },
}


def quick_setup(template: str = "ecs-rds", runbook_dir: str = ".runbook") -> list[str]:
    """Write config.yaml + services.yaml + dirs (reference quickSetup L251)."""
    tpl = TEMPLATES.get(template)
    if tpl is None:
        raise ValueError(f"unknown template '{template}'")
    written: list[str] = []
    os.makedirs(runbook_dir, exist_ok=True)
    for sub in ("runbooks", "skills", "scratchpad", "checkpoints", "learning", "evals", "pending"):
        os.makedirs(os.path.join(runbook_dir, sub), exist_ok=True)
    cfg_path = os.path.join(runbook_dir, "config.yaml")
    if not os.path.exists(cfg_path):
        with open(cfg_path, "w", encoding="utf-8") as f:
            yaml.safe_dump(tpl["config"], f, sort_keys=False)
        written.append(cfg_path)
    svc_path = os.path.join(runbook_dir, "services.yaml")
    if not os.path.exists(svc_path):
        with open(svc_path, "w", encoding="utf-8") as f:
            yaml.safe_dump(tpl["services"], f, sort_keys=False)
        written.append(svc_path)
    return written
