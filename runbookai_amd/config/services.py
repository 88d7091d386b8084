"""services.yaml schema + loading.

Parity with reference src/config/services.ts (277 LoC): zod schemas for
AWS accounts, compute/db/storage/network services, observability
(cloudwatch/datadog/prometheus) — pydantic v2 here.
"""
from __future__ import annotations

import os
from typing import Any, Optional

import yaml
from pydantic import BaseModel, Field

SERVICE_TYPES = (
    "ecs", "eks", "ec2", "lambda", "rds", "dynamodb", "elasticache", "memorydb",
    "s3", "efs", "sqs", "sns", "kinesis", "elbv2", "cloudfront", "apigateway",
    "custom",
)


class AwsAccountEntry(BaseModel):
    account_id: str = Field(default="default", alias="accountId")
    region: str = "us-east-1"
    profile: str = "default"
    role_arn: str = Field(default="", alias="roleArn")
    model_config = {"populate_by_name": True, "extra": "allow"}


class ServiceEntry(BaseModel):
    name: str
    type: str = "custom"
    depends_on: list[Any] = Field(default_factory=list, alias="dependsOn")
    owner: str = ""
    oncall: str = ""
    tier: str = ""
    slack: str = ""
    endpoints: list[str] = Field(default_factory=list)
    model_config = {"populate_by_name": True, "extra": "allow"}


class ObservabilityEntry(BaseModel):
    cloudwatch: dict[str, Any] = Field(default_factory=dict)
    datadog: dict[str, Any] = Field(default_factory=dict)
    prometheus: dict[str, Any] = Field(default_factory=dict)
    model_config = {"extra": "allow"}


class ServicesConfig(BaseModel):
    aws: dict[str, Any] = Field(default_factory=dict)      # {accounts: [...]}
    services: list[ServiceEntry] = Field(default_factory=list)
    observability: ObservabilityEntry = Field(default_factory=ObservabilityEntry)
    model_config = {"extra": "allow"}

    def accounts(self) -> list[AwsAccountEntry]:
        return [AwsAccountEntry.model_validate(a) for a in self.aws.get("accounts", [])]


def load_services_config(path: Optional[str] = None,
                         runbook_dir: str = ".runbook") -> ServicesConfig:
    candidates = [path] if path else [
        os.path.join(runbook_dir, "services.yaml"),
        os.path.join(runbook_dir, "services.yml"),
    ]
    for p in candidates:
        if p and os.path.exists(p):
            with open(p, encoding="utf-8") as f:
                raw = yaml.safe_load(f) or {}
            return ServicesConfig.model_validate(raw)
    return ServicesConfig()


def build_service_graph(config: ServicesConfig):
    """services.yaml -> ServiceGraph for blast-radius analysis."""
    from ..knowledge.store.graph_store import ServiceGraph

    graph = ServiceGraph()
    graph.load_services_config([s.model_dump(by_alias=True) for s in config.services])
    return graph


def validate_services_config(config: ServicesConfig) -> list[str]:
    problems: list[str] = []
    names = set()
    for s in config.services:
        if s.name in names:
            problems.append(f"duplicate service '{s.name}'")
        names.add(s.name)
        if s.type not in SERVICE_TYPES:
            problems.append(f"service '{s.name}': unknown type '{s.type}'")
    for s in config.services:
        for dep in s.depends_on:
            dep_name = dep.get("name") if isinstance(dep, dict) else str(dep)
            if dep_name not in names:
                problems.append(f"service '{s.name}' depends on unknown '{dep_name}'")
    return problems
