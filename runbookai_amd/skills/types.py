"""Skill type contracts.

Parity with reference src/skills/types.ts (79 LoC): SkillDefinition
{id, name, parameters, steps[], riskLevel, applicableServices}; SkillStep
{action=tool|'prompt', parameters, condition, requiresApproval, onError
continue/abort/retry + maxRetries} (L16-48).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Optional


@dataclass
class SkillStep:
    id: str
    action: str                      # a tool name, or the literal 'prompt'
    parameters: dict[str, Any] = field(default_factory=dict)
    prompt: str = ""                 # used when action == 'prompt'
    condition: str = ""              # e.g. "{{steps.check.result.count}} > 0"
    requires_approval: bool = False
    on_error: str = "abort"          # continue | abort | retry
    max_retries: int = 2
    description: str = ""

    @classmethod
    def from_dict(cls, d: dict[str, Any]) -> "SkillStep":
        return cls(
            id=d.get("id", ""),
            action=d.get("action", ""),
            parameters=d.get("parameters", {}) or {},
            prompt=d.get("prompt", ""),
            condition=d.get("condition", ""),
            requires_approval=bool(d.get("requiresApproval", False)),
            on_error=d.get("onError", "abort"),
            max_retries=int(d.get("maxRetries", 2)),
            description=d.get("description", ""),
        )


@dataclass
class SkillDefinition:
    id: str
    name: str
    description: str = ""
    parameters: dict[str, Any] = field(default_factory=dict)   # JSON schema-ish
    steps: list[SkillStep] = field(default_factory=list)
    risk_level: str = "low"
    applicable_services: list[str] = field(default_factory=list)

    @classmethod
    def from_dict(cls, d: dict[str, Any]) -> "SkillDefinition":
        return cls(
            id=d.get("id", d.get("name", "")),
            name=d.get("name", d.get("id", "")),
            description=d.get("description", ""),
            parameters=d.get("parameters", {}) or {},
            steps=[SkillStep.from_dict(s) for s in d.get("steps", [])],
            risk_level=d.get("riskLevel", "low"),
            applicable_services=list(d.get("applicableServices", [])),
        )

    def to_dict(self) -> dict[str, Any]:
        return {
            "id": self.id, "name": self.name, "description": self.description,
            "parameters": self.parameters, "riskLevel": self.risk_level,
            "applicableServices": self.applicable_services,
            "steps": [
                {"id": s.id, "action": s.action, "parameters": s.parameters,
                 "condition": s.condition, "requiresApproval": s.requires_approval,
                 "onError": s.on_error}
                for s in self.steps
            ],
        }
