"""Skill registry: 8 built-ins + YAML user skills from .runbook/skills.

Parity with reference src/skills/registry.ts (153 LoC): built-in
registration (L35-50) + user skill loading (L55-85).
"""
from __future__ import annotations

import os
from typing import Any, Optional

import yaml

from .builtin import BUILTIN_SKILLS
from .types import SkillDefinition


class SkillRegistry:
    def __init__(self) -> None:
        self._skills: dict[str, SkillDefinition] = {}
        for s in BUILTIN_SKILLS:
            self.register(s)

    def register(self, skill: SkillDefinition) -> None:
        self._skills[skill.id] = skill

    def get(self, skill_id: str) -> Optional[SkillDefinition]:
        return self._skills.get(skill_id)

    def list(self) -> list[SkillDefinition]:
        return list(self._skills.values())

    def find_for_service(self, service: str) -> list[SkillDefinition]:
        return [s for s in self._skills.values()
                if not s.applicable_services or service in s.applicable_services]

    def load_user_skills(self, directory: str = ".runbook/skills") -> int:
        """YAML user skills (reference registry.ts:55-85)."""
        if not os.path.isdir(directory):
            return 0
        loaded = 0
        for fn in sorted(os.listdir(directory)):
            if not fn.endswith((".yaml", ".yml")):
                continue
            try:
                with open(os.path.join(directory, fn), encoding="utf-8") as f:
                    data = yaml.safe_load(f)
                if isinstance(data, dict) and data.get("id") or data.get("name"):
                    self.register(SkillDefinition.from_dict(data))
                    loaded += 1
            except (yaml.YAMLError, OSError, AttributeError):
                continue
        return loaded

    def validate(self, skill_id: str) -> dict[str, Any]:
        skill = self.get(skill_id)
        if skill is None:
            return {"valid": False, "errors": [f"unknown skill '{skill_id}'"]}
        errors = []
        ids = set()
        for step in skill.steps:
            if not step.id:
                errors.append("step missing id")
            elif step.id in ids:
                errors.append(f"duplicate step id '{step.id}'")
            ids.add(step.id)
            if not step.action:
                errors.append(f"step '{step.id}' missing action")
            if step.on_error not in ("continue", "abort", "retry"):
                errors.append(f"step '{step.id}' has invalid onError '{step.on_error}'")
        return {"valid": not errors, "errors": errors}


_registry: Optional[SkillRegistry] = None


def get_skill_registry() -> SkillRegistry:
    global _registry
    if _registry is None:
        _registry = SkillRegistry()
    return _registry
