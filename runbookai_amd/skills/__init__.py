from .types import SkillDefinition, SkillStep
from .registry import SkillRegistry, get_skill_registry
from .executor import SkillExecutor

__all__ = ["SkillDefinition", "SkillStep", "SkillRegistry", "get_skill_registry", "SkillExecutor"]
