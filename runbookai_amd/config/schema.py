"""Configuration schema + loading.

Parity with reference src/utils/config.ts (425 LoC): full ConfigSchema
(L193-209) — llm, providers {aws, kubernetes, github, gitlab,
operabilityContext}, incident {pagerduty, opsgenie, slack+events},
knowledge {sources, store, retrieval}, safety {requireApproval,
maxMutationsPerSession 10, cooldown 60s}, agent {maxIterations 10,
maxHypothesisDepth 4, contextThresholdTokens 100000},
integrations.claude.sessionStorage; search path .runbook/config.yaml ->
~/.runbook/ (L222-230); ${ENV} interpolation (L253-270); validate_config
per-provider checks (L292-424). Pydantic v2 replaces zod.
"""
from __future__ import annotations

import os
import re
from typing import Any, Optional

import yaml
from pydantic import BaseModel, Field


class LlmConfig(BaseModel):
    provider: str = "local"              # local (MI355X engine) | mock
    model: str = "llama3-8b"             # llama3-8b | llama3-70b | tiny (tests)
    max_tokens: int = Field(default=1024, alias="maxTokens")
    temperature: float = 0.0
    tensor_parallel: int = Field(default=1, alias="tensorParallel")
    dtype: str = "bf16"
    # HF-format checkpoint dir (safetensors [+ tokenizer.json]); empty =
    # seeded random-init weights (the offline default)
    checkpoint: str = Field(default="", alias="checkpointPath")

    model_config = {"populate_by_name": True, "extra": "allow", "protected_namespaces": ()}


class AwsProviderConfig(BaseModel):
    enabled: bool = True
    region: str = "us-east-1"
    accounts: list[dict[str, Any]] = Field(default_factory=list)
    model_config = {"extra": "allow"}


class ProvidersConfig(BaseModel):
    aws: AwsProviderConfig = Field(default_factory=AwsProviderConfig)
    kubernetes: dict[str, Any] = Field(default_factory=lambda: {"enabled": True})
    github: dict[str, Any] = Field(default_factory=dict)
    gitlab: dict[str, Any] = Field(default_factory=dict)
    operabilityContext: dict[str, Any] = Field(default_factory=dict)
    observability: dict[str, Any] = Field(default_factory=lambda: {"enabled": True})
    model_config = {"extra": "allow"}


class IncidentConfig(BaseModel):
    pagerduty: dict[str, Any] = Field(default_factory=dict)
    opsgenie: dict[str, Any] = Field(default_factory=dict)
    slack: dict[str, Any] = Field(default_factory=dict)
    model_config = {"extra": "allow"}


class KnowledgeConfig(BaseModel):
    sources: list[dict[str, Any]] = Field(default_factory=list)
    store: dict[str, Any] = Field(default_factory=dict)
    retrieval: dict[str, Any] = Field(
        default_factory=lambda: {"mode": "hybrid", "ftsWeight": 0.4, "vectorWeight": 0.6})
    model_config = {"extra": "allow"}


class SafetyConfig(BaseModel):
    require_approval: bool = Field(default=True, alias="requireApproval")
    max_mutations_per_session: int = Field(default=10, alias="maxMutationsPerSession")
    cooldown_seconds: float = Field(default=60.0, alias="cooldownSeconds")
    model_config = {"populate_by_name": True, "extra": "allow"}


class AgentSection(BaseModel):
    max_iterations: int = Field(default=10, alias="maxIterations")
    max_hypothesis_depth: int = Field(default=4, alias="maxHypothesisDepth")
    context_threshold_tokens: int = Field(default=100_000, alias="contextThresholdTokens")
    model_config = {"populate_by_name": True, "extra": "allow"}


class IntegrationsConfig(BaseModel):
    claude: dict[str, Any] = Field(default_factory=dict)
    model_config = {"extra": "allow"}


class Config(BaseModel):
    llm: LlmConfig = Field(default_factory=LlmConfig)
    providers: ProvidersConfig = Field(default_factory=ProvidersConfig)
    incident: IncidentConfig = Field(default_factory=IncidentConfig)
    knowledge: KnowledgeConfig = Field(default_factory=KnowledgeConfig)
    safety: SafetyConfig = Field(default_factory=SafetyConfig)
    agent: AgentSection = Field(default_factory=AgentSection)
    integrations: IntegrationsConfig = Field(default_factory=IntegrationsConfig)
    model_config = {"extra": "allow"}


_ENV_RE = re.compile(r"\$\{([A-Za-z_][A-Za-z0-9_]*)\}")


def _interpolate_env(value: Any) -> Any:
    """${ENV} resolution (reference config.ts:253-270)."""
    if isinstance(value, str):
        return _ENV_RE.sub(lambda m: os.environ.get(m.group(1), ""), value)
    if isinstance(value, dict):
        return {k: _interpolate_env(v) for k, v in value.items()}
    if isinstance(value, list):
        return [_interpolate_env(v) for v in value]
    return value


def config_search_paths(cwd: Optional[str] = None) -> list[str]:
    """Search path (reference config.ts:222-230)."""
    cwd = cwd or os.getcwd()
    return [
        os.path.join(cwd, ".runbook", "config.yaml"),
        os.path.join(cwd, ".runbook", "config.yml"),
        os.path.expanduser("~/.runbook/config.yaml"),
    ]


def load_config(path: Optional[str] = None, cwd: Optional[str] = None) -> Config:
    candidates = [path] if path else config_search_paths(cwd)
    for p in candidates:
        if p and os.path.exists(p):
            with open(p, encoding="utf-8") as f:
                try:
                    raw = yaml.safe_load(f) or {}
                except yaml.YAMLError as e:
                    raise ValueError(
                        f"config file {p} is not valid YAML: "
                        f"{getattr(e, 'problem', e)}") from e
            if not isinstance(raw, dict):
                raise ValueError(f"config file {p} must be a YAML mapping")
            return Config.model_validate(_interpolate_env(raw))
    return Config()


def validate_config(config: Config) -> list[str]:
    """Per-provider consistency checks (reference config.ts:292-424).
    Returns a list of problems (empty = valid)."""
    problems: list[str] = []
    if config.llm.provider not in ("local", "mock"):
        problems.append(
            f"llm.provider '{config.llm.provider}' unsupported: models run locally on "
            "MI355X ('local') or scripted ('mock')")
    if config.llm.tensor_parallel not in (1, 2, 4, 8):
        problems.append("llm.tensorParallel must be 1, 2, 4 or 8 (one node of MI355X)")
    if config.llm.model not in ("llama3-8b", "llama3-70b", "tiny"):
        problems.append(f"llm.model '{config.llm.model}' unknown (llama3-8b / llama3-70b / tiny)")
    if config.llm.model == "llama3-70b" and config.llm.tensor_parallel < 4:
        problems.append("llama3-70b requires tensorParallel >= 4 (weights exceed one GPU at bf16"
                        " only with KV headroom at TP>=4; TP=8 recommended)")
    for i, src in enumerate(config.knowledge.sources):
        if src.get("kind", "filesystem") == "filesystem" and not src.get("path"):
            problems.append(f"knowledge.sources[{i}]: filesystem source needs 'path'")
    if config.safety.max_mutations_per_session < 0:
        problems.append("safety.maxMutationsPerSession must be >= 0")
    slack = config.incident.slack
    if slack.get("enabled") and not (slack.get("botToken") or slack.get("webhookPort")):
        problems.append("incident.slack enabled but neither botToken nor webhookPort set")
    return problems


def set_config_value(path: str, dotted_key: str, value: str) -> None:
    """Dotted config writes, e.g. 'llm.model=llama3-70b'
    (reference cli.tsx:1587-1664 `runbook config --set`)."""
    data: dict[str, Any] = {}
    if os.path.exists(path):
        with open(path, encoding="utf-8") as f:
            data = yaml.safe_load(f) or {}
    cur = data
    keys = dotted_key.split(".")
    for k in keys[:-1]:
        cur = cur.setdefault(k, {})
        if not isinstance(cur, dict):
            raise ValueError(f"cannot set {dotted_key}: {k} is not a mapping")
    # literal parsing: bool/int/float fall out of YAML
    cur[keys[-1]] = yaml.safe_load(value)
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    with open(path, "w", encoding="utf-8") as f:
        yaml.safe_dump(data, f, sort_keys=False)
