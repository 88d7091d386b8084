"""Structured-output contract between the orchestrator and the LLM.

Parity with reference src/agent/llm-parser.ts (570 LoC): HypothesisSchema
(L21-47), HypothesisGenerationSchema 1-5 hypotheses (L54-57),
EvidenceEvaluationSchema with action branch|prune|confirm|continue +
optional subHypotheses (L64-81), TriageResponseSchema (L88-99),
ConclusionSchema (L106-133), RemediationPlanSchema (L140-181),
LogAnalysisSchema (L188-208), extractJSON code-fence/brace extraction
(L215-229), PROMPTS six fill-in templates (L396-559), fillPrompt (L564-570).

MI355X-native addition: every schema also exists as a JSON-schema dict
(`*_JSON_SCHEMA`) consumed by the engine's grammar-constrained sampler
(runbookai_amd/engine/json_fsm.py) — since we own the decoder, schema
discipline is enforced at the logits level rather than hoped for.
"""
from __future__ import annotations

import json
import re
from typing import Any, Optional


class ParseError(Exception):
    pass


# ---------------------------------------------------------------------------
# JSON extraction (reference llm-parser.ts:215-229 + benchmark salvage L111-152)
# ---------------------------------------------------------------------------

_FENCE_RE = re.compile(r"```(?:json)?\s*(.*?)```", re.DOTALL)


def extract_json(text: str) -> str:
    """Pull a JSON object/array out of model output.

    Order: fenced ```json block -> first balanced {...} / [...] span -> raw.
    """
    if not text:
        raise ParseError("empty response")
    m = _FENCE_RE.search(text)
    if m:
        candidate = m.group(1).strip()
        if candidate:
            return candidate
    # first balanced brace span — whichever of { / [ appears first
    starts = [(text.find(c), c, close_c) for c, close_c in (("{", "}"), ("[", "]")) if text.find(c) != -1]
    starts.sort()
    for start, open_ch, close_ch in starts:
        depth = 0
        in_str = False
        esc = False
        for i in range(start, len(text)):
            ch = text[i]
            if in_str:
                if esc:
                    esc = False
                elif ch == "\\":
                    esc = True
                elif ch == '"':
                    in_str = False
                continue
            if ch == '"':
                in_str = True
            elif ch == open_ch:
                depth += 1
            elif ch == close_ch:
                depth -= 1
                if depth == 0:
                    return text[start : i + 1]
    return text.strip()


def _normalize_json_text(text: str) -> str:
    """Salvage common local-model JSON mistakes (reference
    investigation-benchmark.ts:111-152 normalizeJsonResponse):
    trailing commas, single quotes around keys, unquoted literals."""
    t = text.strip()
    t = re.sub(r",\s*([}\]])", r"\1", t)  # trailing commas
    return t


def parse_json(text: str) -> Any:
    # clean output parses as-is — salvage heuristics (fence stripping,
    # brace scanning) only run on failure, so valid JSON whose STRING
    # content contains ``` or braces is never mangled (schema fuzzing,
    # tests/test_llm_parser_fuzz.py)
    try:
        return json.loads(text.strip())
    except (json.JSONDecodeError, AttributeError):
        pass
    raw = extract_json(text)
    try:
        return json.loads(raw)
    except json.JSONDecodeError:
        try:
            return json.loads(_normalize_json_text(raw))
        except json.JSONDecodeError as e:
            raise ParseError(f"unparseable JSON: {e}: {raw[:200]}") from e


# ---------------------------------------------------------------------------
# Schema validators (lightweight; return normalized dicts)
# ---------------------------------------------------------------------------

_CONFIDENCE_LEVELS = ("low", "medium", "high")
_ACTIONS = ("branch", "prune", "confirm", "continue")
_RISKS = ("low", "medium", "high", "critical")


def _as_str_list(v: Any) -> list[str]:
    if isinstance(v, list):
        return [str(x) for x in v if x is not None]
    if isinstance(v, str) and v:
        return [v]
    return []


def _clamp01(v: Any, default: float = 0.5) -> float:
    try:
        return max(0.0, min(1.0, float(v)))
    except (TypeError, ValueError):
        return default


def validate_hypothesis(d: Any) -> dict[str, Any]:
    """Reference HypothesisSchema (llm-parser.ts:21-47)."""
    if not isinstance(d, dict):
        raise ParseError("hypothesis must be an object")
    statement = str(d.get("statement") or d.get("hypothesis") or "").strip()
    if not statement:
        raise ParseError("hypothesis missing statement")
    priority = d.get("priority", 3)
    try:
        priority = max(1, min(5, int(priority)))
    except (TypeError, ValueError):
        priority = 3
    return {
        "statement": statement,
        "rationale": str(d.get("rationale", "")),
        "priority": priority,
        "affectedServices": _as_str_list(d.get("affectedServices")),
        "suggestedQueries": d.get("suggestedQueries") if isinstance(d.get("suggestedQueries"), list) else [],
    }


def parse_hypothesis_generation(text: str) -> list[dict[str, Any]]:
    """Reference HypothesisGenerationSchema: 1-5 hypotheses (llm-parser.ts:54-57)."""
    data = parse_json(text)
    if isinstance(data, dict):
        items = data.get("hypotheses", [])
    elif isinstance(data, list):
        items = data
    else:
        raise ParseError("hypothesis generation: expected object or array")
    out: list[dict[str, Any]] = []
    for item in items:
        try:
            out.append(validate_hypothesis(item))
        except ParseError:
            continue
    if not out:
        raise ParseError("no valid hypotheses parsed")
    return out[:5]


def parse_evidence_evaluation(text: str) -> dict[str, Any]:
    """Reference EvidenceEvaluationSchema (llm-parser.ts:64-81)."""
    d = parse_json(text)
    if not isinstance(d, dict):
        raise ParseError("evaluation must be an object")
    action = str(d.get("action", "continue")).lower()
    if action not in _ACTIONS:
        action = "continue"
    evidence = []
    for ev in d.get("evidence", []) if isinstance(d.get("evidence"), list) else []:
        if isinstance(ev, dict):
            evidence.append(
                {
                    "description": str(ev.get("description", "")),
                    "supports": bool(ev.get("supports", True)),
                    "source": str(ev.get("source", "")),
                }
            )
        elif isinstance(ev, str):
            evidence.append({"description": ev, "supports": True, "source": ""})
    subs = []
    for sub in d.get("subHypotheses", []) if isinstance(d.get("subHypotheses"), list) else []:
        try:
            subs.append(validate_hypothesis(sub))
        except ParseError:
            continue
    return {
        "action": action,
        "confidence": _clamp01(d.get("confidence")),
        "reasoning": str(d.get("reasoning", "")),
        "evidence": evidence,
        "subHypotheses": subs,
    }


def parse_triage_response(text: str) -> dict[str, Any]:
    """Reference TriageResponseSchema (llm-parser.ts:88-99)."""
    d = parse_json(text)
    if not isinstance(d, dict):
        raise ParseError("triage must be an object")
    severity = str(d.get("severity", "medium")).lower()
    if severity not in ("low", "medium", "high", "critical"):
        severity = "medium"
    initial = []
    for h in d.get("initialHypotheses", []) if isinstance(d.get("initialHypotheses"), list) else []:
        try:
            initial.append(validate_hypothesis(h))
        except ParseError:
            continue
    return {
        "summary": str(d.get("summary", "")),
        "symptoms": _as_str_list(d.get("symptoms")),
        "affectedServices": _as_str_list(d.get("affectedServices")),
        "severity": severity,
        "timeline": str(d.get("timeline", "")),
        "initialHypotheses": initial,
    }


def parse_conclusion(text: str) -> dict[str, Any]:
    """Reference ConclusionSchema (llm-parser.ts:106-133)."""
    d = parse_json(text)
    if not isinstance(d, dict):
        raise ParseError("conclusion must be an object")
    root_cause = str(d.get("rootCause") or d.get("root_cause") or "").strip()
    if not root_cause:
        raise ParseError("conclusion missing rootCause")
    confidence = str(d.get("confidence", "medium")).lower()
    if confidence not in _CONFIDENCE_LEVELS:
        confidence = "medium"
    return {
        "rootCause": root_cause,
        "confidence": confidence,
        "summary": str(d.get("summary", "")),
        "affectedServices": _as_str_list(d.get("affectedServices")),
        "evidence": _as_str_list(d.get("evidence")),
        "contributingFactors": _as_str_list(d.get("contributingFactors")),
    }


def parse_remediation_plan(text: str) -> dict[str, Any]:
    """Reference RemediationPlanSchema (llm-parser.ts:140-181)."""
    d = parse_json(text)
    if not isinstance(d, dict):
        raise ParseError("remediation must be an object")
    steps = []
    for s in d.get("steps", []) if isinstance(d.get("steps"), list) else []:
        if isinstance(s, str):
            steps.append(
                {"description": s, "tool": None, "params": {}, "command": None,
                 "risk": "low", "requiresApproval": False, "matchingSkill": None}
            )
            continue
        if not isinstance(s, dict):
            continue
        risk = str(s.get("risk", "low")).lower()
        if risk not in _RISKS:
            risk = "low"
        steps.append(
            {
                "description": str(s.get("description", "")),
                "tool": s.get("tool"),
                "params": s.get("params") if isinstance(s.get("params"), dict) else {},
                "command": s.get("command"),
                "risk": risk,
                "requiresApproval": bool(s.get("requiresApproval", risk in ("high", "critical"))),
                "matchingSkill": s.get("matchingSkill"),
            }
        )
    return {
        "summary": str(d.get("summary", "")),
        "steps": steps,
        "rollback": str(d.get("rollback", "")),
        "matchingSkill": d.get("matchingSkill"),
    }


def parse_log_analysis(text: str) -> dict[str, Any]:
    """Reference LogAnalysisSchema (llm-parser.ts:188-208)."""
    d = parse_json(text)
    if not isinstance(d, dict):
        raise ParseError("log analysis must be an object")
    patterns = []
    for p in d.get("patterns", []) if isinstance(d.get("patterns"), list) else []:
        if isinstance(p, dict):
            patterns.append(
                {
                    "pattern": str(p.get("pattern", "")),
                    "count": int(p.get("count", 0) or 0),
                    "severity": str(p.get("severity", "info")),
                    "sample": str(p.get("sample", "")),
                }
            )
    return {
        "summary": str(d.get("summary", "")),
        "patterns": patterns,
        "services": _as_str_list(d.get("services")),
        "suggestedHypotheses": _as_str_list(d.get("suggestedHypotheses")),
    }


# ---------------------------------------------------------------------------
# JSON schemas for grammar-constrained decoding (engine/json_fsm.py)
# ---------------------------------------------------------------------------

_HYPOTHESIS_ITEM_SCHEMA: dict[str, Any] = {
    "type": "object",
    "properties": {
        "statement": {"type": "string", "maxLength": 200},
        "rationale": {"type": "string", "maxLength": 200},
        "priority": {"type": "integer", "minimum": 1, "maximum": 5},
        "affectedServices": {"type": "array", "items": {"type": "string", "maxLength": 40}, "maxItems": 4},
    },
    "required": ["statement", "rationale", "priority"],
}

HYPOTHESIS_GENERATION_JSON_SCHEMA: dict[str, Any] = {
    "type": "object",
    "properties": {
        "hypotheses": {"type": "array", "items": _HYPOTHESIS_ITEM_SCHEMA, "minItems": 1, "maxItems": 5},
    },
    "required": ["hypotheses"],
}

EVIDENCE_EVALUATION_JSON_SCHEMA: dict[str, Any] = {
    "type": "object",
    "properties": {
        "action": {"enum": ["branch", "prune", "confirm", "continue"]},
        "confidence": {"type": "number", "minimum": 0, "maximum": 1},
        "reasoning": {"type": "string", "maxLength": 300},
        "evidence": {
            "type": "array",
            "maxItems": 4,
            "items": {
                "type": "object",
                "properties": {
                    "description": {"type": "string", "maxLength": 200},
                    "supports": {"type": "boolean"},
                    "source": {"type": "string", "maxLength": 60},
                },
                "required": ["description", "supports"],
            },
        },
        "subHypotheses": {"type": "array", "items": _HYPOTHESIS_ITEM_SCHEMA, "maxItems": 3},
    },
    "required": ["action", "confidence", "reasoning"],
}

TRIAGE_JSON_SCHEMA: dict[str, Any] = {
    "type": "object",
    "properties": {
        "summary": {"type": "string", "maxLength": 400},
        "symptoms": {"type": "array", "items": {"type": "string", "maxLength": 80}, "maxItems": 6},
        "affectedServices": {"type": "array", "items": {"type": "string", "maxLength": 40}, "maxItems": 6},
        "severity": {"enum": ["low", "medium", "high", "critical"]},
        "timeline": {"type": "string", "maxLength": 120},
    },
    "required": ["summary", "symptoms", "affectedServices", "severity"],
}

CONCLUSION_JSON_SCHEMA: dict[str, Any] = {
    "type": "object",
    "properties": {
        "rootCause": {"type": "string", "maxLength": 300},
        "confidence": {"enum": ["low", "medium", "high"]},
        "summary": {"type": "string", "maxLength": 500},
        "affectedServices": {"type": "array", "items": {"type": "string", "maxLength": 40}, "maxItems": 6},
        "evidence": {"type": "array", "items": {"type": "string", "maxLength": 200}, "maxItems": 5},
        "contributingFactors": {"type": "array", "items": {"type": "string", "maxLength": 120}, "maxItems": 4},
    },
    "required": ["rootCause", "confidence", "summary"],
}

REMEDIATION_JSON_SCHEMA: dict[str, Any] = {
    "type": "object",
    "properties": {
        "summary": {"type": "string", "maxLength": 300},
        "steps": {
            "type": "array",
            "minItems": 1,
            "maxItems": 5,
            "items": {
                "type": "object",
                "properties": {
                    "description": {"type": "string", "maxLength": 200},
                    "command": {"type": "string", "maxLength": 120},
                    "risk": {"enum": ["low", "medium", "high", "critical"]},
                    "requiresApproval": {"type": "boolean"},
                },
                "required": ["description", "risk"],
            },
        },
        "rollback": {"type": "string", "maxLength": 200},
    },
    "required": ["summary", "steps"],
}

LOG_ANALYSIS_JSON_SCHEMA: dict[str, Any] = {
    "type": "object",
    "properties": {
        "summary": {"type": "string", "maxLength": 300},
        "patterns": {
            "type": "array",
            "maxItems": 6,
            "items": {
                "type": "object",
                "properties": {
                    "pattern": {"type": "string", "maxLength": 80},
                    "count": {"type": "integer", "minimum": 0},
                    "severity": {"enum": ["info", "warning", "error", "critical"]},
                },
                "required": ["pattern", "severity"],
            },
        },
        "services": {"type": "array", "items": {"type": "string", "maxLength": 40}, "maxItems": 6},
    },
    "required": ["summary", "patterns"],
}


# ---------------------------------------------------------------------------
# Prompt templates (reference llm-parser.ts:396-559)
# ---------------------------------------------------------------------------

PROMPTS: dict[str, str] = {
    "triage": (
        "You are an SRE triaging a production incident.\n\n"
        "Incident query: {query}\n\n"
        "Context gathered so far:\n{context}\n\n"
        "Summarize what is known. Respond with ONLY a JSON object:\n"
        '{{"summary": "...", "symptoms": ["..."], "affectedServices": ["..."], '
        '"severity": "low|medium|high|critical", "timeline": "..."}}'
    ),
    "generateHypotheses": (
        "You are an SRE generating root-cause hypotheses for an incident.\n\n"
        "Triage summary: {triage}\n"
        "Symptoms: {symptoms}\n"
        "Affected services: {services}\n"
        "Relevant knowledge:\n{knowledge}\n\n"
        "Generate 1-5 distinct, testable root-cause hypotheses ranked by likelihood.\n"
        "Respond with ONLY a JSON object:\n"
        '{{"hypotheses": [{{"statement": "...", "rationale": "...", "priority": 1, '
        '"affectedServices": ["..."]}}]}}'
    ),
    "evaluateEvidence": (
        "You are an SRE evaluating evidence for a hypothesis.\n\n"
        "Hypothesis: {hypothesis}\n"
        "Rationale: {rationale}\n\n"
        "Query results:\n{results}\n\n"
        "Decide: confirm (strong supporting evidence), prune (contradicted), "
        "branch (split into more specific sub-hypotheses), or continue (need more data).\n"
        "Respond with ONLY a JSON object:\n"
        '{{"action": "branch|prune|confirm|continue", "confidence": 0.0, "reasoning": "...", '
        '"evidence": [{{"description": "...", "supports": true, "source": "..."}}], '
        '"subHypotheses": [{{"statement": "...", "rationale": "...", "priority": 1}}]}}'
    ),
    "generateConclusion": (
        "You are an SRE writing the conclusion of an incident investigation.\n\n"
        "Investigation summary:\n{summary}\n\n"
        "Confirmed hypotheses:\n{confirmed}\n\n"
        "Evidence collected:\n{evidence}\n\n"
        "Respond with ONLY a JSON object:\n"
        '{{"rootCause": "...", "confidence": "low|medium|high", "summary": "...", '
        '"affectedServices": ["..."], "evidence": ["..."], "contributingFactors": ["..."]}}'
    ),
    "generateRemediation": (
        "You are an SRE planning remediation for a diagnosed incident.\n\n"
        "Root cause: {rootCause}\n"
        "Affected services: {services}\n"
        "Relevant runbooks:\n{runbooks}\n"
        "Available skills (pre-approved automation):\n{skills}\n"
        "Code-fix candidates:\n{codeFixes}\n\n"
        "Plan safe, ordered remediation steps. Mark risky steps requiresApproval.\n"
        "Respond with ONLY a JSON object:\n"
        '{{"summary": "...", "steps": [{{"description": "...", "command": "...", '
        '"risk": "low|medium|high|critical", "requiresApproval": false}}], "rollback": "..."}}'
    ),
    "analyzeLogs": (
        "You are an SRE analyzing application logs.\n\n"
        "Log lines:\n{logs}\n\n"
        "Pattern pre-analysis:\n{patterns}\n\n"
        "Respond with ONLY a JSON object:\n"
        '{{"summary": "...", "patterns": [{{"pattern": "...", "count": 0, '
        '"severity": "info|warning|error|critical"}}], "services": ["..."]}}'
    ),
}

#: Maps each prompt kind to the JSON schema its answer must satisfy — the
#: engine uses this for grammar-constrained decoding.
PROMPT_SCHEMAS: dict[str, dict[str, Any]] = {
    "triage": TRIAGE_JSON_SCHEMA,
    "generateHypotheses": HYPOTHESIS_GENERATION_JSON_SCHEMA,
    "evaluateEvidence": EVIDENCE_EVALUATION_JSON_SCHEMA,
    "generateConclusion": CONCLUSION_JSON_SCHEMA,
    "generateRemediation": REMEDIATION_JSON_SCHEMA,
    "analyzeLogs": LOG_ANALYSIS_JSON_SCHEMA,
}

#: Sentinel embedded into prompts so the engine knows which schema to
#: constrain decoding with (stripped before tokenization).
SCHEMA_TAG_PREFIX = "\x00schema:"


def fill_prompt(kind: str, tag_schema: bool = True, **fields: Any) -> str:
    """Fill one of the PROMPTS templates (reference fillPrompt L564-570).

    When tag_schema is set, prefixes an inline schema tag the local engine
    uses to pick the decode grammar; hosted/mock clients ignore it.
    """
    template = PROMPTS[kind]
    safe = {k: ("" if v is None else str(v)) for k, v in fields.items()}

    class _Default(dict):
        def __missing__(self, key: str) -> str:
            return ""

    body = template.format_map(_Default(**safe))
    if tag_schema and kind in PROMPT_SCHEMAS:
        return f"{SCHEMA_TAG_PREFIX}{kind}\x00{body}"
    return body


def split_schema_tag(prompt: str) -> tuple[Optional[str], str]:
    """Return (schema_kind, clean_prompt)."""
    if prompt.startswith(SCHEMA_TAG_PREFIX):
        rest = prompt[len(SCHEMA_TAG_PREFIX):]
        kind, _, body = rest.partition("\x00")
        return kind, body
    return None, prompt
