"""Prompt builders for the free-form agent loop.

Parity with reference src/agent/prompts.ts (729 LoC): build_system_prompt
with tool/skill lists, region context, methodology, tool-usage policy,
visualization policy, safety rules (L37-223); iteration prompt (L228-266),
knowledge prompt (L271-322), final-answer prompt (L349-393);
context-engineering sections (L402-519); context-aware variants (L524-648).
"""
from __future__ import annotations

from typing import Any, Optional

METHODOLOGY = """## Investigation methodology
1. Triage: understand symptoms, timeline, affected services.
2. Hypothesize: form 1-5 testable root-cause hypotheses, ranked.
3. Investigate: run targeted queries per hypothesis (logs, metrics, infra state).
4. Evaluate: confirm, prune, or branch hypotheses on the evidence.
5. Conclude: state the root cause with confidence and cite evidence.
6. Remediate: propose safe, ordered steps; risky steps need approval."""

TOOL_POLICY = """## Tool usage policy
- Prefer narrow, filtered queries over broad scans.
- Do not repeat a call with identical arguments; drill into existing results
  with get_full_result(resultId) instead.
- Respect soft budgets per tool; when warned, change approach.
- Mutating operations ALWAYS go through the approval flow."""

VISUALIZATION_POLICY = """## Visualization policy
When you present metric trends or architectures in a final answer, include an
ASCII chart (visualize_metrics) or diagram (generate_flowchart /
generate_architecture_diagram) where it clarifies the story."""

SAFETY_RULES = """## Safety rules
- Never execute destructive operations (delete/terminate/purge) without explicit
  approval.
- Never disable alarms or monitoring to silence an incident.
- State uncertainty honestly; prefer "inconclusive" to a fabricated root cause."""


def build_system_prompt(
    tools: Optional[list[dict[str, Any]]] = None,
    skills: Optional[list[str]] = None,
    region: str = "",
    extra_sections: Optional[list[str]] = None,
) -> str:
    parts = [
        "You are Runbook, an expert SRE investigation agent. You diagnose production "
        "incidents hypothesis-first, using the available tools to gather evidence.",
    ]
    if region:
        parts.append(f"AWS region context: {region}")
    if tools:
        lines = ["## Available tools"]
        for t in tools:
            lines.append(f"- {t['name']}: {t.get('description', '')[:120]}")
        parts.append("\n".join(lines))
    if skills:
        parts.append("## Available skills\n" + "\n".join(f"- {s}" for s in skills))
    parts.extend([METHODOLOGY, TOOL_POLICY, VISUALIZATION_POLICY, SAFETY_RULES])
    parts.extend(extra_sections or [])
    return "\n\n".join(p for p in parts if p)


def build_iteration_prompt(
    query: str,
    iteration: int,
    max_iterations: int,
    tiered_context: str,
    memory_summary: str = "",
) -> str:
    parts = [
        f"Investigation query: {query}",
        f"(iteration {iteration}/{max_iterations})",
    ]
    if memory_summary:
        parts.append(memory_summary)
    parts.append("## Evidence so far\n" + tiered_context)
    parts.append(
        "Decide the next step: call tools to gather the evidence your current "
        "hypotheses need, or — if you can already answer — reply without tool calls."
    )
    return "\n\n".join(parts)


def build_knowledge_prompt(query: str, knowledge_markdown: str) -> str:
    return (
        f"Question: {query}\n\n"
        "Answer using ONLY the retrieved knowledge below. Cite sources by their "
        "bracketed numbers. If the knowledge does not answer the question, say so.\n\n"
        f"{knowledge_markdown}"
    )


def build_final_answer_prompt(
    query: str,
    tiered_context: str,
    memory_summary: str = "",
    hypothesis_markdown: str = "",
) -> str:
    parts = [
        f"Investigation query: {query}",
        "## Evidence collected\n" + tiered_context,
    ]
    if memory_summary:
        parts.append(memory_summary)
    if hypothesis_markdown:
        parts.append(hypothesis_markdown)
    parts.append(
        "Write the final answer: root cause (or best current explanation), the "
        "supporting evidence, affected services, confidence (low/medium/high) and "
        "recommended next steps. Be specific and cite evidence resultIds."
    )
    return "\n\n".join(parts)


# -- context-engineering sections (reference L402-519) ------------------------

def build_infra_section(infra_overview: str) -> str:
    return infra_overview or ""


def build_knowledge_availability_section(knowledge_section: str) -> str:
    return knowledge_section or ""


def build_status_section(
    iteration: int, tool_calls: int, services: list[str], active_hypotheses: int
) -> str:
    return (
        "## Investigation status\n"
        f"iteration {iteration} · {tool_calls} tool calls · "
        f"{active_hypotheses} active hypotheses · services: {', '.join(services[:8]) or '—'}"
    )


def build_context_aware_system_prompt(
    tools: Optional[list[dict[str, Any]]],
    skills: Optional[list[str]],
    infra_overview: str = "",
    knowledge_section: str = "",
    service_section: str = "",
    region: str = "",
) -> str:
    """Reference buildContextAwareSystemPrompt (prompts.ts:524-648)."""
    extra = [s for s in (infra_overview, knowledge_section, service_section) if s]
    return build_system_prompt(tools=tools, skills=skills, region=region, extra_sections=extra)
