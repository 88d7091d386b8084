"""Learning loop: investigation result -> postmortem + knowledge updates.

Parity with reference src/learning/loop.ts (703 LoC): single-prompt
generation of {postmortem, knowledgeSuggestions[]} (strict JSON schema in
prompt L233-294); fallback draft on parse failure (L296-339); postmortem
markdown with YAML frontmatter (L351-433); runbook-match scoring
(tokenized title/service overlap L435-460); apply vs propose updates into
.runbook/runbooks / learning/<id>/proposals; artifacts dir
.runbook/learning/<investigation-id>/ (L638) with postmortem-*.md,
knowledge-suggestions.json, investigation-result.json (L622-624).
"""
from __future__ import annotations

import json
import os
import re
import time
from typing import Any, Optional

from ..agent.llm_parser import ParseError, parse_json

LEARNING_PROMPT = """You are an SRE writing the learning artifacts for a completed incident investigation.

Investigation result:
{result}

Respond with ONLY a JSON object:
{{"postmortem": {{"title": "...", "summary": "...", "timeline": ["..."],
  "rootCause": "...", "impact": "...", "actionItems": ["..."]}},
 "knowledgeSuggestions": [{{"kind": "update_runbook|new_runbook|new_known_issue",
  "title": "...", "targetRunbook": "...", "content": "...", "services": ["..."]}}]}}"""

#: JSON schema for grammar-constrained decoding of the learning response.
LEARNING_JSON_SCHEMA: dict[str, Any] = {
    "type": "object",
    "properties": {
        "postmortem": {
            "type": "object",
            "properties": {
                "title": {"type": "string", "maxLength": 120},
                "summary": {"type": "string", "maxLength": 500},
                "timeline": {"type": "array", "items": {"type": "string", "maxLength": 120},
                             "maxItems": 8},
                "rootCause": {"type": "string", "maxLength": 300},
                "impact": {"type": "string", "maxLength": 200},
                "actionItems": {"type": "array", "items": {"type": "string", "maxLength": 150},
                                "maxItems": 6},
            },
            "required": ["title", "summary", "rootCause"],
        },
        "knowledgeSuggestions": {
            "type": "array",
            "maxItems": 4,
            "items": {
                "type": "object",
                "properties": {
                    "kind": {"enum": ["update_runbook", "new_runbook", "new_known_issue"]},
                    "title": {"type": "string", "maxLength": 120},
                    "content": {"type": "string", "maxLength": 600},
                    "services": {"type": "array", "items": {"type": "string", "maxLength": 40},
                                 "maxItems": 5},
                },
                "required": ["kind", "title", "content"],
            },
        },
    },
    "required": ["postmortem", "knowledgeSuggestions"],
}


def _tokenize(text: str) -> set[str]:
    return set(re.findall(r"[a-z0-9]{3,}", text.lower()))


def score_runbook_match(suggestion: dict[str, Any], runbook: dict[str, Any]) -> float:
    """Tokenized title/service overlap (reference L435-460)."""
    st = _tokenize(str(suggestion.get("title", "")) + " " + str(suggestion.get("targetRunbook", "")))
    rt = _tokenize(str(runbook.get("title", "")))
    title_overlap = len(st & rt) / max(1, len(st | rt))
    ss = {s.lower() for s in suggestion.get("services", [])}
    rs = {s.lower() for s in runbook.get("services", [])}
    svc_overlap = len(ss & rs) / max(1, len(ss | rs)) if (ss or rs) else 0.0
    return 0.6 * title_overlap + 0.4 * svc_overlap


def fallback_draft(result: dict[str, Any]) -> dict[str, Any]:
    """Template draft on parse failure (reference L296-339)."""
    root = result.get("rootCause", "unknown")
    return {
        "postmortem": {
            "title": f"Postmortem: {result.get('investigationId', 'investigation')}",
            "summary": result.get("summary", "")[:500] or f"Incident investigation concluded: {root}",
            "timeline": [],
            "rootCause": root,
            "impact": ", ".join(result.get("affectedServices", [])) or "unknown",
            "actionItems": ["Review this auto-drafted postmortem",
                            "Confirm the root cause with service owners"],
        },
        "knowledgeSuggestions": [],
        "fallback": True,
    }


def postmortem_markdown(postmortem: dict[str, Any], investigation_id: str) -> str:
    """Markdown with YAML frontmatter (reference L351-433)."""
    lines = [
        "---",
        f"title: \"{postmortem.get('title', 'Postmortem')}\"",
        "type: postmortem",
        f"investigation: {investigation_id}",
        f"generated: {time.strftime('%Y-%m-%d')}",
        "---",
        "",
        f"# {postmortem.get('title', 'Postmortem')}",
        "",
        "## Summary",
        postmortem.get("summary", ""),
        "",
        "## Root cause",
        postmortem.get("rootCause", ""),
    ]
    if postmortem.get("impact"):
        lines += ["", "## Impact", postmortem["impact"]]
    if postmortem.get("timeline"):
        lines += ["", "## Timeline"] + [f"- {t}" for t in postmortem["timeline"]]
    if postmortem.get("actionItems"):
        lines += ["", "## Action items"] + [f"- [ ] {a}" for a in postmortem["actionItems"]]
    return "\n".join(lines) + "\n"


class LearningLoop:
    def __init__(self, llm: Any, runbook_dir: str = ".runbook",
                 retriever: Any = None, apply_updates: bool = False) -> None:
        self.llm = llm
        self.runbook_dir = runbook_dir
        self.retriever = retriever
        self.apply_updates = apply_updates

    def run(self, result: dict[str, Any]) -> dict[str, Any]:
        investigation_id = result.get("investigationId", "unknown")
        prompt = LEARNING_PROMPT.format(result=json.dumps(result, indent=1, default=str)[:6000])
        try:
            data = parse_json(self.llm.complete(prompt))
            if not isinstance(data, dict) or "postmortem" not in data:
                raise ParseError("missing postmortem")
            data.setdefault("knowledgeSuggestions", [])
        except (ParseError, Exception):  # noqa: BLE001 — fallback draft
            data = fallback_draft(result)

        artifacts_dir = os.path.join(self.runbook_dir, "learning", investigation_id)
        os.makedirs(artifacts_dir, exist_ok=True)
        pm_md = postmortem_markdown(data["postmortem"], investigation_id)
        pm_path = os.path.join(artifacts_dir, f"postmortem-{investigation_id}.md")
        with open(pm_path, "w", encoding="utf-8") as f:
            f.write(pm_md)
        with open(os.path.join(artifacts_dir, "knowledge-suggestions.json"), "w",
                  encoding="utf-8") as f:
            json.dump(data.get("knowledgeSuggestions", []), f, indent=1)
        with open(os.path.join(artifacts_dir, "investigation-result.json"), "w",
                  encoding="utf-8") as f:
            json.dump(result, f, indent=1, default=str)

        applied, proposed = self._route_suggestions(
            data.get("knowledgeSuggestions", []), artifacts_dir)
        # refresh the knowledge base so the postmortem itself is retrievable
        if self.retriever is not None:
            try:
                self.retriever.sync()
            except Exception:  # noqa: BLE001
                pass
        return {
            "postmortemPath": pm_path,
            "postmortem": data["postmortem"],
            "suggestions": data.get("knowledgeSuggestions", []),
            "applied": applied,
            "proposed": proposed,
            "artifactsDir": artifacts_dir,
            "fallback": bool(data.get("fallback")),
        }

    def _route_suggestions(self, suggestions: list[dict[str, Any]],
                           artifacts_dir: str) -> tuple[list[str], list[str]]:
        applied: list[str] = []
        proposed: list[str] = []
        runbooks_dir = os.path.join(self.runbook_dir, "runbooks")
        proposals_dir = os.path.join(artifacts_dir, "proposals")
        existing: list[dict[str, Any]] = []
        if self.retriever is not None:
            try:
                existing = self.retriever.store.list_documents(doc_type="runbook")
            except Exception:  # noqa: BLE001
                existing = []
        for i, s in enumerate(suggestions):
            fname = re.sub(r"[^a-z0-9]+", "-", str(s.get("title", f"suggestion-{i}")).lower()).strip("-")
            body = f"# {s.get('title', '')}\n\n{s.get('content', '')}\n"
            if self.apply_updates and s.get("kind") in ("new_runbook", "update_runbook"):
                target_dir = runbooks_dir
                os.makedirs(target_dir, exist_ok=True)
                if s.get("kind") == "update_runbook" and existing:
                    best = max(existing, key=lambda r: score_runbook_match(s, r))
                    if score_runbook_match(s, best) > 0.3 and best.get("path"):
                        with open(best["path"], "a", encoding="utf-8") as f:
                            f.write(f"\n\n## Learned update ({time.strftime('%Y-%m-%d')})\n"
                                    f"{s.get('content', '')}\n")
                        applied.append(best["path"])
                        continue
                path = os.path.join(target_dir, f"{fname}.md")
                with open(path, "w", encoding="utf-8") as f:
                    f.write(body)
                applied.append(path)
            else:
                os.makedirs(proposals_dir, exist_ok=True)
                path = os.path.join(proposals_dir, f"{fname}.md")
                with open(path, "w", encoding="utf-8") as f:
                    f.write(body)
                proposed.append(path)
        return applied, proposed


def run_learning_loop(llm: Any, result: dict[str, Any], runbook_dir: str = ".runbook",
                      retriever: Any = None, apply_updates: bool = False) -> dict[str, Any]:
    return LearningLoop(llm, runbook_dir, retriever, apply_updates).run(result)
