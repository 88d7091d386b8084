"""Persistent investigation note-taking across iterations.

Parity with reference src/agent/investigation-memory.ts (643 LoC):
InvestigationNote/InvestigationState (L29-89); regex finding extraction
from thinking text extract_from_thinking (L396-451); discovered-services
tracking (L453-462); build_context_summary/build_final_summary (L523-624);
persistence under the scratchpad dir (L198-230); drives knowledge re-query
on new services/symptoms (L633-642).
"""
from __future__ import annotations

import json
import os
import re
from dataclasses import dataclass, field
from typing import Any, Optional

from .types import new_id, now_ms


@dataclass
class InvestigationNote:
    id: str
    kind: str  # finding | symptom | service | decision | question
    text: str
    source: str = ""
    timestamp: int = field(default_factory=now_ms)

    def to_dict(self) -> dict[str, Any]:
        return {"id": self.id, "kind": self.kind, "text": self.text, "source": self.source,
                "timestamp": self.timestamp}


_FINDING_PATTERNS = [
    (re.compile(r"(?:found|observed|detected|confirmed|shows?|indicates?)\s+(?:that\s+)?(.{10,140}?)(?:\.|$)", re.IGNORECASE), "finding"),
    (re.compile(r"(?:symptom|experiencing|seeing)\s*[:\-]?\s*(.{8,120}?)(?:\.|$)", re.IGNORECASE), "symptom"),
    (re.compile(r"(?:error|exception|timeout|alarm)s?\s+(?:in|from|on)\s+([\w\-\.]{3,40})", re.IGNORECASE), "service"),
    (re.compile(r"(?:next|should|need to|will)\s+(?:check|investigate|query|look at)\s+(.{5,100}?)(?:\.|$)", re.IGNORECASE), "question"),
]

_SERVICE_RE = re.compile(r"\b([a-z][a-z0-9]*(?:-[a-z0-9]+)+)\b")  # kebab-case names


class InvestigationMemory:
    def __init__(self, session_id: str, directory: Optional[str] = None) -> None:
        self.session_id = session_id
        self.directory = directory
        self.notes: list[InvestigationNote] = []
        self.discovered_services: list[str] = []
        self.symptoms: list[str] = []
        self._new_services: list[str] = []
        self._new_symptoms: list[str] = []
        self._path = os.path.join(directory, f"{session_id}.memory.json") if directory else None

    # -- persistence (reference L198-230) ------------------------------------

    def init(self) -> None:
        if self._path and os.path.exists(self._path):
            try:
                with open(self._path, encoding="utf-8") as f:
                    data = json.load(f)
                self.notes = [
                    InvestigationNote(id=n["id"], kind=n["kind"], text=n["text"],
                                      source=n.get("source", ""), timestamp=n.get("timestamp", 0))
                    for n in data.get("notes", [])
                ]
                self.discovered_services = data.get("services", [])
                self.symptoms = data.get("symptoms", [])
            except (json.JSONDecodeError, KeyError, OSError):
                pass

    def save(self) -> None:
        if not self._path:
            return
        os.makedirs(os.path.dirname(self._path), exist_ok=True)
        with open(self._path, "w", encoding="utf-8") as f:
            json.dump(
                {
                    "sessionId": self.session_id,
                    "notes": [n.to_dict() for n in self.notes],
                    "services": self.discovered_services,
                    "symptoms": self.symptoms,
                },
                f,
                indent=1,
            )

    # -- note taking ---------------------------------------------------------

    def add_note(self, kind: str, text: str, source: str = "") -> InvestigationNote:
        text = text.strip()
        for n in self.notes:
            if n.kind == kind and n.text == text:
                return n
        note = InvestigationNote(id=new_id("note-"), kind=kind, text=text, source=source)
        self.notes.append(note)
        return note

    def extract_from_thinking(self, thinking: str, source: str = "thinking") -> list[InvestigationNote]:
        """Regex-based finding extraction (reference L396-451)."""
        added: list[InvestigationNote] = []
        for pattern, kind in _FINDING_PATTERNS:
            for m in pattern.finditer(thinking or ""):
                text = m.group(1).strip().rstrip(",;:")
                if len(text) >= 5:
                    before = len(self.notes)
                    note = self.add_note(kind, text, source)
                    if len(self.notes) > before:
                        added.append(note)
        # service discovery (reference L453-462)
        for m in _SERVICE_RE.finditer(thinking or ""):
            self.track_service(m.group(1))
        return added

    def track_service(self, service: str) -> bool:
        service = service.strip().lower()
        blocklist = {"e-g", "i-e", "p99", "p95", "x-ray", "so-called", "re-query", "re-read",
                     "well-known", "long-running", "built-in"}
        if not service or service in blocklist or len(service) < 4:
            return False
        if service not in self.discovered_services:
            self.discovered_services.append(service)
            self._new_services.append(service)
            return True
        return False

    def track_symptom(self, symptom: str) -> bool:
        symptom = symptom.strip()
        if symptom and symptom not in self.symptoms:
            self.symptoms.append(symptom)
            self._new_symptoms.append(symptom)
            return True
        return False

    def drain_new_discoveries(self) -> tuple[list[str], list[str]]:
        """Returns (new_services, new_symptoms) since last call — used for
        just-in-time knowledge re-query (reference L633-642)."""
        s, y = self._new_services, self._new_symptoms
        self._new_services, self._new_symptoms = [], []
        return s, y

    # -- summaries (reference L523-624) --------------------------------------

    def build_context_summary(self, max_notes: int = 12) -> str:
        if not self.notes and not self.discovered_services:
            return ""
        lines = ["## Investigation memory"]
        by_kind: dict[str, list[InvestigationNote]] = {}
        for n in self.notes[-max_notes:]:
            by_kind.setdefault(n.kind, []).append(n)
        for kind in ("finding", "symptom", "decision", "question"):
            items = by_kind.get(kind, [])
            if items:
                lines.append(f"**{kind.title()}s:**")
                lines.extend(f"- {n.text}" for n in items)
        if self.discovered_services:
            lines.append("**Services seen:** " + ", ".join(self.discovered_services[:10]))
        return "\n".join(lines)

    def build_final_summary(self) -> str:
        findings = [n for n in self.notes if n.kind == "finding"]
        if not findings and not self.symptoms:
            return ""
        lines = ["## Investigation summary"]
        if self.symptoms:
            lines.append("**Symptoms:** " + "; ".join(self.symptoms[:8]))
        if findings:
            lines.append("**Key findings:**")
            lines.extend(f"- {n.text}" for n in findings[:10])
        if self.discovered_services:
            lines.append("**Services involved:** " + ", ".join(self.discovered_services[:10]))
        return "\n".join(lines)
