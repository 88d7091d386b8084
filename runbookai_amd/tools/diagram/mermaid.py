"""Mermaid-to-ASCII rendering: flowcharts and sequence diagrams.

Parity with reference src/tools/diagram/mermaid.ts (548 LoC): parse a
subset of mermaid (graph TD/LR, sequenceDiagram) and render as ASCII.
"""
from __future__ import annotations

import re
from typing import Any


_EDGE_RE = re.compile(
    r"^\s*([\w\-\.]+)\s*(?:\[([^\]]*)\]|\(([^\)]*)\))?\s*-[-.]*>?\s*(?:\|([^|]*)\|)?\s*([\w\-\.]+)\s*(?:\[([^\]]*)\]|\(([^\)]*)\))?\s*$"
)
_SEQ_RE = re.compile(r"^\s*([\w\-\.]+)\s*(-{1,2}>>?)\s*([\w\-\.]+)\s*:\s*(.*)$")


def parse_flowchart(mermaid: str) -> tuple[dict[str, str], list[tuple[str, str, str]]]:
    """Returns (node_labels, edges[(src, dst, label)])."""
    nodes: dict[str, str] = {}
    edges: list[tuple[str, str, str]] = []
    for line in mermaid.split("\n"):
        line = line.strip()
        if not line or line.startswith(("graph", "flowchart", "%%", "subgraph", "end")):
            continue
        m = _EDGE_RE.match(line)
        if not m:
            continue
        src, src_l1, src_l2, edge_label, dst, dst_l1, dst_l2 = m.groups()
        nodes.setdefault(src, src_l1 or src_l2 or src)
        if src_l1 or src_l2:
            nodes[src] = src_l1 or src_l2
        nodes.setdefault(dst, dst_l1 or dst_l2 or dst)
        if dst_l1 or dst_l2:
            nodes[dst] = dst_l1 or dst_l2
        edges.append((src, dst, edge_label or ""))
    return nodes, edges


def render_flowchart(mermaid: str) -> str:
    nodes, edges = parse_flowchart(mermaid)
    if not edges:
        return "(empty diagram)"
    # topological-ish layering by BFS from roots
    children: dict[str, list[tuple[str, str]]] = {}
    indegree: dict[str, int] = {n: 0 for n in nodes}
    for s, d, lbl in edges:
        children.setdefault(s, []).append((d, lbl))
        indegree[d] = indegree.get(d, 0) + 1
    roots = [n for n, deg in indegree.items() if deg == 0] or [edges[0][0]]
    lines: list[str] = []
    seen: set[str] = set()

    def walk(node: str, depth: int, edge_label: str) -> None:
        pad = "    " * depth
        arrow = f"--{edge_label}-->" if edge_label else "-->" if depth else ""
        box = f"[{nodes.get(node, node)}]"
        lines.append(f"{pad}{arrow} {box}" if depth else f"{box}")
        if node in seen:
            return
        seen.add(node)
        for child, lbl in children.get(node, []):
            walk(child, depth + 1, lbl)

    for r in roots:
        walk(r, 0, "")
    return "\n".join(lines)


def render_sequence(mermaid: str) -> str:
    steps = []
    actors: list[str] = []
    for line in mermaid.split("\n"):
        m = _SEQ_RE.match(line.strip())
        if not m:
            continue
        src, arrow, dst, msg = m.groups()
        for a in (src, dst):
            if a not in actors:
                actors.append(a)
        steps.append((src, dst, msg, "-->>" in arrow or "-->" == arrow))
    if not steps:
        return "(empty sequence)"
    width = max(len(a) for a in actors) + 2
    header = " | ".join(a.center(width) for a in actors)
    lines = [header, "-" * len(header)]
    for src, dst, msg, dashed in steps:
        si, di = actors.index(src), actors.index(dst)
        lo, hi = min(si, di), max(si, di)
        row = []
        for i, _ in enumerate(actors):
            if i == si:
                row.append(("●" if si <= di else "◀").center(width))
            elif i == di:
                row.append(("▶" if si <= di else "●").center(width))
            elif lo < i < hi:
                row.append(("┈" if dashed else "─") * width)
            else:
                row.append(" " * width)
        lines.append(" | ".join(row) + f"  {msg}")
    return "\n".join(lines)


def render_mermaid(mermaid: str) -> str:
    head = mermaid.strip().split("\n", 1)[0].strip().lower()
    if head.startswith("sequencediagram"):
        return render_sequence(mermaid)
    return render_flowchart(mermaid)


def flowchart_from_spec(nodes: list[dict[str, Any]], edges: list[dict[str, Any]]) -> str:
    """Build + render from a structured spec (generate_flowchart tool)."""
    labels = {n["id"]: n.get("label", n["id"]) for n in nodes}
    lines = ["graph TD"]
    for e in edges:
        lbl = f"|{e['label']}|" if e.get("label") else ""
        src, dst = e["from"], e["to"]
        lines.append(f"    {src}[{labels.get(src, src)}] -->{lbl} {dst}[{labels.get(dst, dst)}]")
    return render_flowchart("\n".join(lines))
