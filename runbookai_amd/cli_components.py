"""Terminal rendering components.

Parity with reference src/cli/components/ (Ink): markdown (417 LoC),
hypothesis-tree (332), confidence-bar (148), table (133), code-block (106)
— re-implemented as ANSI string renderers for a plain terminal.
"""
from __future__ import annotations

import re
from typing import Any, Optional

BOLD = "\033[1m"
DIM = "\033[2m"
ITALIC = "\033[3m"
RESET = "\033[0m"
GREEN = "\033[32m"
YELLOW = "\033[33m"
RED = "\033[31m"
CYAN = "\033[36m"
MAGENTA = "\033[35m"


# -- markdown -----------------------------------------------------------------

_MD_BOLD = re.compile(r"\*\*(.+?)\*\*")
_MD_ITALIC = re.compile(r"(?<!\*)\*([^*]+)\*(?!\*)")
_MD_CODE = re.compile(r"`([^`]+)`")
_MD_HEADER = re.compile(r"^(#{1,4})\s+(.*)$")
_MD_BULLET = re.compile(r"^(\s*)[-*]\s+(.*)$")
_MD_NUM = re.compile(r"^(\s*)(\d+)\.\s+(.*)$")


def render_markdown(text: str, width: int = 100) -> str:
    """Markdown -> ANSI (headers, bold/italic/code, bullets, fences)."""
    out: list[str] = []
    in_fence = False
    fence_lines: list[str] = []
    for line in text.split("\n"):
        if line.strip().startswith("```"):
            if in_fence:
                out.append(render_code_block("\n".join(fence_lines)))
                fence_lines = []
            in_fence = not in_fence
            continue
        if in_fence:
            fence_lines.append(line)
            continue
        m = _MD_HEADER.match(line)
        if m:
            level, title = len(m.group(1)), m.group(2)
            deco = {1: f"{BOLD}{CYAN}", 2: BOLD, 3: f"{BOLD}{DIM}"}.get(level, DIM)
            out.append(f"{deco}{title}{RESET}")
            continue
        line = _MD_CODE.sub(f"{YELLOW}\\1{RESET}", line)
        line = _MD_BOLD.sub(f"{BOLD}\\1{RESET}", line)
        line = _MD_ITALIC.sub(f"{ITALIC}\\1{RESET}", line)
        mb = _MD_BULLET.match(line)
        if mb:
            line = f"{mb.group(1)}• {mb.group(2)}"
        mn = _MD_NUM.match(line)
        if mn:
            line = f"{mn.group(1)}{DIM}{mn.group(2)}.{RESET} {mn.group(3)}"
        out.append(line)
    if fence_lines:
        out.append(render_code_block("\n".join(fence_lines)))
    return "\n".join(out)


def render_code_block(code: str, lang: str = "") -> str:
    lines = code.split("\n")
    w = max((len(l) for l in lines), default=0)
    top = f"{DIM}┌{'─' * (w + 2)}┐{RESET}"
    bottom = f"{DIM}└{'─' * (w + 2)}┘{RESET}"
    body = "\n".join(f"{DIM}│{RESET} {YELLOW}{l.ljust(w)}{RESET} {DIM}│{RESET}" for l in lines)
    return f"{top}\n{body}\n{bottom}"


# -- table --------------------------------------------------------------------

def render_table(headers: list[str], rows: list[list[Any]], max_col: int = 40) -> str:
    def clip(v: Any) -> str:
        s = str(v)
        return s[: max_col - 1] + "…" if len(s) > max_col else s

    cells = [[clip(h) for h in headers]] + [[clip(c) for c in r] for r in rows]
    widths = [max(len(row[i]) for row in cells) for i in range(len(headers))]

    def fmt(row: list[str], deco: str = "") -> str:
        body = " │ ".join(c.ljust(w) for c, w in zip(row, widths))
        return f"{deco}{body}{RESET}" if deco else body

    sep = "─┼─".join("─" * w for w in widths)
    out = [fmt(cells[0], BOLD), sep]
    out.extend(fmt(r) for r in cells[1:])
    return "\n".join(out)


# -- confidence bar -----------------------------------------------------------

def render_confidence_bar(confidence: float, width: int = 20, label: str = "") -> str:
    confidence = max(0.0, min(1.0, confidence))
    filled = int(round(confidence * width))
    color = GREEN if confidence >= 0.7 else (YELLOW if confidence >= 0.4 else RED)
    bar = f"{color}{'█' * filled}{DIM}{'░' * (width - filled)}{RESET}"
    prefix = f"{label} " if label else ""
    return f"{prefix}{bar} {confidence:.0%}"


# -- hypothesis tree ----------------------------------------------------------

_STATUS_DECOR = {
    "active": (CYAN, "○"),
    "investigating": (CYAN, "◐"),
    "confirmed": (GREEN, "✓"),
    "pruned": (DIM, "✗"),
    "branched": (MAGENTA, "⑂"),
}


def render_hypothesis_tree(nodes: list[dict[str, Any]], indent: int = 0) -> str:
    """Renders HypothesisEngine.to_tree_data() output."""
    lines: list[str] = []
    for i, node in enumerate(nodes):
        color, badge = _STATUS_DECOR.get(node.get("status", "active"), (RESET, "·"))
        last = i == len(nodes) - 1
        branch = ("└─ " if last else "├─ ") if indent else ""
        pad = "   " * max(0, indent - 1) + branch
        conf = node.get("confidence", 0.0)
        lines.append(
            f"{pad}{color}{badge}{RESET} {node.get('label', '?')} "
            f"{DIM}({conf:.2f}){RESET}"
        )
        children = node.get("children", [])
        if children:
            lines.append(render_hypothesis_tree(children, indent + 1))
    return "\n".join(l for l in lines if l)
