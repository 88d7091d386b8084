"""ASCII chart rendering: line/bar/sparkline/gauge/histogram.

Parity with reference src/tools/diagram/charts.ts (331 LoC, asciichart).
"""
from __future__ import annotations

from typing import Any, Optional

SPARK_CHARS = "▁▂▃▄▅▆▇█"


def sparkline(values: list[float]) -> str:
    if not values:
        return ""
    lo, hi = min(values), max(values)
    span = (hi - lo) or 1.0
    return "".join(SPARK_CHARS[int((v - lo) / span * (len(SPARK_CHARS) - 1))] for v in values)


def line_chart(values: list[float], height: int = 8, width: Optional[int] = None,
               label: str = "") -> str:
    if not values:
        return "(no data)"
    if width and len(values) > width:
        step = len(values) / width
        values = [values[int(i * step)] for i in range(width)]
    lo, hi = min(values), max(values)
    span = (hi - lo) or 1.0
    rows = []
    for level in range(height - 1, -1, -1):
        threshold = lo + span * level / (height - 1 if height > 1 else 1)
        line = "".join("┤" if abs(v - threshold) <= span / (2 * height) else
                       ("│" if v > threshold else " ") for v in values)
        # simpler: mark cells at-or-above threshold in this row band
        line = ""
        for v in values:
            cell_level = (v - lo) / span * (height - 1)
            line += "●" if round(cell_level) == level else (" " if cell_level < level else "│")
        rows.append(f"{threshold:10.1f} ┤{line}")
    out = "\n".join(rows)
    if label:
        out = f"{label}\n{out}"
    return out


def bar_chart(items: list[tuple[str, float]], width: int = 40) -> str:
    if not items:
        return "(no data)"
    hi = max(v for _, v in items) or 1.0
    label_w = max(len(k) for k, _ in items)
    lines = []
    for k, v in items:
        bar = "█" * max(1, int(v / hi * width)) if v > 0 else ""
        lines.append(f"{k.ljust(label_w)} │{bar} {v:g}")
    return "\n".join(lines)


def gauge(value: float, lo: float = 0.0, hi: float = 100.0, width: int = 30,
          label: str = "") -> str:
    frac = 0.0 if hi <= lo else max(0.0, min(1.0, (value - lo) / (hi - lo)))
    filled = int(frac * width)
    return f"{label}[{'█' * filled}{'░' * (width - filled)}] {value:g} ({frac:.0%})"


def histogram(values: list[float], bins: int = 8, width: int = 30) -> str:
    if not values:
        return "(no data)"
    lo, hi = min(values), max(values)
    span = (hi - lo) or 1.0
    counts = [0] * bins
    for v in values:
        idx = min(bins - 1, int((v - lo) / span * bins))
        counts[idx] += 1
    peak = max(counts) or 1
    lines = []
    for i, c in enumerate(counts):
        lower = lo + span * i / bins
        bar = "█" * max(0, int(c / peak * width))
        lines.append(f"{lower:10.1f} │{bar} {c}")
    return "\n".join(lines)


def visualize(kind: str, data: Any, **opts: Any) -> str:
    """Dispatch for the visualize_metrics tool."""
    if kind == "sparkline":
        return sparkline(list(map(float, data)))
    if kind == "line":
        return line_chart(list(map(float, data)), label=opts.get("label", ""))
    if kind == "bar":
        if isinstance(data, dict):
            items = [(str(k), float(v)) for k, v in data.items()]
        else:
            items = [(str(k), float(v)) for k, v in data]
        return bar_chart(items)
    if kind == "gauge":
        return gauge(float(data), lo=float(opts.get("min", 0)), hi=float(opts.get("max", 100)),
                     label=opts.get("label", ""))
    if kind == "histogram":
        return histogram(list(map(float, data)))
    raise ValueError(f"unknown chart kind '{kind}'")
