"""Self-contained single-file HTML report with hand-rolled SVG charts
(reference: reporting/html/*, 893 LoC — document/sections/svg/style)."""

from __future__ import annotations

import html as _html
from typing import List, Optional

_SEVERITY_COLORS = {"crit": "#d9534f", "warn": "#f0ad4e", "info": "#5bc0de"}

_CSS = """
body { font-family: -apple-system, 'Segoe UI', Roboto, sans-serif;
       margin: 2rem auto; max-width: 960px; color: #222; background: #fafafa; }
h1 { font-size: 1.4rem; } h2 { font-size: 1.1rem; margin-top: 2rem; }
.verdict { padding: 1rem; border-radius: 8px; color: #fff; font-weight: 600; }
.card { background: #fff; border: 1px solid #e0e0e0; border-radius: 8px;
        padding: 1rem; margin: 1rem 0; white-space: pre-wrap;
        font-family: ui-monospace, monospace; font-size: 0.85rem; }
table { border-collapse: collapse; width: 100%; font-size: 0.85rem; }
th, td { border: 1px solid #e0e0e0; padding: 4px 8px; text-align: right; }
th:first-child, td:first-child { text-align: left; }
.badge { display: inline-block; padding: 2px 8px; border-radius: 10px;
         color: #fff; font-size: 0.75rem; margin-left: 0.5rem; }
"""


def _esc(value) -> str:
    return _html.escape(str(value if value is not None else ""))


def _phase_bar_svg(shares: dict) -> str:
    """Horizontal stacked bar of phase shares of step time."""
    palette = {
        "input": "#e07b39",
        "h2d": "#8e44ad",
        "forward": "#2d7dd2",
        "backward": "#1b998b",
        "optimizer": "#97cc04",
        "residual": "#aaaaaa",
    }
    total_w = 600
    x = 0.0
    parts: List[str] = []
    for phase, color in palette.items():
        share = shares.get(phase)
        if share is None or share <= 0:
            continue
        w = max(1.0, min(1.0, share) * total_w)
        parts.append(
            f'<rect x="{x:.0f}" y="0" width="{w:.0f}" height="28" fill="{color}">'
            f"<title>{phase}: {share * 100.0:.1f}%</title></rect>"
        )
        x += w
    if not parts:
        return ""
    legend = " ".join(
        f'<tspan fill="{c}">■</tspan> {p} {shares.get(p, 0) * 100.0:.0f}% '
        for p, c in palette.items()
        if shares.get(p)
    )
    return (
        f'<svg width="{total_w}" height="52" xmlns="http://www.w3.org/2000/svg">'
        + "".join(parts)
        + f'<text x="0" y="46" font-size="11">{legend}</text></svg>'
    )


def _rank_table(section: dict) -> str:
    rows = section.get("groups", {}).get("rows", {})
    metrics = section.get("metadata", {}).get("section_metric_names", [])
    if not rows or not metrics:
        return ""
    header = "<tr><th>rank</th>" + "".join(f"<th>{_esc(m)}</th>" for m in metrics) + "</tr>"
    body = []
    for key in sorted(rows, key=lambda k: (len(k), k)):
        cells = rows[key].get("metrics", {})
        body.append(
            f"<tr><td>{_esc(key)}</td>"
            + "".join(
                "<td>{}</td>".format(
                    f"{cells[m]:.1f}" if isinstance(cells.get(m), float) else _esc(cells.get(m, ""))
                )
                for m in metrics
            )
            + "</tr>"
        )
    return f"<table>{header}{''.join(body)}</table>"


def render_html(payload: dict) -> str:
    primary = payload.get("primary_diagnosis", {})
    severity = primary.get("severity", "info")
    color = _SEVERITY_COLORS.get(severity, "#5bc0de")
    parts = [
        "<!DOCTYPE html><html><head><meta charset='utf-8'>",
        "<title>TraceML-AMD Final Summary</title>",
        f"<style>{_CSS}</style></head><body>",
        "<h1>TraceML-AMD Final Summary"
        f"<span class='badge' style='background:{color}'>{_esc(severity)}</span></h1>",
        f"<div class='verdict' style='background:{color}'>"
        f"{_esc(primary.get('status'))}: {_esc(primary.get('summary'))}</div>",
    ]
    action = primary.get("action")
    if action:
        parts.append(f"<p><b>Next:</b> {_esc(action)}</p>")

    shares = (
        payload.get("step_time", {}).get("evidence_extra", {}).get("shares") or {}
    )
    svg = _phase_bar_svg(shares)
    if svg:
        parts.append("<h2>Step-time phase breakdown</h2>" + svg)

    for name in ("step_time", "step_memory", "system", "process"):
        section = payload.get(name)
        if not section:
            continue
        diag = section.get("diagnosis", {})
        sev = diag.get("severity", "info")
        parts.append(
            f"<h2>{_esc(name)}<span class='badge' "
            f"style='background:{_SEVERITY_COLORS.get(sev, '#5bc0de')}'>"
            f"{_esc(diag.get('status'))}</span></h2>"
        )
        card = section.get("card")
        if card:
            parts.append(f"<div class='card'>{_esc(card)}</div>")
        table = _rank_table(section)
        if table:
            parts.append(table)

    parts.append(
        f"<p style='color:#888;font-size:0.75rem'>generated_at "
        f"{_esc(payload.get('generated_at'))} · schema "
        f"{_esc(payload.get('schema_version'))}</p></body></html>"
    )
    return "".join(parts)
