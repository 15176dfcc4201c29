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


def _dict(value) -> dict:
    """Junk-tolerant coercion: view/compare re-read summaries written by
    crashed runs, so every section access assumes nothing."""
    return value if isinstance(value, dict) else {}


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


_PHASE_PALETTE = {
    "input_wait_ms": ("#e07b39", "input"),
    "h2d_ms": ("#8e44ad", "h2d"),
    "forward_ms": ("#2d7dd2", "forward"),
    "backward_ms": ("#1b998b", "backward"),
    "optimizer_ms": ("#97cc04", "optimizer"),
    "ddp_comm_ms": ("#d05ce3", "ddp comm"),
    "residual_ms": ("#aaaaaa", "residual"),
}


def _per_rank_phase_bars(step_time_section: dict) -> str:
    """One stacked phase-share bar per rank: widths are
    each phase's share of THAT rank's step time, so a straggler's anomalous
    phase is visible at a glance across ranks."""
    rows = _dict(_dict(step_time_section.get("groups")).get("rows"))
    if len(rows) < 1:
        return ""
    total_w, bar_h, gap = 600, 18, 6
    ranks = sorted(rows, key=lambda k: (len(k), k))
    parts: List[str] = []
    y = 0
    for key in ranks:
        metrics = _dict(_dict(rows.get(key)).get("metrics"))
        step = metrics.get("step_time_ms")
        if not isinstance(step, (int, float)):
            step = None
        if not step:
            continue
        parts.append(
            f'<text x="0" y="{y + 13}" font-size="11">r{_esc(key)}</text>'
        )
        x = 30.0
        for metric, (color, label) in _PHASE_PALETTE.items():
            value = metrics.get(metric)
            if metric == "ddp_comm_ms":
                continue  # informational: overlaps backward, not additive
            if not isinstance(value, (int, float)) or value <= 0:
                continue
            w = min(1.0, value / step) * (total_w - 30)
            if w < 0.5:
                continue
            parts.append(
                f'<rect x="{x:.0f}" y="{y}" width="{w:.0f}" height="{bar_h}"'
                f' fill="{color}"><title>r{_esc(key)} {label}: {value:.1f} ms'
                f" ({value / step * 100.0:.1f}%)</title></rect>"
            )
            x += w
        y += bar_h + gap
    if not parts:
        return ""
    legend = " ".join(
        f'<tspan fill="{color}">■</tspan> {label}'
        for metric, (color, label) in _PHASE_PALETTE.items()
        if metric != "ddp_comm_ms"
    )
    height = y + 18
    return (
        "<h2>Per-rank phase shares</h2>"
        f'<svg width="{total_w}" height="{height}" '
        'xmlns="http://www.w3.org/2000/svg">'
        + "".join(parts)
        + f'<text x="0" y="{height - 4}" font-size="11">{legend}</text></svg>'
    )


_RANK_COLORS = ("#2d7dd2", "#1b998b", "#e07b39", "#d05ce3",
                "#97cc04", "#f0ad4e", "#e85d75", "#5bc0de")


def _memory_trend_svg(step_memory_section: dict) -> str:
    """Per-rank peak-allocated line chart over steps with the HBM capacity
    line — memory creep is visible as a rising slope."""
    evidence = _dict(step_memory_section.get("evidence_extra"))
    trend = _dict(evidence.get("trend"))
    series_by_rank = {
        rank: (_dict(t).get("series") or []) for rank, t in trend.items()
    }
    series_by_rank = {
        rank: [p for p in series if isinstance(p, (list, tuple))
               and len(p) == 2 and all(isinstance(v, (int, float)) for v in p)]
        for rank, series in series_by_rank.items()
        if isinstance(series, list)
    }
    all_points = [p for s in series_by_rank.values() for p in s]
    if len(all_points) < 4:
        return ""
    xs = [p[0] for p in all_points]
    ys = [p[1] for p in all_points]
    capacity = evidence.get("capacity_bytes")
    if not isinstance(capacity, (int, float)):
        capacity = None
    x0, x1 = min(xs), max(xs)
    y_top = max(max(ys) * 1.1, (capacity or 0) * 0.25)
    W, H = 600, 120
    gib = 1 << 30
    parts: List[str] = [
        f'<line x1="0" y1="{H}" x2="{W}" y2="{H}" stroke="#ccc"/>'
    ]
    if capacity and capacity <= y_top:
        cap_y = H - capacity / y_top * H
        parts.append(
            f'<line x1="0" y1="{cap_y:.0f}" x2="{W}" y2="{cap_y:.0f}" '
            'stroke="#d9534f" stroke-dasharray="4 3"/>'
            f'<text x="{W - 130}" y="{cap_y - 3:.0f}" font-size="10" '
            f'fill="#d9534f">capacity {capacity / gib:.0f} GiB</text>'
        )
    for i, (rank, series) in enumerate(sorted(series_by_rank.items())):
        if not series:
            continue
        color = _RANK_COLORS[i % len(_RANK_COLORS)]
        points = " ".join(
            f"{((s - x0) / max(1, x1 - x0) * W):.1f},"
            f"{(H - v / y_top * H):.1f}"
            for s, v in series
        )
        parts.append(
            f'<polyline points="{points}" fill="none" stroke="{color}" '
            'stroke-width="1.5"/>'
        )
        slope = _dict(trend.get(rank)).get("slope_bytes_per_step")
        label = f"r{rank}"
        if isinstance(slope, (int, float)) and slope > 1024:
            label += f" (+{slope / (1 << 20):.2f} MiB/step)"
        parts.append(
            f'<text x="{4 + i * 120}" y="12" font-size="10" '
            f'fill="{color}">{_esc(label)}</text>'
        )
    footer = (
        f"peak allocated 0–{y_top / gib:.1f} GiB over steps {x0}–{x1}"
    )
    if capacity:
        footer += f" · capacity {capacity / gib:.0f} GiB"
    parts.append(
        f'<text x="0" y="{H + 14}" font-size="10" fill="#888">{footer}</text>'
    )
    return (
        "<h2>Memory trend</h2>"
        f'<svg width="{W}" height="{H + 18}" '
        'xmlns="http://www.w3.org/2000/svg">' + "".join(parts) + "</svg>"
    )


def _comm_section(step_time_section: dict) -> str:
    """RCCL/xGMI rank-stats table from the step-time evidence (the new
    MI355X comm plane)."""
    comm = _dict(step_time_section.get("evidence_extra")).get(
        "rccl_rank_stats"
    )
    if not isinstance(comm, dict):
        return ""
    ranks = comm.get("ranks") or []
    if not ranks:
        return ""
    parts = ["<h2>RCCL rank stats (xGMI all-gather)</h2>"]
    latency = comm.get("gather_latency_ms")
    if isinstance(latency, (int, float)):
        parts.append(
            f"<p style='color:#666;font-size:0.85rem'>gather latency "
            f"{latency:.2f} ms over xGMI</p>"
        )
    header = (
        "<tr><th>rank</th><th>step</th><th>input ms</th><th>forward ms</th>"
        "<th>backward ms</th><th>optimizer ms</th><th>step ms</th>"
        "<th>ddp comm ms</th></tr>"
    )
    body = []
    for r in ranks:
        if not isinstance(r, dict):
            continue

        def cell(key):
            v = r.get(key)
            return f"{v:.1f}" if isinstance(v, (int, float)) else "—"

        body.append(
            f"<tr><td>r{_esc(r.get('rank'))}</td>"
            f"<td>{_esc(int(r['step']) if r.get('step') is not None else '—')}</td>"
            f"<td>{cell('input_ms')}</td><td>{cell('forward_ms')}</td>"
            f"<td>{cell('backward_ms')}</td><td>{cell('optimizer_ms')}</td>"
            f"<td>{cell('step_ms')}</td><td>{cell('ddp_comm_ms')}</td></tr>"
        )
    parts.append(f"<table>{header}{''.join(body)}</table>")
    return "".join(parts)


def _rank_table(section: dict) -> str:
    rows = _dict(_dict(section.get("groups")).get("rows"))
    metrics = _dict(section.get("metadata")).get("section_metric_names") or []
    if not rows or not metrics:
        return ""
    header = "<tr><th>rank</th>" + "".join(f"<th>{_esc(m)}</th>" for m in metrics) + "</tr>"
    body = []
    for key in sorted(rows, key=lambda k: (len(k), k)):
        cells = _dict(_dict(rows.get(key)).get("metrics"))
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
    primary = _dict(payload.get("primary_diagnosis"))
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

    shares = _dict(
        _dict(_dict(payload.get("step_time")).get("evidence_extra")).get("shares")
    )
    svg = _phase_bar_svg(shares)
    if svg:
        parts.append("<h2>Step-time phase breakdown</h2>" + svg)
    parts.append(_per_rank_phase_bars(_dict(payload.get("step_time"))))
    parts.append(_memory_trend_svg(_dict(payload.get("step_memory"))))
    parts.append(_comm_section(_dict(payload.get("step_time"))))

    for name in ("step_time", "step_memory", "system", "process"):
        section = _dict(payload.get(name))
        if not section:
            continue
        diag = _dict(section.get("diagnosis"))
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
