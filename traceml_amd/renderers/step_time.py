"""Step-time section renderer (reference: renderers/step_time/renderer.py:86-273).

One view model for CLI / dashboard / HTML: verdict, ordered phase rows per
rank (raw + formatted, nullable preserved), phase shares for the stacked
bar, behavior cohorts, worst-vs-median skew of the visible phase, and the
history series for charts.
"""

from __future__ import annotations

from typing import Optional

from traceml_amd.renderers.base import fmt_ms

#: (metric, short label) in display order — the closed event vocabulary
PHASE_ROWS = (
    ("step_time_ms", "step (ms)"),
    ("input_wait_ms", "input"),
    ("h2d_ms", "h2d"),
    ("forward_ms", "forward"),
    ("backward_ms", "backward"),
    ("optimizer_ms", "optimizer"),
    ("ddp_comm_ms", "ddp comm"),
    ("compute_ms", "compute"),
    ("residual_ms", "residual"),
)

#: shares bar order (matches the analyzer's share phases)
SHARE_ORDER = ("input", "h2d", "forward", "backward", "optimizer", "residual")


def _skew(window) -> Optional[dict]:
    """Worst-vs-median step time across ranks (needs >=2 ranks)."""
    values = [
        (r, window.ranks[r].get("step_time_ms"))
        for r in window.ranks_used
        if window.ranks[r].get("step_time_ms") is not None
    ]
    if len(values) < 2:
        return None
    values.sort(key=lambda rv: rv[1])
    median = values[len(values) // 2][1]
    worst_rank, worst = values[-1]
    if not median:
        return None
    return {
        "worst_rank": worst_rank,
        "worst_ms": worst,
        "median_ms": median,
        "skew_fraction": (worst - median) / median,
    }


def render_step_time(window, diagnosis, max_history_points: int = 120) -> dict:
    """Pure view model over an analyzed window + its diagnosis result."""
    rows = []
    for metric, label in PHASE_ROWS:
        cells = {}
        any_value = False
        for rank in window.ranks_used:
            value = window.ranks[rank].get(metric)
            cells[str(rank)] = {"ms": value, "text": fmt_ms(value)}
            any_value = any_value or value is not None
        if any_value:
            rows.append({"metric": metric, "label": label, "cells": cells})

    shares = []
    for phase in SHARE_ORDER:
        value = (window.shares or {}).get(phase)
        if value is not None:
            shares.append({"phase": phase, "fraction": value})

    cohort_of = {}
    for cohort, ranks in (window.cohorts or {}).items():
        for rank in ranks:
            cohort_of[str(rank)] = cohort

    history = {
        str(rank): [[s, round(ms, 3)] for s, ms in series[-max_history_points:]]
        for rank, series in (window.step_series or {}).items()
    }

    primary = diagnosis.primary
    return {
        "section": "step_time",
        "available": window.has_data,
        "diagnosis": primary.to_payload(),
        "issues": [i.to_payload() for i in diagnosis.issues],
        "ranks": [str(r) for r in window.ranks_used],
        "rows": rows,
        "shares": shares,
        "cohorts": cohort_of,
        "skew": _skew(window),
        "history": history,
        "footer": {
            "steps_analyzed": window.steps_analyzed,
            "clock": window.clock,
            "strategy": window.training_strategy,
        },
    }
