"""Step-time degradation trend (reference: diagnostics/step_time/trend.py:165).

Detects a sustained rise of per-step step time across the analyzed window
(e.g., a growing dataloader backlog, thermal throttling, or a memory-creep
side effect) and attaches it as an informational/warn issue — never the
primary verdict on its own."""

from __future__ import annotations

from typing import List, Optional

from traceml_amd.diagnostics.common import DiagnosticIssue
from traceml_amd.diagnostics.trends import fit_trend, split_halves_means

#: relative rise over the window to call it a degradation
TREND_WARN_RELATIVE = 0.25
TREND_MIN_STEPS = 30


def step_time_trend_issue(
    steps: List[int], step_ms_series: List[float]
) -> Optional[DiagnosticIssue]:
    if len(steps) < TREND_MIN_STEPS:
        return None
    trend = fit_trend([float(s) for s in steps], step_ms_series)
    if trend is None or trend.direction != "rising":
        return None
    if trend.relative_delta is None or trend.relative_delta < TREND_WARN_RELATIVE:
        return None
    first_mean, second_mean = split_halves_means(step_ms_series)
    if second_mean <= first_mean * (1 + TREND_WARN_RELATIVE / 2):
        return None  # slope without level shift: noise
    return DiagnosticIssue(
        kind="STEP_TIME_DEGRADING",
        status="STEP TIME DEGRADING",
        severity="warn",
        summary=(
            f"Step time rose {trend.relative_delta * 100.0:.0f}% across the "
            f"window ({first_mean:.1f} -> {second_mean:.1f} ms)."
        ),
        action=(
            "Something is getting slower over time: check dataloader queue "
            "growth, host memory pressure, thermals (see System), or memory "
            "creep (see Step Memory)."
        ),
        metric="step_time_ms",
        score=trend.relative_delta,
        evidence={"type": "trend", **trend.to_payload()},
    )
