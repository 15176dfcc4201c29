"""Step-time rule engine (reference: diagnostics/step_time/rules.py + api.py:40-66).

Verdict kinds: NO_DATA, WARMUP, INCOMPLETE_DATA, BALANCED,
{INPUT,H2D,COMPUTE}_STRAGGLER, STRAGGLER, INPUT_BOUND, H2D_BOUND,
COMPUTE_BOUND, RESIDUAL_HEAVY — severity derives from the policy thresholds
and the warmup gates (warn below the confident window, crit at/after).
FSDP straggler severity is capped at warn (advisory support only).
"""

from __future__ import annotations

import statistics
from typing import List, Optional

from traceml_amd.diagnostics.common import DiagnosticIssue, sort_issues
from traceml_amd.diagnostics.step_time import policy
from traceml_amd.diagnostics.step_time.context import (
    StragglerContext,
    build_straggler_context,
)
from traceml_amd.steptime.model import StepTimeWindow

_STRAGGLER_KINDS = {
    "input": ("INPUT_STRAGGLER", "INPUT STRAGGLER"),
    "h2d": ("H2D_STRAGGLER", "H2D STRAGGLER"),
    "compute": ("COMPUTE_STRAGGLER", "COMPUTE STRAGGLER"),
    None: ("STRAGGLER", "STRAGGLER"),
}

_SHARE_KINDS = {
    "input": ("INPUT_BOUND", "INPUT-BOUND", "input_wait_ms"),
    "h2d": ("H2D_BOUND", "H2D-BOUND", "h2d_ms"),
    "residual": ("RESIDUAL_HEAVY", "RESIDUAL-HEAVY", "residual_ms"),
}

_ACTIONS = {
    "INPUT_BOUND": "Speed up the input pipeline: more DataLoader workers, "
    "faster storage, prefetch, or cached preprocessing.",
    "H2D_BOUND": "Reduce host-to-device transfer cost: pinned memory, "
    "non_blocking=True, larger batches, or move preprocessing to the GPU.",
    "RESIDUAL_HEAVY": "Significant untraced time inside the step: check "
    "logging/validation/checkpointing inside the loop; profile with rocprofv3.",
    "COMPUTE_BOUND": "Step time is dominated by compute — scale up or "
    "optimize kernels; the input pipeline is keeping up.",
    "INPUT_STRAGGLER": "One rank's dataloader is slow; the others wait in "
    "the gradient all-reduce. Fix that rank's input pipeline or storage.",
    "H2D_STRAGGLER": "One rank's host-to-device copies are slow; check its "
    "PCIe/NUMA placement and pinned-memory usage.",
    "COMPUTE_STRAGGLER": "One rank computes slower than its peers; check "
    "its clocks/thermals and colocated load.",
    "STRAGGLER": "One rank lags its peers; its excess is spread across "
    "phases. Inspect that rank's node.",
}


def _median_share(window: StepTimeWindow, metric: str) -> Optional[float]:
    """Authoritative score: median per-rank share of selected step time."""
    shares = []
    for rank in window.ranks_used:
        value = window.ranks[rank].get(metric)
        step = window.ranks[rank].get("step_time_ms")
        if value is not None and step and step > 0:
            shares.append(value / step)
    return statistics.median(shares) if shares else None


def evaluate(window: StepTimeWindow) -> List[DiagnosticIssue]:
    if not window.has_data:
        return [
            DiagnosticIssue(
                kind="NO_DATA",
                status="NO DATA",
                severity="info",
                summary="No step-time telemetry was recorded in this window.",
                action="Wrap your training step in traceml_amd.trace_step(model).",
            )
        ]
    if window.steps_analyzed < policy.MIN_STEPS_WARN:
        return [
            DiagnosticIssue(
                kind="WARMUP",
                status="WARMUP",
                severity="info",
                summary=(
                    f"Only {window.steps_analyzed} aligned step(s) — too few "
                    "for a verdict."
                ),
                action="Let training run longer.",
                evidence={"steps_analyzed": window.steps_analyzed},
            )
        ]
    incomplete_note: Optional[DiagnosticIssue] = None
    if window.missing_signals:
        incomplete_note = DiagnosticIssue(
            kind="INCOMPLETE_DATA",
            status="INCOMPLETE DATA",
            severity="info",
            summary=(
                "Required timing signals were never measured: "
                + ", ".join(window.missing_signals)
            ),
            action=(
                "Enable the corresponding instrumentation (init mode "
                "'auto', or the matching wrap_* helper)."
            ),
            evidence={
                "missing_signals": list(window.missing_signals),
                "signal_coverage": dict(window.signal_coverage),
            },
        )
        # Only the INPUT stream missing (a loop with no DataLoader — the
        # 20k-step production soak is exactly this shape): the traced
        # envelope + forward/backward are complete, so the phase rules can
        # still produce a useful verdict over the measured phases; the
        # incomplete-data note demotes to a secondary issue. Any missing
        # COMPUTE signal keeps incomplete-data as the whole verdict.
        if set(window.missing_signals) - {"dataloader"}:
            return [incomplete_note]

    confident = window.steps_analyzed >= policy.MIN_STEPS_CONFIDENT
    issues: List[DiagnosticIssue] = []

    # -- straggler rules (rank comparison) ---------------------------------
    ctx = build_straggler_context(window)
    if ctx.applicable and ctx.score is not None and ctx.score >= policy.STRAGGLER_WARN:
        severity = (
            "crit" if (ctx.score >= policy.STRAGGLER_CRIT and confident) else "warn"
        )
        if (window.training_strategy or "").lower() == "fsdp":
            severity = "warn"  # FSDP support is advisory (SCHEMA.md:106-109)
        kind, status = _STRAGGLER_KINDS[ctx.cause]
        issues.append(
            DiagnosticIssue(
                kind=kind,
                status=status,
                severity=severity,
                summary=_straggler_summary(ctx),
                action=_ACTIONS[kind],
                metric={
                    "input": "input_wait_ms",
                    "h2d": "h2d_ms",
                    "compute": "compute_ms",
                    None: None,
                }[ctx.cause],
                phase=ctx.cause,
                score=ctx.score,
                ranks=[ctx.culprit_rank],
                evidence={
                    **ctx.evidence,
                    "steps_analyzed": window.steps_analyzed,
                },
            )
        )

    # -- phase-share rules --------------------------------------------------
    for phase, (kind, status, metric) in _SHARE_KINDS.items():
        score = _median_share(window, metric)
        if score is None or score < policy.SHARE_WARN:
            continue
        severity = "crit" if (score >= policy.SHARE_CRIT and confident) else "warn"
        avg_value = window.average.get(metric)
        avg_step = window.average.get("step_time_ms")
        issues.append(
            DiagnosticIssue(
                kind=kind,
                status=status,
                severity=severity,
                summary=(
                    f"{status.replace('-', ' ').title()}: {phase} is "
                    f"{score * 100.0:.1f}% of step time"
                    + (
                        f" ({avg_value:.1f} ms of {avg_step:.1f} ms avg)"
                        if avg_value is not None and avg_step
                        else ""
                    )
                    + "."
                ),
                action=_ACTIONS[kind],
                metric=metric,
                phase=phase,
                score=score,
                share_pct=score * 100.0,
                evidence={
                    "type": "phase_share",
                    "basis": "average",
                    "steps_analyzed": window.steps_analyzed,
                    "score_basis": "median_per_rank_step_time_share",
                    "score_denominator": "selected-clock Step Time per rank",
                    **{
                        m: window.average.get(m)
                        for m in (
                            "input_wait_ms",
                            "step_time_ms",
                            "traced_step_time_ms",
                            "dataloader_fetch_cpu_ms",
                            "h2d_ms",
                            "compute_ms",
                            "residual_ms",
                            "ddp_comm_ms",
                        )
                    },
                    "diagnosis_clock": window.clock,
                },
            )
        )

    # -- compute-bound (informational, unscored) ---------------------------
    compute_share = _median_share(window, "compute_ms")
    if not issues and compute_share is not None and compute_share >= policy.COMPUTE_BOUND_SHARE:
        issues.append(
            DiagnosticIssue(
                kind="COMPUTE_BOUND",
                status="COMPUTE-BOUND",
                severity="info",
                summary=(
                    f"Compute (forward+backward+optimizer) is "
                    f"{compute_share * 100.0:.1f}% of step time — the GPU is "
                    "the bottleneck, which is the healthy state."
                ),
                action=_ACTIONS["COMPUTE_BOUND"],
                metric="compute_ms",
                phase="compute",
                share_pct=compute_share * 100.0,
                evidence={
                    "type": "phase_share",
                    "basis": "median_per_rank",
                    "steps_analyzed": window.steps_analyzed,
                    "diagnosis_clock": window.clock,
                },
            )
        )

    # -- degradation trend (supporting issue, never primary on its own) ----
    trend_issue = _trend_issue(window)
    if trend_issue is not None:
        issues.append(trend_issue)

    non_trend = [i for i in issues if i.kind != "STEP_TIME_DEGRADING"]
    if not non_trend:
        issues.insert(
            0,
            DiagnosticIssue(
                kind="BALANCED",
                status="BALANCED",
                severity="info",
                summary="No phase dominates step time and ranks are in step.",
                action="No step-time action needed.",
                evidence={
                    "steps_analyzed": window.steps_analyzed,
                    "diagnosis_clock": window.clock,
                },
            )
        )
    ordered = sort_issues(issues)
    # The degradation trend is supporting evidence, never THE diagnosis
    # when any other finding (including BALANCED/COMPUTE_BOUND) exists.
    if len(ordered) > 1 and ordered[0].kind == "STEP_TIME_DEGRADING":
        trend = ordered.pop(0)
        ordered.insert(1, trend)
    if incomplete_note is not None:
        ordered.append(incomplete_note)  # secondary caveat, never primary
    return ordered


def _trend_issue(window: StepTimeWindow):
    """Mean-across-ranks per-step step-time series -> degradation check."""
    from traceml_amd.diagnostics.step_time.trend import step_time_trend_issue

    per_step: dict = {}
    for series in window.step_series.values():
        for step, ms in series:
            per_step.setdefault(step, []).append(ms)
    if not per_step:
        return None
    steps = sorted(per_step)
    means = [sum(per_step[s]) / len(per_step[s]) for s in steps]
    return step_time_trend_issue(steps, means)


def _straggler_summary(ctx: StragglerContext) -> str:
    cause_text = {
        "input": "slow input pipeline",
        "h2d": "slow host-to-device transfers",
        "compute": "slow compute",
        None: "mixed causes",
    }[ctx.cause]
    return (
        f"Rank r{ctx.culprit_rank} lags its peers ({cause_text}): visible "
        f"phase {ctx.culprit_visible_ms:.1f} ms vs median "
        f"{ctx.victim_visible_ms:.1f} ms; peers lose "
        f"{ctx.score * 100.0:.1f}% of step time waiting."
    )
