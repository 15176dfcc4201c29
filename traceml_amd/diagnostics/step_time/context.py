"""Straggler context: rank-comparison evidence for the step-time rules
(reference: diagnostics/step_time/context.py:162-306 and the
rank-straggler-policy doc).

The attribution inversion that is easy to get backwards: in DDP the visible
phase is **backward**, which on healthy ranks is inflated by waiting in the
gradient all-reduce for the late rank. So the **culprit is the rank with
the LOWEST visible value** (it shows up late and never waits) and the
victim is the median rank. The culprit's real problem lives in its other
phases — input wait, H2D, or its own compute — and a cause is only named
when its excess covers >= 80% of the visible gap.

MI355X upgrade: when the explicit ``ddp_comm`` phase was measured (RCCL
bucket timing on the comm stream), it is attached as corroborating
evidence — victims show large ddp_comm, the culprit small — but the
classification keeps the reference's visible-phase semantics so verdicts
stay comparable.
"""

from __future__ import annotations

import statistics
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from traceml_amd.diagnostics.step_time import policy
from traceml_amd.steptime.model import StepTimeWindow

STRATEGY_VISIBLE_PHASES = {
    "ddp": ("backward_ms",),
    "distributed_unknown": ("backward_ms",),
    "fsdp": ("forward_ms", "backward_ms"),
}


@dataclass
class StragglerContext:
    applicable: bool = False
    culprit_rank: Optional[int] = None
    victim_rank: Optional[int] = None
    culprit_visible_ms: Optional[float] = None
    victim_visible_ms: Optional[float] = None
    victim_step_time_ms: Optional[float] = None
    score: Optional[float] = None
    cause: Optional[str] = None  # "input" | "h2d" | "compute" | None
    cause_coverage: Optional[float] = None
    visible_phases: tuple = ()
    per_rank_visible: Dict[int, float] = field(default_factory=dict)
    evidence: Dict = field(default_factory=dict)


def _visible_value(values, phases) -> Optional[float]:
    total = 0.0
    for phase in phases:
        v = values.get(phase)
        if v is None:
            return None
        total += v
    return total


def build_straggler_context(window: StepTimeWindow) -> StragglerContext:
    ctx = StragglerContext()
    strategy = (window.training_strategy or "ddp").lower()
    phases = STRATEGY_VISIBLE_PHASES.get(strategy, ("backward_ms",))
    ctx.visible_phases = phases

    ranks = window.ranks_used
    if len(ranks) < 2:
        return ctx

    visible: Dict[int, float] = {}
    for rank in ranks:
        value = _visible_value(window.ranks[rank], phases)
        if value is None:
            return ctx  # incomplete visible signal -> no straggler verdict
        visible[rank] = value
    ctx.per_rank_visible = visible
    ctx.applicable = True

    # Culprit = LOWEST visible; victim = median rank by visible value.
    culprit = min(ranks, key=lambda r: (visible[r], r))
    med_value = statistics.median(visible.values())
    victim = min(
        (r for r in ranks if r != culprit),
        key=lambda r: (abs(visible[r] - med_value), r),
    )
    ctx.culprit_rank = culprit
    ctx.victim_rank = victim
    ctx.culprit_visible_ms = visible[culprit]
    ctx.victim_visible_ms = visible[victim]
    victim_step = window.ranks[victim].get("step_time_ms")
    ctx.victim_step_time_ms = victim_step

    gap = visible[victim] - visible[culprit]
    if victim_step is None or victim_step <= 0 or gap <= 0:
        ctx.score = 0.0
        return ctx
    ctx.score = gap / victim_step

    # Cause attribution: which of the culprit's hidden phases explains the gap?
    def excess(metric_names) -> Optional[float]:
        c_total = v_total = 0.0
        for m in metric_names:
            c = window.ranks[culprit].get(m)
            v = window.ranks[victim].get(m)
            if c is None or v is None:
                return None
            c_total += c
            v_total += v
        return c_total - v_total

    causes = {
        "input": excess(("input_wait_ms",)),
        "h2d": excess(("h2d_ms",)),
        # own-work proxy outside the visible phase (backward wait hides
        # the culprit's true backward compute)
        "compute": excess(
            tuple(
                m
                for m in ("forward_ms", "optimizer_ms")
                if m not in phases
            )
        ),
    }
    best_cause, best_excess = None, 0.0
    for cause, value in causes.items():
        if value is not None and value > best_excess:
            best_cause, best_excess = cause, value
    coverage = best_excess / gap if gap > 0 else 0.0
    ctx.cause_coverage = coverage
    if best_cause is not None and coverage >= policy.STRAGGLER_CAUSE_COVERAGE:
        ctx.cause = best_cause

    ctx.evidence = {
        "type": "rank_comparison",
        "visible_phases": list(phases),
        "median": {"rank": victim, "value_ms": visible[victim]},
        "worst": {"rank": culprit, "value_ms": visible[culprit]},
        "delta_ms": gap,
        "ratio": (visible[victim] / visible[culprit]) if visible[culprit] else None,
        "cause_excess_ms": {k: v for k, v in causes.items() if v is not None},
        "cause_coverage": coverage,
    }
    # MI355X: corroborating measured comm-wait evidence when available.
    ddp = {
        r: window.ranks[r].get("ddp_comm_ms")
        for r in ranks
        if window.ranks[r].get("ddp_comm_ms") is not None
    }
    if ddp:
        ctx.evidence["ddp_comm_ms_per_rank"] = ddp
    return ctx
