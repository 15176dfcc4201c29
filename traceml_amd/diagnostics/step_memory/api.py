"""Step-memory diagnosis: pressure, rank imbalance, creep
(reference: diagnostics/step_memory/{api,rules,trend}.py, ~1.7k LoC).

Input: per-rank step-memory series (step, peak_allocated, peak_reserved,
capacity). Verdicts:
* HIGH_MEMORY_PRESSURE (reserved/capacity >= 92% warn / 97% crit)
* RANK_MEMORY_IMBALANCE (skew >= 20/30%, gated on pressure >= 30/50%)
* MEMORY_CREEP_CONFIRMED / MEMORY_CREEP_EARLY (conservative: >= 800 steps,
  >= 512 MiB absolute growth, sustained slope, weak recovery)
* NORMAL / NO_DATA
"""

from __future__ import annotations

import sqlite3
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from traceml_amd.diagnostics.common import (
    DiagnosticIssue,
    DiagnosticResult,
    sort_issues,
)
from traceml_amd.diagnostics.step_memory import policy


@dataclass
class MemorySeries:
    global_rank: int
    steps: List[int] = field(default_factory=list)
    peak_allocated: List[Optional[int]] = field(default_factory=list)
    peak_reserved: List[Optional[int]] = field(default_factory=list)
    capacity: Optional[int] = None
    identity: dict = field(default_factory=dict)
    #: HIP caching-allocator churn: total blocking-malloc retries and the
    #: latest live segment count over the window (None = not measured)
    alloc_retries_total: Optional[int] = None
    segments_latest: Optional[int] = None


def load_memory_series(db_path: str) -> Dict[int, MemorySeries]:
    try:
        conn = sqlite3.connect(f"file:{db_path}?mode=ro", uri=True)
        conn.row_factory = sqlite3.Row
    except sqlite3.Error:
        return {}
    try:
        try:
            rows = conn.execute(
                """
                SELECT global_rank, local_rank, node_rank, hostname,
                       world_size, local_world_size, step,
                       peak_allocated_bytes, peak_reserved_bytes,
                       device_capacity_bytes, alloc_retries, segments
                FROM step_memory_samples ORDER BY global_rank, step, id
                """
            ).fetchall()
        except sqlite3.Error:
            # pre-churn-column DB (additive schema evolution): legacy set
            try:
                rows = conn.execute(
                    """
                    SELECT global_rank, local_rank, node_rank, hostname,
                           world_size, local_world_size, step,
                           peak_allocated_bytes, peak_reserved_bytes,
                           device_capacity_bytes, NULL AS alloc_retries,
                           NULL AS segments
                    FROM step_memory_samples ORDER BY global_rank, step, id
                    """
                ).fetchall()
            except sqlite3.Error:
                return {}
    finally:
        conn.close()
    series: Dict[int, MemorySeries] = {}
    for r in rows:
        rank = r["global_rank"]
        if rank is None:
            continue
        s = series.get(rank)
        if s is None:
            s = MemorySeries(global_rank=int(rank))
            series[rank] = s
        if s.steps and s.steps[-1] == r["step"]:
            # dedupe by (rank, step): keep newest
            s.steps.pop()
            s.peak_allocated.pop()
            s.peak_reserved.pop()
        s.steps.append(int(r["step"] or 0))
        s.peak_allocated.append(r["peak_allocated_bytes"])
        s.peak_reserved.append(r["peak_reserved_bytes"])
        if r["device_capacity_bytes"]:
            s.capacity = int(r["device_capacity_bytes"])
        if r["alloc_retries"] is not None and r["alloc_retries"] > 0:
            s.alloc_retries_total = (s.alloc_retries_total or 0) + int(
                r["alloc_retries"]
            )
        if r["segments"] is not None:
            s.segments_latest = int(r["segments"])
        s.identity = {
            "global_rank": rank,
            "local_rank": r["local_rank"],
            "node_rank": r["node_rank"],
            "hostname": r["hostname"],
            "local_world_size": r["local_world_size"],
            "world_size": r["world_size"],
        }
    return series


def _pressure(s: MemorySeries) -> Optional[float]:
    reserved = [v for v in s.peak_reserved if v is not None]
    if not reserved or not s.capacity:
        return None
    return max(reserved) / s.capacity


def _slope(xs: List[int], ys: List[float]) -> float:
    n = len(xs)
    if n < 2:
        return 0.0
    mx = sum(xs) / n
    my = sum(ys) / n
    denom = sum((x - mx) ** 2 for x in xs)
    if denom == 0:
        return 0.0
    return sum((x - mx) * (y - my) for x, y in zip(xs, ys)) / denom


def _creep_issue(s: MemorySeries) -> Optional[DiagnosticIssue]:
    pairs = [
        (step, float(alloc))
        for step, alloc in zip(s.steps, s.peak_allocated)
        if alloc is not None
    ]
    if len(pairs) < 2:
        return None
    steps_span = pairs[-1][0] - pairs[0][0] + 1
    values = [v for _, v in pairs]
    delta = values[-1] - values[0]
    peak = max(values)
    recovery = (peak - values[-1]) / max(1.0, peak - values[0]) if peak > values[0] else 1.0
    slope = _slope([p[0] for p in pairs], values)

    confirmed = (
        steps_span >= policy.CREEP_MIN_STEPS
        and delta >= policy.CREEP_MIN_DELTA_BYTES
        and slope >= policy.CREEP_SLOPE_MIN_BYTES_PER_STEP
        and recovery <= policy.CREEP_RECOVERY_MAX_FRACTION
    )
    early = (
        not confirmed
        and steps_span >= policy.CREEP_EARLY_MIN_STEPS
        and delta >= policy.CREEP_MIN_DELTA_BYTES / 2
        and slope >= policy.CREEP_SLOPE_MIN_BYTES_PER_STEP
        and recovery <= policy.CREEP_RECOVERY_MAX_FRACTION
    )
    if not (confirmed or early):
        return None
    kind = "MEMORY_CREEP_CONFIRMED" if confirmed else "MEMORY_CREEP_EARLY"
    return DiagnosticIssue(
        kind=kind,
        status="MEMORY CREEP" if confirmed else "MEMORY CREEP (EARLY)",
        severity="warn" if confirmed else "info",
        summary=(
            f"Rank r{s.global_rank}: peak allocated grew "
            f"{delta / (1 << 20):.0f} MiB over {steps_span} steps "
            f"({slope:.0f} B/step) without recovering."
        ),
        action=(
            "Look for tensors retained across steps: growing python lists of "
            "loss tensors, missing .detach(), caches keyed by step."
        ),
        metric="peak_allocated_bytes",
        ranks=[s.global_rank],
        score=delta / (1 << 30),
        evidence={
            "steps_span": steps_span,
            "delta_bytes": delta,
            "slope_bytes_per_step": slope,
            "recovery_fraction": recovery,
        },
    )


def diagnose_step_memory(series: Dict[int, MemorySeries]) -> DiagnosticResult:
    if not series:
        return DiagnosticResult(
            issues=[
                DiagnosticIssue(
                    kind="NO_DATA",
                    status="NO DATA",
                    severity="info",
                    summary="No step-memory telemetry recorded.",
                    action="Memory watermarks require a GPU run under trace_step.",
                )
            ]
        )
    measured = {
        r: s
        for r, s in series.items()
        if any(v is not None for v in s.peak_reserved)
    }
    if not measured:
        return DiagnosticResult(
            issues=[
                DiagnosticIssue(
                    kind="NO_GPU",
                    status="NO GPU",
                    severity="info",
                    summary="Step-memory rows exist but no GPU watermarks were "
                    "measured (CPU run).",
                    action="",
                )
            ]
        )

    issues: List[DiagnosticIssue] = []
    pressures = {r: _pressure(s) for r, s in measured.items()}
    pressures = {r: p for r, p in pressures.items() if p is not None}

    for rank, pressure in sorted(pressures.items()):
        if pressure >= policy.PRESSURE_WARN:
            severity = "crit" if pressure >= policy.PRESSURE_CRIT else "warn"
            issues.append(
                DiagnosticIssue(
                    kind="HIGH_MEMORY_PRESSURE",
                    status="HIGH MEMORY PRESSURE",
                    severity=severity,
                    summary=(
                        f"Rank r{rank}: peak reserved is {pressure * 100.0:.1f}% "
                        "of device memory — OOM risk."
                    ),
                    action=(
                        "Reduce batch size / activation memory, or shard more "
                        "(the MI355X has 288 GB HBM3E per GPU — check for "
                        "fragmentation before shrinking the model)."
                    ),
                    metric="peak_reserved_bytes",
                    ranks=[rank],
                    score=pressure,
                    evidence={"pressure": pressure},
                )
            )

    if len(pressures) >= 2:
        max_rank = max(pressures, key=lambda r: pressures[r])
        min_p = min(pressures.values())
        max_p = pressures[max_rank]
        skew = (max_p - min_p) / max_p if max_p > 0 else 0.0
        if skew >= policy.IMBALANCE_WARN and max_p >= policy.IMBALANCE_PRESSURE_GATE_WARN:
            severity = (
                "crit"
                if (
                    skew >= policy.IMBALANCE_CRIT
                    and max_p >= policy.IMBALANCE_PRESSURE_GATE_CRIT
                )
                else "warn"
            )
            issues.append(
                DiagnosticIssue(
                    kind="RANK_MEMORY_IMBALANCE",
                    status="RANK MEMORY IMBALANCE",
                    severity=severity,
                    summary=(
                        f"Peak reserved memory is skewed {skew * 100.0:.0f}% "
                        f"across ranks (max on r{max_rank})."
                    ),
                    action="Check for uneven sharding or rank-0-only buffers.",
                    metric="peak_reserved_bytes",
                    ranks=[max_rank],
                    score=skew,
                    skew_pct=skew * 100.0,
                    evidence={"pressures": pressures},
                )
            )

    for s in measured.values():
        creep = _creep_issue(s)
        if creep is not None:
            issues.append(creep)

    # allocator churn: any blocking-malloc retry means the HIP caching
    # allocator flushed its cache mid-step (fragmentation / oversubscribed
    # reserve) — visible long before an OOM, invisible in peaks alone
    for rank, s in sorted(measured.items()):
        if s.alloc_retries_total:
            issues.append(
                DiagnosticIssue(
                    kind="ALLOCATOR_CHURN",
                    status="ALLOCATOR CHURN",
                    severity="warn",
                    summary=(
                        f"Rank r{rank}: the HIP caching allocator hit "
                        f"{s.alloc_retries_total} blocking-malloc "
                        "retr{} during the window — cache flushes stall "
                        "the stream and precede OOMs.".format(
                            "y" if s.alloc_retries_total == 1 else "ies"
                        )
                    ),
                    action=(
                        "Reduce fragmentation: avoid many transient "
                        "odd-sized allocations, or set "
                        "PYTORCH_HIP_ALLOC_CONF=expandable_segments:True "
                        "(288 GB HBM3E leaves headroom — churn is usually "
                        "fragmentation, not capacity)."
                    ),
                    metric="alloc_retries",
                    ranks=[rank],
                    score=float(s.alloc_retries_total),
                    evidence={
                        "alloc_retries": s.alloc_retries_total,
                        "segments": s.segments_latest,
                    },
                )
            )

    if not issues:
        issues.append(
            DiagnosticIssue(
                kind="NORMAL",
                status="NORMAL",
                severity="info",
                summary="Step memory is stable and within budget.",
                action="",
            )
        )
    return DiagnosticResult(issues=sort_issues(issues))
