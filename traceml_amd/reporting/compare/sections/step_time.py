"""Step-time compare section (reference: compare/sections/step_time.py).

Clock-aware: each summary selects one window clock (GPU iff complete);
phase metrics are only compared when both runs used the SAME clock —
otherwise the comparison falls back to the CPU-clock step time that both
always carry, with an explanatory note (a GPU-clock 40 ms vs CPU-clock
38 ms is measurement skew, not a regression).
"""

from __future__ import annotations

from typing import Dict, List, Optional

from traceml_amd.reporting.compare.model import CompareSection
from traceml_amd.reporting.compare.sections.base import (
    diagnosis_transition,
    global_average,
    section_available,
    time_metric,
)

PHASE_METRICS = (
    ("input_wait_ms", "Input wait"),
    ("h2d_ms", "H2D"),
    ("forward_ms", "Forward"),
    ("backward_ms", "Backward"),
    ("optimizer_ms", "Optimizer"),
    ("ddp_comm_ms", "DDP comm"),
    ("compute_ms", "Compute"),
    ("residual_ms", "Residual"),
)


def _selected_clock(section: Optional[dict]) -> Optional[str]:
    step = global_average(section, "step_time_ms")
    if step is None:
        return None
    gpu = global_average(section, "step_time_gpu_ms")
    if gpu is not None and abs(gpu - step) < 1e-9:
        return "gpu"
    cpu = global_average(section, "step_time_cpu_ms")
    if cpu is not None and abs(cpu - step) < 1e-9:
        return "cpu"
    return "cpu" if cpu is not None else None


def compare_step_time(lhs_payload: dict, rhs_payload: dict) -> CompareSection:
    lhs = lhs_payload.get("step_time")
    rhs = rhs_payload.get("step_time")
    notes: List[str] = []

    lhs_clock = _selected_clock(lhs)
    rhs_clock = _selected_clock(rhs)
    clocks_match = (
        lhs_clock is not None and rhs_clock is not None
        and lhs_clock == rhs_clock
    )

    if clocks_match:
        step_key = "step_time_ms"
        step_label = f"Step time ({lhs_clock} clock)"
        lhs_step = global_average(lhs, "step_time_ms")
        rhs_step = global_average(rhs, "step_time_ms")
    else:
        # fall back to the CPU clock both summaries always measure
        step_key = "step_time_ms"
        step_label = "Step time (cpu clock, mixed-clock fallback)"
        lhs_step = global_average(lhs, "step_time_cpu_ms")
        rhs_step = global_average(rhs, "step_time_cpu_ms")
        if lhs_step is None or rhs_step is None:
            lhs_step = global_average(lhs, "step_time_ms")
            rhs_step = global_average(rhs, "step_time_ms")
        if lhs_clock or rhs_clock:
            notes.append(
                "selected clocks differ "
                f"(A: {lhs_clock or 'n/a'}, B: {rhs_clock or 'n/a'}); "
                "step time compared on the common CPU clock, per-phase "
                "metrics withheld"
            )

    metrics: Dict[str, object] = {
        step_key: time_metric(step_key, step_label, lhs_step, rhs_step)
    }
    if clocks_match:
        for key, label in PHASE_METRICS:
            metrics[key] = time_metric(
                key, label, global_average(lhs, key), global_average(rhs, key)
            )

    section = CompareSection(
        name="step_time",
        available=section_available(lhs, rhs),
        metrics=metrics,
        diagnosis=diagnosis_transition(lhs, rhs),
        notes=notes,
    )
    return section


def per_rank_step_time(lhs_payload: dict, rhs_payload: dict) -> List[dict]:
    def rank_values(payload):
        rows = (
            payload.get("step_time", {}).get("groups", {}).get("rows", {})
        )
        return {
            key: row.get("metrics", {}).get("step_time_ms")
            for key, row in rows.items()
        }

    b_rows = rank_values(lhs_payload)
    c_rows = rank_values(rhs_payload)
    out = []
    for key in sorted(set(b_rows) | set(c_rows), key=lambda k: (len(k), k)):
        b = b_rows.get(key)
        c = c_rows.get(key)
        delta = (c - b) if (b is not None and c is not None) else None
        out.append(
            {
                "rank": key,
                "baseline": b,
                "candidate": c,
                "delta": delta,
                "pct": (delta / b * 100.0) if (delta is not None and b) else None,
            }
        )
    return out
