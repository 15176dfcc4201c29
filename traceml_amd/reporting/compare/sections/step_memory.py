"""Step-memory compare section (reference: compare/sections/step_memory.py):
peak allocated/reserved movement in bytes plus the worst-rank peaks, so a
single-rank blowup is visible even when averages move little."""

from __future__ import annotations

from typing import Optional

from traceml_amd.reporting.compare.model import CompareSection
from traceml_amd.reporting.compare.sections.base import (
    bytes_metric,
    diagnosis_transition,
    global_average,
    section_available,
)


def _worst(section: Optional[dict], metric: str) -> Optional[float]:
    if not isinstance(section, dict):
        return None
    entry = section.get("global", {}).get("worst", {}).get(metric)
    if isinstance(entry, dict):
        value = entry.get("value")
    else:
        value = entry
    return float(value) if isinstance(value, (int, float)) else None


def compare_step_memory(lhs_payload: dict, rhs_payload: dict) -> CompareSection:
    lhs = lhs_payload.get("step_memory")
    rhs = rhs_payload.get("step_memory")
    return CompareSection(
        name="step_memory",
        available=section_available(lhs, rhs),
        metrics={
            "peak_allocated_bytes": bytes_metric(
                "peak_allocated_bytes",
                "Peak allocated (avg)",
                global_average(lhs, "peak_allocated_bytes"),
                global_average(rhs, "peak_allocated_bytes"),
            ),
            "peak_reserved_bytes": bytes_metric(
                "peak_reserved_bytes",
                "Peak reserved (avg)",
                global_average(lhs, "peak_reserved_bytes"),
                global_average(rhs, "peak_reserved_bytes"),
            ),
            "peak_allocated_bytes_worst": bytes_metric(
                "peak_allocated_bytes_worst",
                "Peak allocated (worst rank)",
                _worst(lhs, "peak_allocated_bytes"),
                _worst(rhs, "peak_allocated_bytes"),
            ),
            "peak_reserved_bytes_worst": bytes_metric(
                "peak_reserved_bytes_worst",
                "Peak reserved (worst rank)",
                _worst(lhs, "peak_reserved_bytes"),
                _worst(rhs, "peak_reserved_bytes"),
            ),
        },
        diagnosis=diagnosis_transition(lhs, rhs),
    )
