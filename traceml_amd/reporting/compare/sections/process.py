"""Process compare section (reference: compare/sections/process.py):
per-process RSS / GPU memory movement plus the profiler's own measured
self-overhead — compare is how a tracing regression in traceml itself
would be caught."""

from __future__ import annotations

from traceml_amd.reporting.compare.model import CompareSection
from traceml_amd.reporting.compare.sections.base import (
    bytes_metric,
    diagnosis_transition,
    global_average,
    points_metric,
    section_available,
    time_metric,
)


def compare_process(lhs_payload: dict, rhs_payload: dict) -> CompareSection:
    lhs = lhs_payload.get("process")
    rhs = rhs_payload.get("process")

    def us_metric(key, label):
        lhs_us = global_average(lhs, key)
        rhs_us = global_average(rhs, key)
        m = time_metric(
            key, label,
            None if lhs_us is None else lhs_us / 1000.0,
            None if rhs_us is None else rhs_us / 1000.0,
        )
        m.unit = "ms"
        return m

    return CompareSection(
        name="process",
        available=section_available(lhs, rhs),
        metrics={
            "ram_bytes": bytes_metric(
                "ram_bytes",
                "Process RSS avg",
                global_average(lhs, "ram_bytes"),
                global_average(rhs, "ram_bytes"),
                direction="context",
            ),
            "gpu_mem_used_bytes": bytes_metric(
                "gpu_mem_used_bytes",
                "Process GPU allocated avg",
                global_average(lhs, "gpu_mem_used_bytes"),
                global_average(rhs, "gpu_mem_used_bytes"),
            ),
            "gpu_mem_reserved_bytes": bytes_metric(
                "gpu_mem_reserved_bytes",
                "Process GPU reserved avg",
                global_average(lhs, "gpu_mem_reserved_bytes"),
                global_average(rhs, "gpu_mem_reserved_bytes"),
            ),
            "cpu_percent": points_metric(
                "cpu_percent",
                "Process CPU avg",
                global_average(lhs, "cpu_percent"),
                global_average(rhs, "cpu_percent"),
            ),
            "traceml_self_overhead_ms": us_metric(
                "traceml_self_overhead_us", "TraceML self-overhead/step"
            ),
        },
        diagnosis=diagnosis_transition(lhs, rhs),
    )
