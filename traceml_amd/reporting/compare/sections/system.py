"""System compare section (reference: compare/sections/system.py). These
are CONTEXT metrics: GPU util dropping explains a step-time regression but
never drives the verdict on its own."""

from __future__ import annotations

from traceml_amd.reporting.compare.model import CompareSection
from traceml_amd.reporting.compare.sections.base import (
    bytes_metric,
    diagnosis_transition,
    global_average,
    points_metric,
    section_available,
)


def compare_system(lhs_payload: dict, rhs_payload: dict) -> CompareSection:
    lhs = lhs_payload.get("system")
    rhs = rhs_payload.get("system")
    return CompareSection(
        name="system",
        available=section_available(lhs, rhs),
        metrics={
            "gpu_util_percent": points_metric(
                "gpu_util_percent",
                "GPU util avg",
                global_average(lhs, "gpu_util_percent"),
                global_average(rhs, "gpu_util_percent"),
            ),
            "gpu_mem_percent": points_metric(
                "gpu_mem_percent",
                "GPU memory avg",
                global_average(lhs, "gpu_mem_percent"),
                global_average(rhs, "gpu_mem_percent"),
            ),
            "gpu_temp_c": points_metric(
                "gpu_temp_c",
                "GPU temp avg (°C)",
                global_average(lhs, "gpu_temp_c"),
                global_average(rhs, "gpu_temp_c"),
            ),
            "gpu_power_w": points_metric(
                "gpu_power_w",
                "GPU power avg (W)",
                global_average(lhs, "gpu_power_w"),
                global_average(rhs, "gpu_power_w"),
            ),
            "cpu_percent": points_metric(
                "cpu_percent",
                "Host CPU avg",
                global_average(lhs, "cpu_percent"),
                global_average(rhs, "cpu_percent"),
            ),
            "ram_bytes": bytes_metric(
                "ram_bytes",
                "Host RAM avg",
                global_average(lhs, "ram_bytes"),
                global_average(rhs, "ram_bytes"),
                direction="context",
            ),
        },
        diagnosis=diagnosis_transition(lhs, rhs),
    )
