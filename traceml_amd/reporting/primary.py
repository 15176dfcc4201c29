"""Primary diagnosis: the top-level "why was training slow?" verdict,
promoted from already-built section payloads — never from raw telemetry
(reference: reporting/primary_diagnosis.py:582; SCHEMA.md:62-169).

Selection policy:
* straggler + phase-share kinds promote from step_time.diagnosis
* LOW_GPU_UTILIZATION_UNEXPLAINED when step_time is BALANCED but System
  reports LOW/MODERATE_GPU_UTILIZATION
* NO_CLEAR_PERFORMANCE_BOTTLENECK when BALANCED and utilization is fine
* INSUFFICIENT_STEP_TIME_DATA for NO_DATA / WARMUP / INCOMPLETE_DATA
"""

from __future__ import annotations

from typing import Optional

_PROMOTABLE = {
    "INPUT_STRAGGLER",
    "COMPUTE_STRAGGLER",
    "H2D_STRAGGLER",
    "STRAGGLER",
    "INPUT_BOUND",
    "H2D_BOUND",
    "COMPUTE_BOUND",
    "RESIDUAL_HEAVY",
}

_INSUFFICIENT = {"NO_DATA", "WARMUP", "INCOMPLETE_DATA"}


def _gpu_util_issue(system_payload: dict) -> Optional[dict]:
    for issue in system_payload.get("issues", []):
        if issue.get("kind") in ("LOW_GPU_UTILIZATION", "MODERATE_GPU_UTILIZATION"):
            return issue
    return None


def build_primary_diagnosis(
    step_time_payload: dict, system_payload: dict
) -> dict:
    st_diag = step_time_payload.get("diagnosis") or {}
    kind = st_diag.get("kind")

    if kind in _PROMOTABLE:
        return {
            "kind": kind,
            "status": st_diag.get("status"),
            "severity": st_diag.get("severity", "info"),
            "section": "step_time",
            "scope": "performance",
            "summary": st_diag.get("summary", ""),
            "action": st_diag.get("action", ""),
            "evidence": st_diag.get("evidence", {}),
        }

    if kind in _INSUFFICIENT:
        evidence = {"step_time_status": st_diag.get("status")}
        if kind == "INCOMPLETE_DATA":
            evidence.update(
                {
                    "missing_signals": st_diag.get("evidence", {}).get(
                        "missing_signals", []
                    ),
                    "signal_coverage": st_diag.get("evidence", {}).get(
                        "signal_coverage", {}
                    ),
                }
            )
        return {
            "kind": "INSUFFICIENT_STEP_TIME_DATA",
            "status": "INSUFFICIENT STEP TIME DATA",
            "severity": "info",
            "section": "step_time",
            "scope": "performance",
            "summary": st_diag.get("summary", "Not enough step-time data."),
            "action": st_diag.get("action", ""),
            "evidence": {"type": "insufficient_data", **evidence},
        }

    # BALANCED (or COMPUTE_BOUND already handled above)
    util_issue = _gpu_util_issue(system_payload)
    if util_issue is not None:
        return {
            "kind": "LOW_GPU_UTILIZATION_UNEXPLAINED",
            "status": "LOW GPU UTILIZATION (UNEXPLAINED)",
            "severity": "warn",
            "section": "system",
            "scope": "performance",
            "summary": (
                "Step timing looks balanced, but GPU utilization is low — "
                "time is being lost somewhere the step instrumentation "
                "cannot see."
            ),
            "action": (
                "Profile one step with rocprofv3 (kernel trace) to find gaps "
                "between kernels; check for host synchronization points."
            ),
            "evidence": {
                "type": "utilization_fallback",
                "gpu_util_avg_percent": util_issue.get("evidence", {}).get(
                    "util_avg"
                ),
            },
        }

    return {
        "kind": "NO_CLEAR_PERFORMANCE_BOTTLENECK",
        "status": "NO CLEAR BOTTLENECK",
        "severity": "info",
        "section": "step_time",
        "scope": "performance",
        "summary": "No phase dominates and GPU utilization is healthy.",
        "action": "",
        "evidence": {"type": "no_clear_bottleneck"},
    }
