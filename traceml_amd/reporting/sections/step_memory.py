"""Step Memory section builder (reference: reporting/sections/step_memory/*)."""

from __future__ import annotations

from traceml_amd.diagnostics.step_memory.api import (
    diagnose_step_memory,
    load_memory_series,
)
from traceml_amd.reporting.schema import (
    STEP_MEMORY_METRICS,
    empty_section_payload,
    fill_metric_maps,
)


def build(db_path: str) -> dict:
    payload = empty_section_payload(STEP_MEMORY_METRICS, index_by="global_rank")
    series = load_memory_series(db_path)
    payload.update(diagnose_step_memory(series).to_payload())

    per_rank = {}
    latest_step = None
    for rank, s in sorted(series.items()):
        alloc = [v for v in s.peak_allocated if v is not None]
        reserved = [v for v in s.peak_reserved if v is not None]
        per_rank[str(rank)] = {
            "peak_allocated_bytes": max(alloc) if alloc else None,
            "peak_reserved_bytes": max(reserved) if reserved else None,
        }
        if s.steps:
            latest_step = max(latest_step or 0, s.steps[-1])

    md = payload["metadata"]
    md["mode"] = "single_node" if series else "no_data"
    md["samples"] = sum(len(s.steps) for s in series.values()) or None
    md["global_ranks_seen"] = sorted(series)
    md["global_ranks_used"] = sorted(per_rank)
    md["training_latest_step"] = latest_step
    md["training_total_steps"] = latest_step

    payload["global"]["window"]["kind"] = "step_window"
    payload["global"]["window"]["alignment"] = "common_steps"
    fill_metric_maps(payload, STEP_MEMORY_METRICS, per_rank)
    for rank, s in series.items():
        row = payload["groups"]["rows"].get(str(rank))
        if row is not None:
            row["identity"] = dict(s.identity)

    diag = payload.get("diagnosis") or {}
    gib = 1 << 30
    lines = ["Step Memory"]
    for rank, values in per_rank.items():
        alloc = values["peak_allocated_bytes"]
        reserved = values["peak_reserved_bytes"]
        if alloc is None and reserved is None:
            lines.append(f"  r{rank}: not measured (CPU run)")
        else:
            lines.append(
                f"  r{rank}: peak allocated "
                f"{(alloc or 0) / gib:.1f} GiB, reserved {(reserved or 0) / gib:.1f} GiB"
            )
    if diag:
        lines.append(f"  Verdict: {diag.get('status')} — {diag.get('summary')}")
    payload["card"] = "\n".join(lines)
    return payload
