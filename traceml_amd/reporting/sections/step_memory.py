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

    # common-step alignment (the section's declared contract): compare peak
    # watermarks over the steps EVERY rank reported, so a rank that died
    # early does not skew the comparison window
    common_steps = None
    for s in series.values():
        steps = set(s.steps)
        common_steps = steps if common_steps is None else (common_steps & steps)
    common_steps = common_steps or set()

    per_rank = {}
    latest_step = None
    window_start = window_end = None
    for rank, s in sorted(series.items()):
        pairs = [
            (step, alloc, reserved)
            for step, alloc, reserved in zip(
                s.steps, s.peak_allocated, s.peak_reserved
            )
            if not common_steps or step in common_steps
        ]
        alloc = [a for _, a, _ in pairs if a is not None]
        reserved = [r for _, _, r in pairs if r is not None]
        per_rank[str(rank)] = {
            "peak_allocated_bytes": max(alloc) if alloc else None,
            "peak_reserved_bytes": max(reserved) if reserved else None,
        }
        if pairs:
            window_start = pairs[0][0] if window_start is None else min(window_start, pairs[0][0])
            window_end = pairs[-1][0] if window_end is None else max(window_end, pairs[-1][0])
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
    payload["global"]["window"]["start_step"] = window_start
    payload["global"]["window"]["end_step"] = window_end
    payload["global"]["window"]["steps_analyzed"] = (
        len(common_steps) or None
    )
    fill_metric_maps(payload, STEP_MEMORY_METRICS, per_rank)
    for rank, s in series.items():
        row = payload["groups"]["rows"].get(str(rank))
        if row is not None:
            row["identity"] = dict(s.identity)

    # evidence for charts (HTML report / dashboard): capacity, trend slope
    # and a downsampled per-rank allocated series
    from traceml_amd.renderers.step_memory import _slope_bytes_per_step

    trend = {}
    capacity = None
    for rank, s in sorted(series.items()):
        if s.capacity is not None:
            capacity = max(capacity or 0, s.capacity)
        stride = max(1, len(s.steps) // 60)
        points = [
            [s.steps[i], s.peak_allocated[i]]
            for i in range(0, len(s.steps), stride)
            if s.peak_allocated[i] is not None
        ]
        if points:
            trend[str(rank)] = {
                "slope_bytes_per_step": _slope_bytes_per_step(
                    s.steps, s.peak_allocated
                ),
                "series": points,
            }
    payload["evidence_extra"] = {
        "capacity_bytes": capacity,
        "trend": trend,
    }

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
