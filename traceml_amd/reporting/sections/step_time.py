"""Step Time section builder: projects the shared pipeline result to the
public section JSON (reference: reporting/sections/step_time/builder.py:405)."""

from __future__ import annotations

from traceml_amd.reporting.schema import (
    STEP_TIME_METRICS,
    empty_section_payload,
    fill_metric_maps,
)
from traceml_amd.steptime.pipeline import StepTimePipeline


def build(db_path: str) -> dict:
    payload = empty_section_payload(STEP_TIME_METRICS, index_by="global_rank")
    result = StepTimePipeline(db_path, profile="summary").run()
    window = result.window

    payload.update(result.diagnosis.to_payload())  # diagnosis + issues

    md = payload["metadata"]
    md["samples"] = window.steps_analyzed * max(1, len(window.ranks_used))
    md["global_ranks_seen"] = list(window.ranks_seen)
    md["global_ranks_used"] = list(window.ranks_used)
    md["training_total_steps"] = window.end_step
    md["training_latest_step"] = window.end_step
    nodes = {
        i.node_rank
        for i in window.identities.values()
        if i.node_rank is not None
    }
    md["nodes_observed"] = len(nodes) if nodes else (1 if window.has_data else None)
    md["gpus_observed"] = len(window.ranks_used) if window.clock == "gpu" else None
    md["mode"] = (
        "no_data"
        if not window.has_data
        else ("multi_node" if len(nodes) > 1 else "single_node")
    )

    g = payload["global"]
    g["window"] = {
        "kind": "step_window",
        "alignment": "common_steps",
        "samples": md["samples"],
        "steps_analyzed": window.steps_analyzed or None,
        "start_step": window.start_step,
        "end_step": window.end_step,
        "completed_step": window.end_step,
        "window_size": window.steps_analyzed or None,
    }

    per_rank = {
        str(rank): window.ranks[rank].as_dict() for rank in window.ranks_used
    }
    fill_metric_maps(payload, STEP_TIME_METRICS, per_rank)
    for rank in window.ranks_used:
        identity = window.identities.get(rank)
        if identity is not None:
            payload["groups"]["rows"][str(rank)]["identity"] = identity.as_dict()

    payload["evidence_extra"] = {
        "diagnosis_clock": window.clock,
        "signal_coverage": window.signal_coverage,
        "shares": window.shares,
        "cohorts": window.cohorts,
        "training_strategy": window.training_strategy,
    }
    comm = _latest_rank_stats(db_path)
    if comm:
        payload["evidence_extra"]["rccl_rank_stats"] = comm
    payload["card"] = _card(window, payload)
    return payload


def _latest_rank_stats(db_path: str) -> dict:
    """Most recent RCCL-over-xGMI gather (per-rank skew + gather latency)."""
    import json as _json
    import sqlite3

    try:
        conn = sqlite3.connect(f"file:{db_path}?mode=ro", uri=True)
        try:
            row = conn.execute(
                "SELECT timestamp, world_size_gathered, ranks_json, "
                "gather_latency_ms, gather_latency_ms_mean "
                "FROM rank_stats ORDER BY id DESC LIMIT 1"
            ).fetchone()
        finally:
            conn.close()
    except sqlite3.Error:
        return {}
    if not row or not row[2]:
        return {}
    try:
        ranks = _json.loads(row[2])
    except ValueError:
        return {}
    return {
        "timestamp": row[0],
        "world_size": row[1],
        "ranks": ranks,
        "gather_latency_ms": row[3],
        "gather_latency_ms_mean": row[4],
    }


def _card(window, payload) -> str:
    if not window.has_data:
        return "Step Time: no data."
    avg = window.average
    lines = [
        f"Step Time ({window.clock} clock, {window.steps_analyzed} aligned steps, "
        f"ranks {', '.join(str(r) for r in window.ranks_used)})"
    ]

    def fmt(metric, label):
        value = avg.get(metric)
        if value is None:
            return None
        share = None
        step = avg.get("step_time_ms")
        if step and metric not in ("step_time_ms",):
            share = f" ({value / step * 100.0:.1f}%)"
        return f"  {label:<14} {value:9.1f} ms{share or ''}"

    for metric, label in (
        ("step_time_ms", "Step"),
        ("input_wait_ms", "Input wait"),
        ("h2d_ms", "H2D"),
        ("forward_ms", "Forward"),
        ("backward_ms", "Backward"),
        ("optimizer_ms", "Optimizer"),
        ("ddp_comm_ms", "DDP comm"),
        ("residual_ms", "Residual"),
    ):
        line = fmt(metric, label)
        if line:
            lines.append(line)
    diag = payload.get("diagnosis") or {}
    if diag:
        lines.append(f"  Verdict: {diag.get('status')} — {diag.get('summary')}")
    return "\n".join(lines)
