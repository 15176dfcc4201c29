"""Plain-dict view models over the pipelines + section loaders."""

from __future__ import annotations

from typing import Dict

#: display order + labels for the per-rank phase table
STEP_TIME_TABLE_METRICS = (
    ("step_time_ms", "step (ms)"),
    ("input_wait_ms", "input"),
    ("h2d_ms", "h2d"),
    ("forward_ms", "forward"),
    ("backward_ms", "backward"),
    ("optimizer_ms", "optimizer"),
    ("ddp_comm_ms", "ddp comm"),
    ("residual_ms", "residual"),
)


def step_time_view(window, diagnosis) -> dict:
    return {
        "diagnosis": diagnosis.primary.to_payload(),
        "issues": [i.to_payload() for i in diagnosis.issues],
        "steps_analyzed": window.steps_analyzed,
        "clock": window.clock,
        "strategy": window.training_strategy,
        "shares": window.shares,
        "ranks": {str(r): window.ranks[r].as_dict() for r in window.ranks_used},
        "table_metrics": [m for m, _ in STEP_TIME_TABLE_METRICS],
    }


def memory_view(db_path: str) -> Dict[str, dict]:
    from traceml_amd.diagnostics.step_memory.api import load_memory_series

    out: Dict[str, dict] = {}
    for rank, series in load_memory_series(db_path).items():
        alloc = [v for v in series.peak_allocated if v is not None]
        reserved = [v for v in series.peak_reserved if v is not None]
        out[str(rank)] = {
            "alloc": max(alloc) if alloc else None,
            "reserved": max(reserved) if reserved else None,
            "capacity": series.capacity,
        }
    return out


def system_view(db_path: str) -> dict:
    from traceml_amd.diagnostics.system.api import load_system_context

    ctx = load_system_context(db_path)
    return {
        "cpu_percent": ctx.cpu_percent_avg,
        "ram_percent": ctx.ram_percent_avg,
        "gpus": {
            str(i): {
                "util": g.get("util"),
                "mem_used": g.get("mem_used"),
                "mem_total": g.get("mem_total"),
                "temp": g.get("temp"),
                "power": g.get("power"),
            }
            for i, g in ctx.gpus.items()
        },
    }


def issues_view(db_path: str, step_time_diagnosis) -> list:
    """All sections' current findings, one flat severity-sorted list."""
    from traceml_amd.diagnostics.common import SEVERITY_ORDER
    from traceml_amd.diagnostics.process.api import (
        diagnose_process,
        load_process_context,
    )
    from traceml_amd.diagnostics.step_memory.api import (
        diagnose_step_memory,
        load_memory_series,
    )
    from traceml_amd.diagnostics.system.api import (
        diagnose_system,
        load_system_context,
    )

    issues = []
    for section, result in (
        ("step_time", step_time_diagnosis),
        ("step_memory", diagnose_step_memory(load_memory_series(db_path))),
        ("system", diagnose_system(load_system_context(db_path))),
        ("process", diagnose_process(load_process_context(db_path))),
    ):
        for issue in result.issues:
            payload = issue.to_payload()
            payload["section"] = section
            issues.append(payload)
    issues.sort(key=lambda i: -SEVERITY_ORDER.get(i.get("severity"), 0))
    return issues


def stdout_tail_view(db_path: str, n: int = 12) -> list:
    import sqlite3

    try:
        conn = sqlite3.connect(f"file:{db_path}?mode=ro", uri=True)
        try:
            rows = conn.execute(
                "SELECT stream, line FROM stdout_stderr ORDER BY id DESC LIMIT ?",
                (n,),
            ).fetchall()
        finally:
            conn.close()
        return [{"stream": s, "line": l} for s, l in reversed(rows)]
    except sqlite3.Error:
        return []


def comm_view(db_path: str) -> dict:
    """Latest RCCL rank-stats gather (xGMI latency + per-rank skew)."""
    import json as _json
    import sqlite3

    try:
        conn = sqlite3.connect(f"file:{db_path}?mode=ro", uri=True)
        try:
            row = conn.execute(
                "SELECT ranks_json FROM rank_stats ORDER BY id DESC LIMIT 1"
            ).fetchone()
        finally:
            conn.close()
        if not row or not row[0]:
            return {}
        return {"ranks": _json.loads(row[0])}
    except (sqlite3.Error, ValueError):
        return {}


def history_view(window, max_points: int = 120) -> dict:
    """Per-rank step-time series for charts: {rank: [[step, ms], ...]}."""
    out = {}
    for rank, series in window.step_series.items():
        out[str(rank)] = [[s, round(ms, 3)] for s, ms in series[-max_points:]]
    return out


def live_view(db_path: str, session=None) -> dict:
    """One payload for live surfaces; optionally freshness-bridged via a
    LiveStepTimeSession. ``sections`` carries the per-section renderer view
    models (renderers/{step_time,step_memory,system,process,comm}.py) that
    the CLI, dashboard and HTML report all share; the flat top-level keys
    are kept for backward compatibility."""
    if session is not None:
        result, freshness = session.tick()
    else:
        from traceml_amd.steptime.pipeline import StepTimePipeline

        result = StepTimePipeline(db_path, profile="live").run()
        freshness = "live" if result.window.has_data else "cold"

    from traceml_amd.diagnostics.process.api import (
        diagnose_process,
        load_process_context,
    )
    from traceml_amd.diagnostics.step_memory.api import (
        diagnose_step_memory,
        load_memory_series,
    )
    from traceml_amd.diagnostics.system.api import (
        diagnose_system,
        load_system_context,
    )
    from traceml_amd.renderers.comm import load_latest_gather, render_comm
    from traceml_amd.renderers.process import render_process
    from traceml_amd.renderers.step_memory import render_step_memory
    from traceml_amd.renderers.step_time import render_step_time
    from traceml_amd.renderers.system import render_system

    from traceml_amd.diagnostics.model_diagnostics import (
        compose_model_diagnostics,
    )

    memory_series = load_memory_series(db_path)
    memory_diagnosis = diagnose_step_memory(memory_series)
    system_ctx = load_system_context(db_path)
    process_ctx = load_process_context(db_path)
    model_combined = compose_model_diagnostics(
        result.diagnosis, memory_diagnosis
    )
    sections = {
        "step_time": render_step_time(result.window, result.diagnosis),
        "step_memory": render_step_memory(memory_series, memory_diagnosis),
        "system": render_system(system_ctx, diagnose_system(system_ctx)),
        "process": render_process(process_ctx, diagnose_process(process_ctx)),
        "comm": render_comm(load_latest_gather(db_path)),
        # combined model-health card: most severe of step_time/step_memory
        # with the originating domain in evidence (the reference's
        # model-diagnostics dashboard card)
        "model": {
            "section": "model",
            "available": result.window.has_data,
            "diagnosis": model_combined.primary.to_payload(),
            "issues": [i.to_payload() for i in model_combined.issues],
        },
    }
    payload = {
        "freshness": freshness,
        "sections": sections,
        "step_time": step_time_view(result.window, result.diagnosis),
        "memory": memory_view(db_path),
        "system": system_view(db_path),
        "issues": issues_view(db_path, result.diagnosis),
        "stdout": stdout_tail_view(db_path),
        "comm": comm_view(db_path),
        "history": history_view(result.window),
    }
    return payload
