"""Plain-dict view models over the pipelines + section loaders."""

from __future__ import annotations

from typing import Dict, Optional

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


def live_view(db_path: str, session=None) -> dict:
    """One payload for live surfaces; optionally freshness-bridged via a
    LiveStepTimeSession."""
    if session is not None:
        result, freshness = session.tick()
    else:
        from traceml_amd.steptime.pipeline import StepTimePipeline

        result = StepTimePipeline(db_path, profile="live").run()
        freshness = "live" if result.window.has_data else "cold"
    payload = {
        "freshness": freshness,
        "step_time": step_time_view(result.window, result.diagnosis),
        "memory": memory_view(db_path),
        "system": system_view(db_path),
    }
    return payload
