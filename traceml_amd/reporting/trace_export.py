"""Chrome-trace export: ``traceml-amd export-trace <telemetry.sqlite> -o trace.json``.

Renders the per-rank step-time history as a chrome://tracing / Perfetto
timeline — one process row per rank, phases laid out sequentially inside
each step envelope. Phase layout inside a step is reconstructed from the
aggregated per-step durations (input → h2d → forward → backward →
optimizer; ddp_comm on its own lane since it overlaps backward), anchored
at the step's flush timestamp, so cross-rank skew is visible at a glance
even though sub-phase start offsets are not stored.
"""

from __future__ import annotations

import json
from typing import List, Optional

from traceml_amd.steptime.repository import SQLiteStepTimeRepository

_SEQUENTIAL_PHASES = [
    ("dataloader", "input_wait"),
    ("h2d", "h2d"),
    ("forward", "forward"),
    ("backward", "backward"),
    ("optimizer", "optimizer"),
]

_PHASE_COLORS = {
    "input_wait": "thread_state_iowait",
    "h2d": "thread_state_runnable",
    "forward": "thread_state_running",
    "backward": "rail_animation",
    "optimizer": "rail_response",
    "ddp_comm": "rail_idle",
    "step": "generic_work",
}


def _duration_ms(events: dict, signal: str) -> Optional[float]:
    cell = events.get(signal)
    if cell is None:
        return None
    value = cell.get("gpu_ms")
    if value is None:
        value = cell.get("cpu_ms")
    return value


def build_chrome_trace(db_path: str, max_steps: Optional[int] = None) -> dict:
    repo = SQLiteStepTimeRepository(db_path)
    rows = repo.load_summary()
    if max_steps is not None:
        by_rank: dict = {}
        for row in rows:
            by_rank.setdefault(row.global_rank, []).append(row)
        rows = [r for rs in by_rank.values() for r in rs[-max_steps:]]

    events: List[dict] = []
    for row in rows:
        pid = row.global_rank
        traced = _duration_ms(row.events, "traced") or 0.0
        input_ms = _duration_ms(row.events, "dataloader") or 0.0
        step_total = input_ms + traced
        # anchor: the flush timestamp marks the END of the step
        start_us = (row.timestamp - step_total / 1000.0) * 1e6
        events.append(
            {
                "name": f"step {row.step}",
                "ph": "X",
                "ts": start_us,
                "dur": step_total * 1000.0,
                "pid": pid,
                "tid": 0,
                "cname": _PHASE_COLORS["step"],
                "args": {"step": row.step},
            }
        )
        cursor_us = start_us
        for signal, label in _SEQUENTIAL_PHASES:
            ms = _duration_ms(row.events, signal)
            if ms is None or ms <= 0:
                continue
            events.append(
                {
                    "name": label,
                    "ph": "X",
                    "ts": cursor_us,
                    "dur": ms * 1000.0,
                    "pid": pid,
                    "tid": 1,
                    "cname": _PHASE_COLORS.get(label, "generic_work"),
                    "args": {"step": row.step},
                }
            )
            cursor_us += ms * 1000.0
        ddp = _duration_ms(row.events, "ddp_comm")
        if ddp:
            # overlaps backward: right-aligned to the step end on lane 2
            events.append(
                {
                    "name": "ddp_comm",
                    "ph": "X",
                    "ts": start_us + (step_total - ddp) * 1000.0,
                    "dur": ddp * 1000.0,
                    "pid": pid,
                    "tid": 2,
                    "cname": _PHASE_COLORS["ddp_comm"],
                    "args": {"step": row.step},
                }
            )

    ranks = sorted({e["pid"] for e in events})
    meta: List[dict] = []
    for rank in ranks:
        meta.append(
            {
                "name": "process_name",
                "ph": "M",
                "pid": rank,
                "args": {"name": f"rank {rank}"},
            }
        )
        for tid, name in ((0, "step"), (1, "phases"), (2, "comm")):
            meta.append(
                {
                    "name": "thread_name",
                    "ph": "M",
                    "pid": rank,
                    "tid": tid,
                    "args": {"name": name},
                }
            )
    return {"traceEvents": meta + events, "displayTimeUnit": "ms"}


def export_chrome_trace(
    db_path: str, out_path: str, max_steps: Optional[int] = None
) -> int:
    trace = build_chrome_trace(db_path, max_steps=max_steps)
    with open(out_path, "w", encoding="utf-8") as f:
        json.dump(trace, f)
    return len(trace["traceEvents"])
