"""Comm section renderer — the RCCL/xGMI rank-stats card (new MI355X
capability; no reference counterpart — the reference has no comm plane of
its own, SURVEY §5 "Distributed communication backend").

Latest gathered matrix: per-rank step/input/backward/ddp-comm times, xGMI
all-gather latency (last + running mean), input/step skew across ranks and
the straggler candidate (slowest visible rank).
"""

from __future__ import annotations

import json as _json
import sqlite3
from typing import Optional


def load_latest_gather(db_path: str) -> Optional[dict]:
    try:
        conn = sqlite3.connect(f"file:{db_path}?mode=ro", uri=True)
        try:
            row = conn.execute(
                "SELECT timestamp, world_size_gathered, gather_latency_ms, "
                "gather_latency_ms_mean, ranks_json FROM rank_stats "
                "ORDER BY id DESC LIMIT 1"
            ).fetchone()
        finally:
            conn.close()
    except sqlite3.Error:
        return None
    if not row or not row[4]:
        return None
    try:
        ranks = _json.loads(row[4])
    except ValueError:
        return None
    return {
        "timestamp": row[0],
        "world_size": row[1],
        "gather_latency_ms": row[2],
        "gather_latency_ms_mean": row[3],
        "ranks": ranks,
    }


def _spread(values) -> Optional[dict]:
    present = [v for v in values if v is not None]
    if len(present) < 2:
        return None
    lo, hi = min(present), max(present)
    return {"min": lo, "max": hi, "spread_ms": hi - lo}


def render_comm(gather: Optional[dict]) -> dict:
    """gather: output of load_latest_gather (or a drained exchange row)."""
    if not gather or not gather.get("ranks"):
        return {"section": "comm", "available": False}
    ranks = gather["ranks"]
    rows = [
        {
            "rank": str(int(r.get("rank", i))),
            "step": None if r.get("step") is None else int(r["step"]),
            "input_ms": r.get("input_ms"),
            "forward_ms": r.get("forward_ms"),
            "backward_ms": r.get("backward_ms"),
            "optimizer_ms": r.get("optimizer_ms"),
            "step_ms": r.get("step_ms"),
            "ddp_comm_ms": r.get("ddp_comm_ms"),
            "peak_alloc_bytes": (
                None
                if not r.get("peak_alloc_bytes")
                else int(r["peak_alloc_bytes"])
            ),
        }
        for i, r in enumerate(ranks)
    ]
    step_values = [r["step_ms"] for r in rows]
    slowest = None
    present = [(r["rank"], r["step_ms"]) for r in rows if r["step_ms"] is not None]
    if len(present) >= 2:
        slowest = max(present, key=lambda rv: rv[1])[0]
    return {
        "section": "comm",
        "available": True,
        "world_size": gather.get("world_size"),
        "gather_latency_ms": gather.get("gather_latency_ms"),
        "gather_latency_ms_mean": gather.get("gather_latency_ms_mean"),
        "rows": rows,
        "input_skew": _spread([r["input_ms"] for r in rows]),
        "step_skew": _spread(step_values),
        "slowest_rank": slowest,
    }
