"""SQLite step-time repository: snapshot reads for live + summary surfaces
(reference: step_time/sqlite.py:85-264).

``load_live`` reads a bounded per-rank tail; ``load_summary`` reads the full
retention window. Event payloads are restricted-JSON decoded; rows that fail
to decode are skipped (never crash a surface on a bad row).
"""

from __future__ import annotations

import json
import sqlite3
from typing import List, Optional

from traceml_amd.reporting.config import LIVE_WINDOW_ROWS, SUMMARY_WINDOW_ROWS
from traceml_amd.steptime.model import STEP_TIME_EVENT_NAMES, StepTimeSourceRow


def _connect(db_path: str) -> sqlite3.Connection:
    conn = sqlite3.connect(f"file:{db_path}?mode=ro", uri=True)
    conn.row_factory = sqlite3.Row
    return conn


def normalize_step_time_events(raw: Optional[str]) -> Optional[dict]:
    """events_json -> {signal_key: {duration_ms, cpu_ms, gpu_ms, n_calls, is_gpu}}.

    Hot path for summary loads (10k+ rows): json values are already
    numbers/None from our own writer, so coercion only runs for the
    defensive string/garbage case.
    """
    if not raw:
        return None
    try:
        decoded = json.loads(raw)
    except (ValueError, TypeError):
        return None
    if type(decoded) is not dict:
        return None
    events = {}
    names = STEP_TIME_EVENT_NAMES
    for wire_name, cell in decoded.items():
        signal = names.get(wire_name)
        if signal is None and wire_name.startswith("_traceml_user:"):
            # custom trace_time regions survive as "user:<name>" signals —
            # ignored by the analyzer's derivations, visible to inspect /
            # export-trace / custom tooling
            signal = "user:" + wire_name.split(":", 1)[1]
        if signal is None or type(cell) is not dict:
            continue
        duration = cell.get("duration_ms")
        cpu = cell.get("cpu_ms")
        gpu = cell.get("gpu_ms")
        # fast path: our own writer emits numbers/None; anything else
        # (strings, lists, bools from a hostile peer) goes through _num
        if duration is not None and type(duration) not in (int, float):
            duration = _num(duration)
        if cpu is not None and type(cpu) not in (int, float):
            cpu = _num(cpu)
        if gpu is not None and type(gpu) not in (int, float):
            gpu = _num(gpu)
        events[signal] = {
            "duration_ms": duration,
            "cpu_ms": cpu,
            "gpu_ms": gpu,
            "n_calls": cell.get("n_calls") or 0,
            "is_gpu": bool(cell.get("is_gpu")),
        }
    return events


def _num(value) -> Optional[float]:
    if value is None:
        return None
    try:
        return float(value)
    except (TypeError, ValueError):
        return None


class SQLiteStepTimeRepository:
    def __init__(self, db_path: str) -> None:
        self.db_path = db_path

    def _load(self, per_rank_limit: int) -> List[StepTimeSourceRow]:
        try:
            conn = _connect(self.db_path)
        except sqlite3.Error:
            return []
        try:
            try:
                cursor = conn.execute(
                    """
                    SELECT id, global_rank, local_rank, node_rank, hostname,
                           world_size, local_world_size, timestamp, step,
                           events_json
                    FROM (
                      SELECT *, ROW_NUMBER() OVER (
                        PARTITION BY global_rank ORDER BY id DESC
                      ) AS rn FROM step_time_samples
                    ) WHERE rn <= ? ORDER BY global_rank, step
                    """,
                    (per_rank_limit,),
                )
                raw_rows = cursor.fetchall()
            except sqlite3.Error:
                return []
        finally:
            conn.close()
        rows: List[StepTimeSourceRow] = []
        for r in raw_rows:
            events = normalize_step_time_events(r["events_json"])
            if events is None or r["step"] is None or r["global_rank"] is None:
                continue
            rows.append(
                StepTimeSourceRow(
                    row_id=int(r["id"]),
                    global_rank=int(r["global_rank"]),
                    step=int(r["step"]),
                    timestamp=float(r["timestamp"] or 0.0),
                    events=events,
                    node_rank=r["node_rank"],
                    local_rank=r["local_rank"],
                    hostname=r["hostname"],
                    world_size=r["world_size"],
                    local_world_size=r["local_world_size"],
                )
            )
        return rows

    def load_live(self) -> List[StepTimeSourceRow]:
        return self._load(LIVE_WINDOW_ROWS)

    def load_new_rows(self, since_id: int) -> tuple:
        """Incremental live read: rows with id > since_id, ordered by id.
        Returns (rows, max_id_seen). Powers the live session's cursor reuse
        so a UI tick costs only the new rows, not a full window re-read."""
        try:
            conn = _connect(self.db_path)
        except sqlite3.Error:
            return ([], since_id)
        try:
            try:
                raw_rows = conn.execute(
                    """
                    SELECT id, global_rank, local_rank, node_rank, hostname,
                           world_size, local_world_size, timestamp, step,
                           events_json
                    FROM step_time_samples WHERE id > ? ORDER BY id
                    """,
                    (since_id,),
                ).fetchall()
            except sqlite3.Error:
                return ([], since_id)
        finally:
            conn.close()
        rows: List[StepTimeSourceRow] = []
        max_id = since_id
        for r in raw_rows:
            max_id = max(max_id, int(r["id"]))
            events = normalize_step_time_events(r["events_json"])
            if events is None or r["step"] is None or r["global_rank"] is None:
                continue
            rows.append(
                StepTimeSourceRow(
                    row_id=int(r["id"]),
                    global_rank=int(r["global_rank"]),
                    step=int(r["step"]),
                    timestamp=float(r["timestamp"] or 0.0),
                    events=events,
                    node_rank=r["node_rank"],
                    local_rank=r["local_rank"],
                    hostname=r["hostname"],
                    world_size=r["world_size"],
                    local_world_size=r["local_world_size"],
                )
            )
        return (rows, max_id)

    def load_summary(self) -> List[StepTimeSourceRow]:
        return self._load(SUMMARY_WINDOW_ROWS)

    def load_training_strategy(self) -> Optional[str]:
        try:
            conn = _connect(self.db_path)
        except sqlite3.Error:
            return None
        try:
            try:
                row = conn.execute(
                    "SELECT training_strategy FROM runtime_environment "
                    "ORDER BY id DESC LIMIT 1"
                ).fetchone()
            except sqlite3.Error:
                return None
        finally:
            conn.close()
        return row["training_strategy"] if row else None
