"""One load→analyze→diagnose path for every surface (CLI, summary, compare)
(reference: step_time/pipeline.py:71-277)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

from traceml_amd.diagnostics.common import DiagnosticResult
from traceml_amd.diagnostics.step_time.api import diagnose_step_time_window
from traceml_amd.steptime.analyzer import StepTimeAnalyzer
from traceml_amd.steptime.model import StepTimeWindow
from traceml_amd.steptime.repository import SQLiteStepTimeRepository


@dataclass
class StepTimePipelineResult:
    window: StepTimeWindow
    diagnosis: DiagnosticResult


class StepTimePipeline:
    def __init__(self, db_path: str, profile: str = "summary") -> None:
        self.repository = SQLiteStepTimeRepository(db_path)
        self.profile = profile

    def run(self) -> StepTimePipelineResult:
        rows = (
            self.repository.load_summary()
            if self.profile == "summary"
            else self.repository.load_live()
        )
        strategy = self.repository.load_training_strategy() or "ddp"
        window = StepTimeAnalyzer().analyze(rows, training_strategy=strategy)
        diagnosis = diagnose_step_time_window(window)
        return StepTimePipelineResult(window=window, diagnosis=diagnosis)


#: freshness states for live surfaces (reference: pipeline.py LiveStepTimeSession)
FRESH_COLD = "cold"  # never produced a window
FRESH_LIVE = "live"  # current tick produced a fresh window
FRESH_BRIDGED = "bridged"  # current tick empty; serving the last good window
FRESH_EXPIRED = "expired"  # last good window older than the TTL


class LiveStepTimeSession:
    """Live-surface wrapper: re-runs the pipeline per UI tick, bridges over
    transient empty reads with the last good result, and reports freshness
    so displays can dim stale data instead of flashing empty
    (reference: step_time/pipeline.py:160-277, 30 s TTL)."""

    def __init__(self, db_path: str, ttl_sec: float = 30.0) -> None:
        self.pipeline = StepTimePipeline(db_path, profile="live")
        self.ttl_sec = ttl_sec
        self._last_good: StepTimePipelineResult = None
        self._last_good_at: float = 0.0
        self._last_end_step = None
        # incremental cursor: (rank, step) -> newest SourceRow; id watermark
        self._cursor_id = 0
        self._rows: dict = {}

    def _incremental_rows(self):
        from traceml_amd.reporting.config import LIVE_WINDOW_ROWS

        new_rows, self._cursor_id = self.pipeline.repository.load_new_rows(
            self._cursor_id
        )
        for row in new_rows:
            key = (row.global_rank, row.step)
            old = self._rows.get(key)
            if old is None or row.row_id > old.row_id:
                self._rows[key] = row
        # bound the cache: keep the newest LIVE_WINDOW_ROWS steps per rank
        by_rank: dict = {}
        for (rank, step) in self._rows:
            by_rank.setdefault(rank, []).append(step)
        for rank, steps in by_rank.items():
            if len(steps) > LIVE_WINDOW_ROWS:
                steps.sort()
                for step in steps[: len(steps) - LIVE_WINDOW_ROWS]:
                    del self._rows[(rank, step)]
        return list(self._rows.values())

    def tick(self):
        """Returns (result, freshness). Incremental: each tick reads only
        rows appended since the last one (cursor reuse)."""
        import time as _time

        from traceml_amd.diagnostics.step_time.api import diagnose_step_time_window
        from traceml_amd.steptime.analyzer import StepTimeAnalyzer

        before = len(self._rows)
        before_cursor = self._cursor_id
        rows = self._incremental_rows()
        got_new = self._cursor_id > before_cursor or len(self._rows) != before
        strategy = (
            self.pipeline.repository.load_training_strategy() or "ddp"
        )
        window = StepTimeAnalyzer().analyze(rows, training_strategy=strategy)
        result = StepTimePipelineResult(
            window=window, diagnosis=diagnose_step_time_window(window)
        )
        now = _time.time()
        if not result.window.has_data:
            return result, FRESH_COLD
        # freshness = recency of NEW telemetry, not of the re-read: the
        # cached window keeps serving while we classify how stale it is
        if got_new or self._last_good is None:
            self._last_good = result
            self._last_good_at = now
            self._last_end_step = result.window.end_step
            return result, FRESH_LIVE
        self._last_good = result
        if now - self._last_good_at <= self.ttl_sec:
            return result, FRESH_BRIDGED
        return result, FRESH_EXPIRED
