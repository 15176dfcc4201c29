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

    def tick(self):
        """Returns (result, freshness)."""
        import time as _time

        result = self.pipeline.run()
        now = _time.time()
        if result.window.has_data:
            # new data, or same window re-read — both count as live
            self._last_good = result
            self._last_good_at = now
            self._last_end_step = result.window.end_step
            return result, FRESH_LIVE
        if self._last_good is None:
            return result, FRESH_COLD
        if now - self._last_good_at <= self.ttl_sec:
            return self._last_good, FRESH_BRIDGED
        return self._last_good, FRESH_EXPIRED
