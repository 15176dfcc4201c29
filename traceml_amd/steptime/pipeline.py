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
