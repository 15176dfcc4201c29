"""Step-time diagnosis entry point (reference: diagnostics/step_time/api.py:213)."""

from __future__ import annotations

from traceml_amd.diagnostics.common import DiagnosticResult
from traceml_amd.diagnostics.step_time.rules import evaluate
from traceml_amd.steptime.model import StepTimeWindow


def diagnose_step_time_window(window: StepTimeWindow) -> DiagnosticResult:
    return DiagnosticResult(issues=evaluate(window))
