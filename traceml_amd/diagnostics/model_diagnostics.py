"""Model-diagnostics composer: combines the step-time and step-memory
results into one "model health" card for the dashboard surfaces
(reference: diagnostics/model_diagnostics.py:468, registry-driven)."""

from __future__ import annotations

from typing import Optional

from traceml_amd.diagnostics.common import (
    SEVERITY_ORDER,
    DiagnosticIssue,
    DiagnosticResult,
    sort_issues,
)


def compose_model_diagnostics(
    step_time: Optional[DiagnosticResult],
    step_memory: Optional[DiagnosticResult],
) -> DiagnosticResult:
    """Merge both domains; primary = most severe (step-time wins ties —
    performance beats health in the model card ordering)."""
    issues = []
    if step_time is not None:
        for issue in step_time.issues:
            tagged = DiagnosticIssue(**{**issue.to_payload()})
            tagged.evidence = {**issue.evidence, "domain": "step_time"}
            issues.append(tagged)
    if step_memory is not None:
        for issue in step_memory.issues:
            tagged = DiagnosticIssue(**{**issue.to_payload()})
            tagged.evidence = {**issue.evidence, "domain": "step_memory"}
            issues.append(tagged)
    if not issues:
        issues = [
            DiagnosticIssue(
                kind="NO_DATA", status="NO DATA", severity="info",
                summary="No model telemetry.", action="",
            )
        ]
    ordered = sorted(
        issues,
        key=lambda i: (
            -SEVERITY_ORDER.get(i.severity, 0),
            0 if i.evidence.get("domain") == "step_time" else 1,
            -(i.score if i.score is not None else -1.0),
        ),
    )
    return DiagnosticResult(issues=ordered)


def model_card(result: DiagnosticResult) -> str:
    primary = result.primary
    lines = [f"Model: {primary.status} [{primary.severity}] — {primary.summary}"]
    for issue in result.issues[1:4]:
        lines.append(f"  also: {issue.status} [{issue.severity}]")
    return "\n".join(lines)
