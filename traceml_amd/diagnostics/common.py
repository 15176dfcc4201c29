"""Shared diagnostic types (reference: diagnostics/common.py:38-120).

``DiagnosticResult.issues`` is the canonical sorted list; ``issues[0]`` IS
the primary diagnosis (invariant enforced at construction). Neutral states
(BALANCED, NO_DATA, …) use the same issue shape as actionable findings.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

SEVERITY_ORDER = {"crit": 2, "warn": 1, "info": 0}


@dataclass
class DiagnosticIssue:
    kind: str
    status: str
    severity: str = "info"  # info | warn | crit
    summary: str = ""
    action: str = ""
    metric: Optional[str] = None
    phase: Optional[str] = None
    score: Optional[float] = None
    share_pct: Optional[float] = None
    skew_pct: Optional[float] = None
    ranks: List[int] = field(default_factory=list)
    evidence: Dict = field(default_factory=dict)

    def to_payload(self) -> dict:
        return {
            "kind": self.kind,
            "status": self.status,
            "severity": self.severity,
            "summary": self.summary,
            "action": self.action,
            "metric": self.metric,
            "phase": self.phase,
            "score": self.score,
            "share_pct": self.share_pct,
            "skew_pct": self.skew_pct,
            "ranks": list(self.ranks),
            "evidence": dict(self.evidence),
        }


@dataclass
class DiagnosticResult:
    issues: List[DiagnosticIssue]

    def __post_init__(self) -> None:
        if not self.issues:
            raise ValueError("DiagnosticResult requires at least one issue")

    @property
    def primary(self) -> DiagnosticIssue:
        return self.issues[0]

    def to_payload(self) -> dict:
        return {
            "diagnosis": self.primary.to_payload(),
            "issues": [i.to_payload() for i in self.issues],
        }


def sort_issues(issues: List[DiagnosticIssue]) -> List[DiagnosticIssue]:
    """Severity first, then score (desc), stable otherwise."""
    return sorted(
        issues,
        key=lambda i: (
            -SEVERITY_ORDER.get(i.severity, 0),
            -(i.score if i.score is not None else -1.0),
        ),
    )
