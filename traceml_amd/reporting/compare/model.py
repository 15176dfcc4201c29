"""Typed compare contracts (reference: reporting/compare/model.py:107).

``CompareMetric`` is one metric compared across two runs; ``CompareSection``
groups the metrics + diagnosis transition of one summary section; the
verdict layer consumes these without re-reading the summaries.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

#: metric movement statuses
REGRESSION = "REGRESSION"
IMPROVEMENT = "IMPROVEMENT"
NEUTRAL = "NEUTRAL"
INCOMPARABLE = "INCOMPARABLE"
CONTEXT = "CONTEXT"  # informational direction (system util etc.)

#: significance grades, ordered
NEGLIGIBLE = "negligible"
MODERATE = "moderate"
MATERIAL = "material"
SIGNIFICANCE_ORDER = {NEGLIGIBLE: 0, MODERATE: 1, MATERIAL: 2}


@dataclass
class CompareMetric:
    key: str
    label: str
    unit: str  # "ms" | "bytes" | "percent" | "fraction"
    direction: str  # "higher_is_worse" | "context"
    baseline: Optional[float]
    candidate: Optional[float]
    status: str = INCOMPARABLE
    significance: Optional[str] = None

    @property
    def delta(self) -> Optional[float]:
        if self.baseline is None or self.candidate is None:
            return None
        return self.candidate - self.baseline

    @property
    def pct(self) -> Optional[float]:
        if self.delta is None or not self.baseline:
            return None
        return self.delta / self.baseline * 100.0

    def to_payload(self) -> dict:
        return {
            "key": self.key,
            "label": self.label,
            "unit": self.unit,
            "direction": self.direction,
            "baseline": self.baseline,
            "candidate": self.candidate,
            "delta": self.delta,
            "pct": self.pct,
            "status": self.status,
            "significance": self.significance,
        }


@dataclass
class DiagnosisTransition:
    from_kind: Optional[str]
    from_severity: Optional[str]
    to_kind: Optional[str]
    to_severity: Optional[str]
    direction: str = "unchanged"  # resolved|improved|worsened|changed|unchanged

    def to_payload(self) -> dict:
        return {
            "from": {"kind": self.from_kind, "severity": self.from_severity},
            "to": {"kind": self.to_kind, "severity": self.to_severity},
            "direction": self.direction,
        }


@dataclass
class CompareSection:
    name: str
    available: bool
    metrics: Dict[str, CompareMetric] = field(default_factory=dict)
    diagnosis: Optional[DiagnosisTransition] = None
    notes: List[str] = field(default_factory=list)

    def metric(self, key: str) -> Optional[CompareMetric]:
        return self.metrics.get(key)

    def to_payload(self) -> dict:
        return {
            "available": self.available,
            "diagnosis_transition": (
                self.diagnosis.to_payload() if self.diagnosis else None
            ),
            "metrics": {k: m.to_payload() for k, m in self.metrics.items()},
            "notes": list(self.notes),
        }
