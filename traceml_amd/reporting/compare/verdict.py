"""Rule-chain verdict selection (reference: reporting/compare/verdict.py:506).

A fixed priority ladder picks the primary compare finding; earlier rules
win. The ladder is conservative: data problems beat conclusions, mixed
signals beat single-family conclusions, measured step-time movement beats
memory movement, and diagnosis-rank changes (BALANCED → STRAGGLER) only
decide when no metric moved materially.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional

from traceml_amd.reporting.compare import policy
from traceml_amd.reporting.compare.model import (
    IMPROVEMENT,
    MATERIAL,
    MODERATE,
    REGRESSION,
    CompareSection,
)


class VerdictPriority:
    MISSING_PRIMARY_SIGNALS = 1
    PARTIAL_PRIMARY_SIGNALS = 2
    MIXED_PRIMARY_SIGNALS = 3
    STEP_TIME_REGRESSION = 4
    STEP_TIME_IMPROVEMENT = 5
    STEP_MEMORY_REGRESSION = 6
    STEP_MEMORY_IMPROVEMENT = 7
    STEP_TIME_MODERATE = 8
    STEP_TIME_DIAGNOSIS_REGRESSION = 9
    STEP_MEMORY_DIAGNOSIS_REGRESSION = 10
    EQUIVALENT = 11


@dataclass
class VerdictFinding:
    verdict: str  # REGRESSION | IMPROVEMENT | NEUTRAL | MIXED | INCOMPARABLE
    priority: int
    significance: Optional[str]
    title: str
    detail: str

    def to_payload(self) -> dict:
        return {
            "verdict": self.verdict,
            "priority": self.priority,
            "significance": self.significance,
            "title": self.title,
            "detail": self.detail,
        }


def _fmt_pct(pct: Optional[float]) -> str:
    if pct is None:
        return "n/a"
    sign = "+" if pct >= 0 else ""
    return f"{sign}{pct:.1f}%"


def _step_metric(sections: Dict[str, CompareSection]):
    st = sections.get("step_time")
    return st.metric("step_time_ms") if st else None


def _memory_material_direction(
    sections: Dict[str, CompareSection]
) -> Optional[str]:
    sm = sections.get("step_memory")
    if not sm:
        return None
    for key in ("peak_reserved_bytes", "peak_allocated_bytes",
                "peak_reserved_bytes_worst", "peak_allocated_bytes_worst"):
        m = sm.metric(key)
        if m and m.significance == MATERIAL and m.status in (
            REGRESSION, IMPROVEMENT,
        ):
            return m.status
    return None


def decide_verdict(sections: Dict[str, CompareSection]) -> VerdictFinding:
    step = _step_metric(sections)
    st_section = sections.get("step_time")

    # 1. no step-time signal at all
    if st_section is None or not st_section.available:
        return VerdictFinding(
            "INCOMPARABLE", VerdictPriority.MISSING_PRIMARY_SIGNALS, None,
            "No step-time signal",
            "Neither run carries step-time measurements; nothing to compare.",
        )
    # 2. only one side measured
    if step is None or step.delta is None:
        return VerdictFinding(
            "INCOMPARABLE", VerdictPriority.PARTIAL_PRIMARY_SIGNALS, None,
            "Partial step-time signal",
            "Only one run carries a comparable step time "
            "(warmup-only run, or clocks with no common measurement).",
        )

    step_material = step.significance == MATERIAL and step.status in (
        REGRESSION, IMPROVEMENT,
    )
    memory_dir = _memory_material_direction(sections)

    # 3. materially opposite movements in the two primary families
    if step_material and memory_dir and memory_dir != step.status:
        return VerdictFinding(
            "MIXED", VerdictPriority.MIXED_PRIMARY_SIGNALS, MATERIAL,
            "Mixed signals",
            f"Step time moved {_fmt_pct(step.pct)} ({step.status.lower()}) "
            f"while peak memory shows a material {memory_dir.lower()}.",
        )
    # 4/5. material step-time movement decides
    if step_material:
        priority = (
            VerdictPriority.STEP_TIME_REGRESSION
            if step.status == REGRESSION
            else VerdictPriority.STEP_TIME_IMPROVEMENT
        )
        return VerdictFinding(
            step.status, priority, MATERIAL,
            f"Step time {step.status.lower()}",
            f"Average step time moved {_fmt_pct(step.pct)} "
            f"({step.baseline:.1f} → {step.candidate:.1f} ms).",
        )
    # 6/7. material memory movement
    if memory_dir:
        sm = sections["step_memory"]
        m = next(
            m for m in sm.metrics.values()
            if m.significance == MATERIAL and m.status == memory_dir
        )
        priority = (
            VerdictPriority.STEP_MEMORY_REGRESSION
            if memory_dir == REGRESSION
            else VerdictPriority.STEP_MEMORY_IMPROVEMENT
        )
        gib = (m.delta or 0) / (1 << 30)
        return VerdictFinding(
            memory_dir, priority, MATERIAL,
            f"Peak memory {memory_dir.lower()}",
            f"{m.label} moved {gib:+.2f} GiB with step time steady.",
        )
    # 8. moderate step-time movement
    if step.significance == MODERATE and step.status in (
        REGRESSION, IMPROVEMENT,
    ):
        return VerdictFinding(
            step.status, VerdictPriority.STEP_TIME_MODERATE, MODERATE,
            f"Step time {step.status.lower()} (moderate)",
            f"Average step time moved {_fmt_pct(step.pct)} — above noise, "
            "below the material threshold "
            f"({policy.STEP_PCT_MATERIAL:.0f}%).",
        )
    # 9/10. diagnosis-rank movement with steady metrics
    for name, prio in (
        ("step_time", VerdictPriority.STEP_TIME_DIAGNOSIS_REGRESSION),
        ("step_memory", VerdictPriority.STEP_MEMORY_DIAGNOSIS_REGRESSION),
    ):
        section = sections.get(name)
        diag = section.diagnosis if section else None
        if diag is None:
            continue
        from_rank = policy.kind_rank(name, diag.from_kind)
        to_rank = policy.kind_rank(name, diag.to_kind)
        if to_rank > from_rank:
            return VerdictFinding(
                REGRESSION, prio, MODERATE,
                f"{name.replace('_', ' ')} diagnosis worsened",
                f"Diagnosis moved {diag.from_kind} → {diag.to_kind} with "
                "metric averages inside the noise thresholds.",
            )
    # 11. equivalent
    return VerdictFinding(
        "NEUTRAL", VerdictPriority.EQUIVALENT, None,
        "No significant change",
        f"Step time moved {_fmt_pct(step.pct)} — within the "
        f"{policy.STEP_PCT_MODERATE:.0f}% noise threshold; no material "
        "memory movement.",
    )
