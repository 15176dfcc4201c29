"""Shared helpers for per-section comparers (reference:
reporting/compare/sections/base.py)."""

from __future__ import annotations

from typing import Any, Dict, Optional

from traceml_amd.reporting.compare import policy
from traceml_amd.reporting.compare.model import (
    CONTEXT,
    IMPROVEMENT,
    INCOMPARABLE,
    NEGLIGIBLE,
    NEUTRAL,
    REGRESSION,
    CompareMetric,
    DiagnosisTransition,
)

_SEVERITY_RANK = {"info": 0, "warn": 1, "crit": 2}


def global_average(section: Optional[dict], metric: str) -> Optional[float]:
    if not isinstance(section, dict):
        return None
    value = section.get("global", {}).get("average", {}).get(metric)
    return float(value) if isinstance(value, (int, float)) else None


def section_available(lhs: Optional[dict], rhs: Optional[dict]) -> bool:
    def has_data(section):
        if not isinstance(section, dict):
            return False
        avg = section.get("global", {}).get("average", {})
        return any(v is not None for v in avg.values()) if avg else False

    return has_data(lhs) or has_data(rhs)


def diagnosis_transition(
    lhs: Optional[dict], rhs: Optional[dict]
) -> DiagnosisTransition:
    def diag(section):
        d = (section or {}).get("diagnosis")
        return d if isinstance(d, dict) else {}

    b, c = diag(lhs), diag(rhs)
    b_sev = _SEVERITY_RANK.get(b.get("severity"), 0)
    c_sev = _SEVERITY_RANK.get(c.get("severity"), 0)
    if c_sev < b_sev:
        direction = "resolved" if c_sev == 0 else "improved"
    elif c_sev > b_sev:
        direction = "worsened"
    elif b.get("kind") != c.get("kind"):
        direction = "changed"
    else:
        direction = "unchanged"
    return DiagnosisTransition(
        from_kind=b.get("kind"),
        from_severity=b.get("severity"),
        to_kind=c.get("kind"),
        to_severity=c.get("severity"),
        direction=direction,
    )


def _status(
    metric: CompareMetric, significance: Optional[str]
) -> str:
    if metric.delta is None:
        return INCOMPARABLE
    if metric.direction == "context":
        return CONTEXT
    if significance in (None, NEGLIGIBLE):
        return NEUTRAL
    return REGRESSION if metric.delta > 0 else IMPROVEMENT


def time_metric(
    key: str, label: str, lhs: Optional[float], rhs: Optional[float],
    direction: str = "higher_is_worse",
) -> CompareMetric:
    m = CompareMetric(key, label, "ms", direction, lhs, rhs)
    m.significance = policy.significance_time(m.delta, m.baseline)
    m.status = _status(m, m.significance)
    return m


def bytes_metric(
    key: str, label: str, lhs: Optional[float], rhs: Optional[float],
    direction: str = "higher_is_worse",
) -> CompareMetric:
    m = CompareMetric(key, label, "bytes", direction, lhs, rhs)
    m.significance = policy.significance_bytes(m.delta)
    m.status = _status(m, m.significance)
    return m


def points_metric(
    key: str, label: str, lhs: Optional[float], rhs: Optional[float],
    direction: str = "context",
) -> CompareMetric:
    m = CompareMetric(key, label, "percent", direction, lhs, rhs)
    m.significance = policy.significance_points(m.delta)
    m.status = _status(m, m.significance)
    return m
