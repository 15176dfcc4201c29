"""``traceml-amd compare A.json B.json`` (reference: reporting/compare/*, ~2k LoC).

Schema-version-aware reader, per-metric significance classification, and a
rule-chain verdict: REGRESSION / IMPROVEMENT / NEUTRAL / MIXED.
"""

from __future__ import annotations

import json
import sys
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

from traceml_amd.reporting.compare import policy

_COMPARED_METRICS: List[Tuple[str, str, str]] = [
    # (section, metric, unit)
    ("step_time", "step_time_ms", "ms"),
    ("step_time", "input_wait_ms", "ms"),
    ("step_time", "h2d_ms", "ms"),
    ("step_time", "compute_ms", "ms"),
    ("step_time", "forward_ms", "ms"),
    ("step_time", "backward_ms", "ms"),
    ("step_time", "optimizer_ms", "ms"),
    ("step_time", "ddp_comm_ms", "ms"),
    ("step_time", "residual_ms", "ms"),
    ("step_memory", "peak_allocated_bytes", "bytes"),
    ("step_memory", "peak_reserved_bytes", "bytes"),
]


@dataclass
class MetricDelta:
    section: str
    metric: str
    unit: str
    baseline: Optional[float]
    candidate: Optional[float]
    status: str  # REGRESSION | IMPROVEMENT | NEUTRAL | INCOMPARABLE

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


def load_summary(path: str) -> dict:
    with open(path, "r", encoding="utf-8") as f:
        payload = json.load(f)
    version = payload.get("schema_version")
    if version is None:
        raise ValueError(f"{path}: not a traceml final summary (no schema_version)")
    return payload


def _metric_value(payload: dict, section: str, metric: str) -> Optional[float]:
    value = payload.get(section, {}).get("global", {}).get("average", {}).get(metric)
    return float(value) if isinstance(value, (int, float)) else None


def _classify(delta: Optional[float], baseline: Optional[float], unit: str) -> str:
    if delta is None:
        return "INCOMPARABLE"
    floor = (
        policy.ABSOLUTE_BYTES_FLOOR if unit == "bytes" else policy.ABSOLUTE_MS_FLOOR
    )
    if abs(delta) < floor:
        return "NEUTRAL"
    if baseline and abs(delta / baseline) < policy.RELATIVE_SIGNIFICANCE:
        return "NEUTRAL"
    return "REGRESSION" if delta > 0 else "IMPROVEMENT"


def compare_payloads(baseline: dict, candidate: dict) -> dict:
    deltas: List[MetricDelta] = []
    for section, metric, unit in _COMPARED_METRICS:
        b = _metric_value(baseline, section, metric)
        c = _metric_value(candidate, section, metric)
        delta = (c - b) if (b is not None and c is not None) else None
        deltas.append(
            MetricDelta(section, metric, unit, b, c, _classify(delta, b, unit))
        )

    statuses = {d.status for d in deltas if d.status != "INCOMPARABLE"}
    headline_metric = next(
        (d for d in deltas if d.metric == "step_time_ms"), None
    )
    if "REGRESSION" in statuses and "IMPROVEMENT" in statuses:
        verdict = "MIXED"
    elif "REGRESSION" in statuses:
        verdict = "REGRESSION"
    elif "IMPROVEMENT" in statuses:
        verdict = "IMPROVEMENT"
    elif statuses:
        verdict = "NEUTRAL"
    else:
        verdict = "INCOMPARABLE"
    # headline step-time movement dominates the verdict when significant
    if headline_metric and headline_metric.status in ("REGRESSION", "IMPROVEMENT"):
        verdict = headline_metric.status

    return {
        "verdict": verdict,
        "baseline_diagnosis": baseline.get("primary_diagnosis", {}).get("kind"),
        "candidate_diagnosis": candidate.get("primary_diagnosis", {}).get("kind"),
        "metrics": [
            {
                "section": d.section,
                "metric": d.metric,
                "unit": d.unit,
                "baseline": d.baseline,
                "candidate": d.candidate,
                "delta": d.delta,
                "pct": d.pct,
                "status": d.status,
            }
            for d in deltas
        ],
    }


def render_compare(result: dict) -> str:
    lines = [
        f"TraceML-AMD Compare Verdict: {result['verdict']}",
        f"  diagnosis: {result['baseline_diagnosis']} -> "
        f"{result['candidate_diagnosis']}",
        "",
        f"  {'metric':<28} {'baseline':>12} {'candidate':>12} {'delta':>14}  status",
    ]
    for m in result["metrics"]:
        if m["baseline"] is None and m["candidate"] is None:
            continue

        def fmt(v):
            if v is None:
                return "—"
            if m["unit"] == "bytes":
                return f"{v / (1 << 30):.2f}G"
            return f"{v:.1f}"

        delta_s = "—"
        if m["delta"] is not None:
            sign = "+" if m["delta"] >= 0 else ""
            delta_s = f"{sign}{fmt(m['delta'])}"
            if m["pct"] is not None:
                delta_s += f" ({sign}{m['pct']:.1f}%)"
        lines.append(
            f"  {m['metric']:<28} {fmt(m['baseline']):>12} "
            f"{fmt(m['candidate']):>12} {delta_s:>14}  {m['status']}"
        )
    return "\n".join(lines)


def compare_files(path_a: str, path_b: str) -> int:
    try:
        baseline = load_summary(path_a)
        candidate = load_summary(path_b)
    except (OSError, ValueError) as exc:
        print(f"compare: {exc}", file=sys.stderr)
        return 1
    result = compare_payloads(baseline, candidate)
    print(render_compare(result))
    return 0
