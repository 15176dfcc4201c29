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
        "diagnosis_transition": _diagnosis_transition(baseline, candidate),
        "per_rank_step_time": _per_rank_compare(
            baseline, candidate, "step_time", "step_time_ms"
        ),
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


_SEVERITY_RANK = {"info": 0, "warn": 1, "crit": 2}


def _diagnosis_transition(baseline: dict, candidate: dict) -> dict:
    b = baseline.get("primary_diagnosis", {})
    c = candidate.get("primary_diagnosis", {})
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
    return {
        "from": {"kind": b.get("kind"), "severity": b.get("severity")},
        "to": {"kind": c.get("kind"), "severity": c.get("severity")},
        "direction": direction,
    }


def _per_rank_compare(
    baseline: dict, candidate: dict, section: str, metric: str
) -> List[dict]:
    def rank_values(payload):
        rows = payload.get(section, {}).get("groups", {}).get("rows", {})
        return {
            key: row.get("metrics", {}).get(metric)
            for key, row in rows.items()
        }

    b_rows = rank_values(baseline)
    c_rows = rank_values(candidate)
    out = []
    for key in sorted(set(b_rows) | set(c_rows), key=lambda k: (len(k), k)):
        b = b_rows.get(key)
        c = c_rows.get(key)
        delta = (c - b) if (b is not None and c is not None) else None
        out.append(
            {
                "rank": key,
                "baseline": b,
                "candidate": c,
                "delta": delta,
                "pct": (delta / b * 100.0) if (delta is not None and b) else None,
            }
        )
    return out


def render_compare(result: dict) -> str:
    transition = result.get("diagnosis_transition", {})
    lines = [
        f"TraceML-AMD Compare Verdict: {result['verdict']}",
        f"  diagnosis: {result['baseline_diagnosis']} -> "
        f"{result['candidate_diagnosis']}"
        + (f"  ({transition.get('direction')})" if transition else ""),
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
    per_rank = [
        r for r in result.get("per_rank_step_time", [])
        if r["baseline"] is not None or r["candidate"] is not None
    ]
    if len(per_rank) > 1:
        lines.append("")
        lines.append("  step_time_ms by rank:")
        for r in per_rank:
            delta_s = "—"
            if r["delta"] is not None:
                sign = "+" if r["delta"] >= 0 else ""
                delta_s = f"{sign}{r['delta']:.1f}"
                if r["pct"] is not None:
                    delta_s += f" ({sign}{r['pct']:.1f}%)"
            b = "—" if r["baseline"] is None else f"{r['baseline']:.1f}"
            c = "—" if r["candidate"] is None else f"{r['candidate']:.1f}"
            lines.append(f"    r{r['rank']:<4} {b:>10} -> {c:>10}  {delta_s}")
    return "\n".join(lines)


def compare_files(
    path_a: str, path_b: str, fail_on_regression: bool = False
) -> int:
    try:
        baseline = load_summary(path_a)
        candidate = load_summary(path_b)
    except (OSError, ValueError) as exc:
        print(f"compare: {exc}", file=sys.stderr)
        return 1
    result = compare_payloads(baseline, candidate)
    print(render_compare(result))
    if fail_on_regression and result["verdict"] == "REGRESSION":
        return 4  # CI perf gate
    return 0
