"""``traceml-amd compare A.json B.json`` (reference: reporting/compare/*,
~2k LoC: io.py:200 schema-versioned readers, sections/, policy.py:216,
verdict.py:506 rule chain).

Pipeline: strict schema-versioned load → normalize → per-section comparers
(step_time clock-aware, step_memory, system, process) → priority rule-chain
verdict → sectioned text render. ``--fail-on-regression`` turns the verdict
into a CI gate (exit 4).
"""

from __future__ import annotations

import sys
from typing import Dict, List, Optional

from traceml_amd.reporting.compare import io as compare_io
from traceml_amd.reporting.compare.model import CompareSection
from traceml_amd.reporting.compare.sections import (
    compare_process,
    compare_step_memory,
    compare_step_time,
    compare_system,
    per_rank_step_time,
)
from traceml_amd.reporting.compare.verdict import decide_verdict

#: kept for callers of the old reader API
load_summary = compare_io.load_summary


def compare_payloads(baseline: dict, candidate: dict) -> dict:
    """Full sectioned comparison of two loaded final summaries."""
    baseline, notes_a = compare_io.normalize_summary(dict(baseline))
    candidate, notes_b = compare_io.normalize_summary(dict(candidate))
    sections: Dict[str, CompareSection] = {
        "step_time": compare_step_time(baseline, candidate),
        "step_memory": compare_step_memory(baseline, candidate),
        "system": compare_system(baseline, candidate),
        "process": compare_process(baseline, candidate),
    }
    finding = decide_verdict(sections)

    # flat list kept for existing consumers (CI scripts, old tests)
    flat: List[dict] = []
    for name in ("step_time", "step_memory"):
        for key, metric in sections[name].metrics.items():
            p = metric.to_payload()
            flat.append(
                {
                    "section": name,
                    "metric": key,
                    "unit": p["unit"],
                    "baseline": p["baseline"],
                    "candidate": p["candidate"],
                    "delta": p["delta"],
                    "pct": p["pct"],
                    "status": p["status"],
                    "significance": p["significance"],
                }
            )

    def _primary_kind(payload: dict):
        diag = payload.get("primary_diagnosis")
        return diag.get("kind") if isinstance(diag, dict) else None

    st_diag = sections["step_time"].diagnosis
    return {
        "verdict": finding.verdict,
        "finding": finding.to_payload(),
        "baseline_diagnosis": _primary_kind(baseline),
        "candidate_diagnosis": _primary_kind(candidate),
        "diagnosis_transition": (
            st_diag.to_payload() if st_diag else None
        ),
        "sections": {k: s.to_payload() for k, s in sections.items()},
        "per_rank_step_time": per_rank_step_time(baseline, candidate),
        "metrics": flat,
        "notes": notes_a + notes_b,
    }


def _fmt_value(value: Optional[float], unit: str) -> str:
    if value is None:
        return "—"
    if unit == "bytes":
        return f"{value / (1 << 30):.2f}G"
    if unit == "percent":
        return f"{value:.1f}"
    return f"{value:.1f}"


def _fmt_delta(m: dict) -> str:
    if m["delta"] is None:
        return "—"
    sign = "+" if m["delta"] >= 0 else ""
    out = f"{sign}{_fmt_value(m['delta'], m['unit'])}"
    if m.get("pct") is not None:
        out += f" ({sign}{m['pct']:.1f}%)"
    return out


_SECTION_TITLES = {
    "step_time": "Step time",
    "step_memory": "Step memory",
    "system": "System (context)",
    "process": "Process (context)",
}


def render_compare(result: dict, label_a: str = "A", label_b: str = "B") -> str:
    finding = result.get("finding", {})
    transition = result.get("diagnosis_transition") or {}
    lines = [
        f"TraceML-AMD Compare Verdict: {result['verdict']}"
        + (
            f"  [{finding['significance']}]"
            if finding.get("significance")
            else ""
        ),
        f"  {finding.get('title', '')}: {finding.get('detail', '')}",
        f"  runs: {label_a} (baseline) vs {label_b} (candidate)",
        f"  diagnosis: {result['baseline_diagnosis']} -> "
        f"{result['candidate_diagnosis']}"
        + (
            f"  ({transition.get('direction')})"
            if transition.get("direction")
            else ""
        ),
    ]
    for note in result.get("notes", []):
        lines.append(f"  note: {note}")

    for name, title in _SECTION_TITLES.items():
        section = result.get("sections", {}).get(name)
        if not section:
            continue
        metrics = [
            dict(m, metric=key)
            for key, m in section.get("metrics", {}).items()
            if m["baseline"] is not None or m["candidate"] is not None
        ]
        if not metrics and not section.get("notes"):
            continue
        lines.append("")
        diag = section.get("diagnosis_transition") or {}
        diag_txt = ""
        if diag.get("from", {}).get("kind") or diag.get("to", {}).get("kind"):
            diag_txt = (
                f"   [{diag['from'].get('kind')} -> {diag['to'].get('kind')}"
                f", {diag.get('direction')}]"
            )
        lines.append(f"  {title}{diag_txt}")
        for note in section.get("notes", []):
            lines.append(f"    note: {note}")
        if metrics:
            lines.append(
                f"    {'metric':<30} {'baseline':>12} {'candidate':>12} "
                f"{'delta':>18}  status"
            )
        for m in metrics:
            status = m["status"]
            if m.get("significance") in ("moderate", "material") and status in (
                "REGRESSION", "IMPROVEMENT",
            ):
                status += f" ({m['significance']})"
            lines.append(
                f"    {m.get('label') or m['metric']:<30} "
                f"{_fmt_value(m['baseline'], m['unit']):>12} "
                f"{_fmt_value(m['candidate'], m['unit']):>12} "
                f"{_fmt_delta(m):>18}  {status}"
            )

    per_rank = [
        r
        for r in result.get("per_rank_step_time", [])
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
        pair = compare_io.load_pair(path_a, path_b)
    except (OSError, ValueError) as exc:
        print(f"compare: {exc}", file=sys.stderr)
        return 1
    result = compare_payloads(pair["baseline"], pair["candidate"])
    result["notes"] = pair["notes"] + result.get("notes", [])
    print(
        render_compare(
            result, pair["baseline_label"], pair["candidate_label"]
        )
    )
    if fail_on_regression and result["verdict"] == "REGRESSION":
        return 4  # CI perf gate
    return 0
