"""Final report generator → ``final_summary.{json,txt}``
(reference: reporting/final.py:47-990).

Top-level shape (schema 1.7-compatible): schema_version / generated_at /
duration_s / meta / primary_diagnosis / system / process / step_time /
step_memory / text. ``text`` is the verdict-first compact report:
TraceML Verdict, Why, Next, then the section cards.
"""

from __future__ import annotations

import datetime
import os
import sqlite3
from typing import Optional

from traceml_amd.reporting.primary import build_primary_diagnosis
from traceml_amd.reporting.schema import SCHEMA_VERSION
from traceml_amd.reporting.sections import process as process_section
from traceml_amd.reporting.sections import step_memory as step_memory_section
from traceml_amd.reporting.sections import step_time as step_time_section
from traceml_amd.reporting.sections import system as system_section
from traceml_amd.utils.atomic_io import atomic_write_json, atomic_write_text

_SECTION_BUILDERS = [
    ("system", system_section.build),
    ("process", process_section.build),
    ("step_time", step_time_section.build),
    ("step_memory", step_memory_section.build),
]


def _run_duration_s(db_path: str) -> Optional[float]:
    try:
        conn = sqlite3.connect(f"file:{db_path}?mode=ro", uri=True)
        try:
            lo_hi = []
            for table in ("step_time_samples", "system_samples", "process_samples"):
                try:
                    row = conn.execute(
                        f"SELECT MIN(timestamp), MAX(timestamp) FROM {table}"
                    ).fetchone()
                    if row and row[0] is not None:
                        lo_hi.append(row)
                except sqlite3.Error:
                    continue
            if not lo_hi:
                return None
            lo = min(r[0] for r in lo_hi)
            hi = max(r[1] for r in lo_hi)
            return max(0.0, hi - lo)
        finally:
            conn.close()
    except sqlite3.Error:
        return None


class FinalReportGenerator:
    def __init__(self, db_path: str, run_name: Optional[str] = None) -> None:
        self.db_path = db_path
        self.run_name = run_name

    def generate(self) -> dict:
        sections = {}
        for name, builder in _SECTION_BUILDERS:
            try:
                sections[name] = builder(self.db_path)
            except Exception as exc:  # a broken section never kills the report
                sections[name] = {
                    "metadata": {"mode": "no_data", "error": repr(exc)},
                    "diagnosis": {
                        "kind": "SECTION_ERROR",
                        "status": "SECTION ERROR",
                        "severity": "info",
                        "summary": f"Section failed to build: {exc!r}",
                        "action": "",
                    },
                    "issues": [],
                    "global": {},
                    "groups": {"by": "global_rank", "rows": {}},
                    "units": {},
                    "card": "",
                }

        primary = build_primary_diagnosis(
            sections.get("step_time", {}), sections.get("system", {})
        )

        st_md = sections.get("step_time", {}).get("metadata", {})
        ranks_seen = st_md.get("global_ranks_seen") or []
        world_size = None
        rows = sections.get("step_time", {}).get("groups", {}).get("rows", {})
        for row in rows.values():
            ws = row.get("identity", {}).get("world_size")
            if ws:
                world_size = ws
                break
        payload = {
            "schema_version": SCHEMA_VERSION,
            "generated_at": datetime.datetime.now(datetime.timezone.utc).isoformat(),
            "duration_s": _run_duration_s(self.db_path),
            "meta": {
                "run_name": self.run_name,
                "mode": st_md.get("mode", "no_data"),
                "world_size": world_size or (len(ranks_seen) or None),
                "nodes_observed": st_md.get("nodes_observed"),
                "gpus_observed": st_md.get("gpus_observed"),
            },
            "primary_diagnosis": primary,
            **sections,
        }
        payload["text"] = build_verdict_text(payload)
        return payload


def _junk_tolerant_dict(value) -> dict:
    return value if isinstance(value, dict) else {}


def build_verdict_text(payload: dict) -> str:
    primary = _junk_tolerant_dict(payload.get("primary_diagnosis"))
    lines = [
        "TraceML-AMD Verdict: "
        + str(primary.get("status", "UNKNOWN"))
        + f"  [{primary.get('severity', 'info')}]",
        "Why:  " + str(primary.get("summary", "")),
    ]
    action = primary.get("action")
    if action:
        lines.append("Next: " + str(action))
    lines.append("")
    lines.append("Sections:")
    for name in ("step_time", "step_memory", "system", "process"):
        diag = _junk_tolerant_dict(
            _junk_tolerant_dict(payload.get(name)).get("diagnosis")
        )
        if diag:
            lines.append(
                f"  {name:<12} {diag.get('status', ''):<28} "
                f"[{diag.get('severity', 'info')}]"
            )
    lines.append("")
    rank_table = _rank_evidence_table(payload)
    if rank_table:
        lines.append(rank_table)
        lines.append("")
    for name in ("step_time", "step_memory", "system", "process"):
        card = _junk_tolerant_dict(payload.get(name)).get("card")
        if card:
            lines.append(card)
            lines.append("")
    return "\n".join(lines).rstrip() + "\n"


def _rank_evidence_table(payload: dict) -> str:
    """Compact per-rank step-time evidence (multi-rank runs only)."""
    rows = _junk_tolerant_dict(
        _junk_tolerant_dict(
            _junk_tolerant_dict(payload.get("step_time")).get("groups")
        ).get("rows")
    )
    if len(rows) < 2:
        return ""
    metrics = [
        ("step_time_ms", "step"),
        ("input_wait_ms", "input"),
        ("forward_ms", "fwd"),
        ("backward_ms", "bwd"),
        ("optimizer_ms", "opt"),
        ("ddp_comm_ms", "comm"),
        ("residual_ms", "resid"),
    ]
    keys = sorted(rows, key=lambda k: (len(k), k))
    used = [
        (m, label)
        for m, label in metrics
        if any(rows[k]["metrics"].get(m) is not None for k in keys)
    ]
    if not used:
        return ""
    header = "  rank   " + "".join(f"{label:>9}" for _, label in used) + "   (ms)"
    lines = ["Per-rank step time:", header]
    for key in keys:
        cells = rows[key]["metrics"]
        line = f"  r{key:<6}"
        for m, _label in used:
            value = cells.get(m)
            line += f"{value:>9.1f}" if value is not None else f"{'—':>9}"
        lines.append(line)
    return "\n".join(lines)


def write_summary_artifacts(
    payload: dict, session_dir: str, html: bool = False
) -> dict:
    from traceml_amd.sdk import protocol

    paths = {
        "json": protocol.summary_json_path(session_dir),
        "txt": protocol.summary_txt_path(session_dir),
    }
    atomic_write_json(paths["json"], payload)
    atomic_write_text(paths["txt"], payload.get("text", ""))
    if html:
        from traceml_amd.reporting.html.document import render_html

        paths["html"] = protocol.summary_html_path(session_dir)
        atomic_write_text(paths["html"], render_html(payload))
    return paths


def _enrich_actions_from_code_manifest(payload: dict, session_dir: str) -> None:
    """Cross-reference the AST code manifest: when the verdict is
    input-related and the user's DataLoader was constructed with
    num_workers=0 (or unset), say so in the action — the most common cause,
    named concretely."""
    primary = payload.get("primary_diagnosis", {})
    kind = primary.get("kind")
    if kind not in ("INPUT_BOUND", "INPUT_STRAGGLER", "H2D_BOUND"):
        return
    import json as _json

    try:
        with open(
            os.path.join(session_dir, "code_manifest.json"), encoding="utf-8"
        ) as f:
            manifest = _json.load(f)
    except (OSError, ValueError):
        return

    def _apply(hint: str) -> None:
        primary["action"] = primary.get("action", "") + hint
        st_diag = payload.get("step_time", {}).get("diagnosis")
        if st_diag:
            st_diag["action"] = st_diag.get("action", "") + hint

    for call in manifest.get("calls", []):
        if call.get("call") != "DataLoader":
            continue
        kwargs = call.get("kwargs", {})
        if kind in ("INPUT_BOUND", "INPUT_STRAGGLER"):
            workers = kwargs.get("num_workers", 0)
            if workers in (0, None):
                _apply(
                    f" Your script constructs DataLoader with num_workers="
                    f"{workers} (line {call.get('line')}): loading happens on "
                    "the training thread — start with num_workers=8 and "
                    "pin_memory=True."
                )
                break
        else:  # H2D_BOUND
            pin = kwargs.get("pin_memory", False)
            if pin in (False, None):
                _apply(
                    f" Your script constructs DataLoader without pin_memory "
                    f"(line {call.get('line')}): page-locked staging enables "
                    "async H2D — set pin_memory=True and pass "
                    "non_blocking=True to .to(device)."
                )
                break


def generate_summary(
    db_path: str,
    session_dir: str,
    run_name: Optional[str] = None,
    html: bool = False,
) -> dict:
    payload = FinalReportGenerator(db_path, run_name=run_name).generate()
    _enrich_actions_from_code_manifest(payload, session_dir)
    payload["text"] = build_verdict_text(payload)  # re-render with the hint
    write_summary_artifacts(payload, session_dir, html=html)
    return payload
