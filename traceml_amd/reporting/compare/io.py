"""Schema-versioned summary readers for compare (reference:
reporting/compare/io.py:200).

Strict loading: a file must be a traceml final summary (schema_version
present, known major version, the four top-level sections present — absent
sections are normalized to empty payloads with a note so older/partial
summaries stay comparable). Run labels come from the summary's run name or
the file stem, de-generified (final_summary.json → parent directory name).
"""

from __future__ import annotations

import json
import os
from typing import Dict, List, Tuple

REQUIRED_SECTIONS = ("system", "process", "step_time", "step_memory")
SUPPORTED_SCHEMA_MAJOR = 1

#: stems too generic to identify a run; fall back to the directory name
_GENERIC_STEMS = {"final_summary", "summary", "run", "output"}


class SummaryLoadError(ValueError):
    pass


def load_summary(path: str) -> dict:
    """Load + validate one final summary; raises SummaryLoadError."""
    if not os.path.isfile(path):
        raise SummaryLoadError(f"{path}: no such file")
    try:
        with open(path, "r", encoding="utf-8") as f:
            payload = json.load(f)
    except (OSError, ValueError) as exc:
        raise SummaryLoadError(f"{path}: not readable JSON ({exc})") from exc
    if not isinstance(payload, dict):
        raise SummaryLoadError(f"{path}: not a JSON object")
    version = payload.get("schema_version")
    if version is None:
        raise SummaryLoadError(
            f"{path}: not a traceml final summary (no schema_version)"
        )
    try:
        major = int(float(version))
    except (TypeError, ValueError):
        raise SummaryLoadError(f"{path}: bad schema_version {version!r}")
    if major != SUPPORTED_SCHEMA_MAJOR:
        raise SummaryLoadError(
            f"{path}: schema_version {version} unsupported "
            f"(this build compares major version {SUPPORTED_SCHEMA_MAJOR}.x)"
        )
    return payload


def normalize_summary(payload: dict) -> Tuple[dict, List[str]]:
    """Fill absent sections with empty payloads; returns (payload, notes).

    Additive schema evolution means a newer reader must accept older
    summaries: a missing section compares as unavailable, never as an
    error (reference: additive envelope evolution, architecture.md:70)."""
    notes: List[str] = []
    for section in REQUIRED_SECTIONS:
        block = payload.get(section)
        if not isinstance(block, dict):
            payload[section] = {
                "metadata": {},
                "diagnosis": None,
                "issues": [],
                "global": {},
                "groups": {"by": "global_rank", "rows": {}},
            }
            notes.append(
                f"section '{section}' absent in this summary "
                f"(schema {payload.get('schema_version')}); "
                "compared as unavailable"
            )
    return payload, notes


def run_label(path: str, payload: dict) -> str:
    """Human label for a run: meta run name > de-generified file stem."""
    meta = payload.get("meta")
    if isinstance(meta, dict):
        for key in ("run_name", "session_id"):
            value = meta.get(key)
            if isinstance(value, str) and value.strip():
                return value.strip()
    stem = os.path.splitext(os.path.basename(path))[0]
    if stem.lower() in _GENERIC_STEMS:
        parent = os.path.basename(os.path.dirname(os.path.abspath(path)))
        if parent:
            return parent
    return stem


def load_pair(path_a: str, path_b: str) -> Dict[str, object]:
    """Load baseline+candidate with labels and normalization notes."""
    baseline = load_summary(path_a)
    candidate = load_summary(path_b)
    baseline, notes_a = normalize_summary(baseline)
    candidate, notes_b = normalize_summary(candidate)
    return {
        "baseline": baseline,
        "candidate": candidate,
        "baseline_label": run_label(path_a, baseline),
        "candidate_label": run_label(path_b, candidate),
        "notes": [f"A: {n}" for n in notes_a] + [f"B: {n}" for n in notes_b],
    }
