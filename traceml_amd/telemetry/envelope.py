"""Canonical telemetry wire payload.

``{"meta": {rank identity…, "sampler": str, "timestamp": float},
   "body": {"tables": {table_name: [row, …]}}}``

No version field — the envelope evolves additively (reference:
telemetry/envelope.py:37-166). ``normalize_telemetry_envelope`` accepts the
canonical shape and returns None for anything unintelligible; unknown extra
meta keys are preserved.
"""

from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

META_REQUIRED = ("sampler",)


def build_telemetry_envelope(
    identity_meta: Dict[str, Any],
    sampler: str,
    tables: Dict[str, List[dict]],
    timestamp: Optional[float] = None,
) -> dict:
    meta = dict(identity_meta)
    meta["sampler"] = sampler
    meta["timestamp"] = time.time() if timestamp is None else timestamp
    return {"meta": meta, "body": {"tables": tables}}


def normalize_telemetry_envelope(payload: Any) -> Optional[dict]:
    if not isinstance(payload, dict):
        return None
    meta = payload.get("meta")
    body = payload.get("body")
    if not isinstance(meta, dict) or not isinstance(body, dict):
        return None
    if not isinstance(meta.get("sampler"), str):
        return None
    tables = body.get("tables")
    if not isinstance(tables, dict):
        return None
    clean_tables: Dict[str, List[dict]] = {}
    for name, rows in tables.items():
        if not isinstance(name, str) or not isinstance(rows, list):
            continue
        clean_tables[name] = [r for r in rows if isinstance(r, dict)]
    return {"meta": meta, "body": {"tables": clean_tables}}
