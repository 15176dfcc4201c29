"""In-run summary access: ``summary()`` and ``final_summary()``
(reference: sdk/summary_client.py:56-153, summary_projection.py:67).

``final_summary(wait=True)`` writes a request file (rank 0) and polls for
the aggregator to produce ``final_summary.json``. ``summary()`` returns a
flat, tracker-friendly dict (W&B/MLflow-ready) projected from the same
payload.
"""

from __future__ import annotations

import json
import os
import time
import uuid
from typing import Any, Dict, Optional

from traceml_amd.runtime.identity import resolve_runtime_identity
from traceml_amd.runtime.session import get_session_id, session_dir
from traceml_amd.runtime.settings import TraceMLSettings
from traceml_amd.sdk import protocol
from traceml_amd.utils.atomic_io import atomic_write_json


def _session_dir() -> str:
    # the settings the process was init()ed with win over raw env (a direct
    # init(logs_dir=..., session_id=...) launch has no TRACEML_* env set)
    from traceml_amd.sdk import initial

    config = initial.get_active_config()
    settings = (
        config.settings if config is not None else TraceMLSettings.from_env()
    )
    return session_dir(settings.logs_dir, get_session_id(settings.session_id))


def final_summary(
    wait: bool = True,
    timeout_sec: float = 60.0,
    poll_interval_sec: float = 0.5,
) -> Optional[Dict[str, Any]]:
    sdir = _session_dir()
    json_path = protocol.summary_json_path(sdir)
    identity = resolve_runtime_identity()
    if identity.global_rank == 0:
        os.makedirs(protocol.control_dir(sdir), exist_ok=True)
        atomic_write_json(
            protocol.request_path(sdir),
            {"request_id": uuid.uuid4().hex, "requested_at": time.time()},
        )
    if not wait:
        return _read_json(json_path)
    deadline = time.time() + timeout_sec
    last_mtime = None
    while time.time() < deadline:
        if os.path.exists(json_path):
            mtime = os.path.getmtime(json_path)
            if last_mtime is None:
                last_mtime = mtime
            payload = _read_json(json_path)
            if payload is not None:
                return payload
        time.sleep(poll_interval_sec)
    return _read_json(json_path)


def _read_json(path: str) -> Optional[dict]:
    try:
        with open(path, "r", encoding="utf-8") as f:
            return json.load(f)
    except (OSError, ValueError):
        return None


def compact_summary(payload: Optional[dict]) -> Dict[str, Any]:
    """Flatten a final-summary payload into tracker-friendly scalars."""
    out: Dict[str, Any] = {}
    if not payload:
        return out
    primary = payload.get("primary_diagnosis") or {}
    out["traceml/verdict"] = primary.get("status")
    out["traceml/verdict_kind"] = primary.get("kind")
    out["traceml/verdict_severity"] = primary.get("severity")
    step_time = (payload.get("step_time") or {}).get("global") or {}
    for metric, value in (step_time.get("average") or {}).items():
        if isinstance(value, (int, float)):
            out[f"traceml/step_time/{metric}"] = value
    memory = (payload.get("step_memory") or {}).get("global") or {}
    for metric, value in (memory.get("average") or {}).items():
        if isinstance(value, (int, float)):
            out[f"traceml/step_memory/{metric}"] = value
    system = (payload.get("system") or {}).get("global") or {}
    for metric, value in (system.get("average") or {}).items():
        if isinstance(value, (int, float)):
            out[f"traceml/system/{metric}"] = value
    return out


def summary(timeout_sec: float = 30.0) -> Dict[str, Any]:
    return compact_summary(final_summary(wait=True, timeout_sec=timeout_sec))
