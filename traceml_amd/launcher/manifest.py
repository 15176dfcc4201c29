"""Run manifests (reference: launcher/manifest.py:58-228).

``manifest.json`` tracks launch metadata + status transitions
starting → running → completed/failed; ``code_manifest.json`` captures a
static scan of the user script (utils/ast_analysis).
"""

from __future__ import annotations

import os
import sys
import time
from typing import List, Optional

from traceml_amd.utils.atomic_io import atomic_write_json

STATUS_STARTING = "starting"
STATUS_RUNNING = "running"
STATUS_COMPLETED = "completed"
STATUS_FAILED = "failed"


def manifest_path(session_dir: str) -> str:
    return os.path.join(session_dir, "manifest.json")


def write_run_manifest(
    session_dir: str,
    status: str,
    script: str = "",
    script_args: Optional[List[str]] = None,
    world_size: int = 1,
    nnodes: int = 1,
    run_name: Optional[str] = None,
    extra: Optional[dict] = None,
) -> None:
    path = manifest_path(session_dir)
    payload = {
        "status": status,
        "updated_at": time.time(),
        "script": script,
        "script_args": list(script_args or []),
        "world_size": world_size,
        "nnodes": nnodes,
        "run_name": run_name,
        "python": sys.version.split()[0],
        "argv": sys.argv,
    }
    if extra:
        payload.update(extra)
    atomic_write_json(path, payload)


def update_status(session_dir: str, status: str, extra: Optional[dict] = None) -> None:
    import json

    path = manifest_path(session_dir)
    try:
        with open(path, "r", encoding="utf-8") as f:
            payload = json.load(f)
    except (OSError, ValueError):
        payload = {}
    payload["status"] = status
    payload["updated_at"] = time.time()
    if extra:
        payload.update(extra)
    atomic_write_json(path, payload)


def write_code_manifest(session_dir: str, script: str) -> None:
    try:
        from traceml_amd.utils.ast_analysis import scan_script

        payload = scan_script(script)
    except Exception as exc:
        payload = {"error": repr(exc)}
    atomic_write_json(os.path.join(session_dir, "code_manifest.json"), payload)


def collect_existing_artifacts(session_dir: str) -> List[str]:
    found = []
    for name in (
        "manifest.json",
        "code_manifest.json",
        "final_summary.json",
        "final_summary.txt",
        "final_summary.html",
        "finalization_warning.json",
        "finalization_error.json",
    ):
        if os.path.exists(os.path.join(session_dir, name)):
            found.append(name)
    return found
