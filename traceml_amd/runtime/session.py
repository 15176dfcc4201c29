"""Session identity + per-rank directory naming (reference: runtime/session.py)."""

from __future__ import annotations

import os
import time
from typing import Optional

_session_id: Optional[str] = None


def generate_session_id() -> str:
    return time.strftime("%Y%m%d-%H%M%S") + f"-{os.getpid() % 10000:04d}"


def get_session_id(explicit: Optional[str] = None) -> str:
    global _session_id
    if explicit:
        _session_id = explicit
        return _session_id
    if _session_id is None:
        _session_id = os.environ.get("TRACEML_SESSION_ID") or generate_session_id()
    return _session_id


def rank_dir_name(local_rank: int) -> str:
    return f"r{local_rank}"


def session_dir(logs_dir: str, session_id: str) -> str:
    return os.path.join(logs_dir, session_id)


def reset_for_tests() -> None:
    global _session_id
    _session_id = None
