"""On-demand summary file-RPC service (reference: aggregator/summary_service.py:27-143).

Polls the session's ``control/final_summary_request.json``; on a new request
(re)generates the summary from the live SQLite (after a flush barrier) and
writes the artifacts, so in-run ``summary()``/``final_summary()`` calls work
before the run ends.
"""

from __future__ import annotations

import json
import logging
import os
from typing import Optional

from traceml_amd.reporting.final import generate_summary
from traceml_amd.sdk import protocol
from traceml_amd.utils.atomic_io import atomic_write_json

logger = logging.getLogger(__name__)


class FinalSummaryService:
    def __init__(
        self,
        session_dir: str,
        db_path: str,
        sqlite_writer=None,
        run_name: Optional[str] = None,
        html: bool = False,
    ) -> None:
        self.session_dir = session_dir
        self.db_path = db_path
        self.sqlite_writer = sqlite_writer
        self.run_name = run_name
        self.html = html
        self._last_request_id: Optional[str] = None

    def poll(self) -> bool:
        """Answer a pending request; returns True if a summary was produced."""
        request_path = protocol.request_path(self.session_dir)
        try:
            with open(request_path, "r", encoding="utf-8") as f:
                request = json.load(f)
        except (OSError, ValueError):
            return False
        if not isinstance(request, dict):
            return False  # junk request file: ignore until overwritten
        request_id = request.get("request_id")
        if not request_id or request_id == self._last_request_id:
            return False
        self._last_request_id = request_id
        try:
            if self.sqlite_writer is not None:
                self.sqlite_writer.force_flush(timeout=5.0)
            generate_summary(
                self.db_path,
                self.session_dir,
                run_name=self.run_name,
                html=self.html,
            )
            atomic_write_json(
                protocol.response_path(self.session_dir),
                {"request_id": request_id, "status": "ok"},
            )
            return True
        except Exception as exc:
            logger.warning("traceml_amd: summary request failed", exc_info=True)
            try:
                atomic_write_json(
                    protocol.response_path(self.session_dir),
                    {"request_id": request_id, "status": "error", "error": repr(exc)},
                )
            except OSError:
                pass
            return False
