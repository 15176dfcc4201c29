"""The aggregator: TCP ingest → SQLite persist → live display → deterministic
finalization (reference: aggregator/trace_aggregator.py:89-587).

Event-driven loop: wake on TCP data (or timeout), drain envelopes into the
async SQLite writer, consume ``rank_finished`` controls, rate-limit a UI
tick, answer in-run summary file-RPC requests.

Stop sequence (``stop()``):
1. settle — keep the TCP server open until every expected rank sent
   rank_finished OR the settle budget elapses (budget = finalize_timeout
   minus the SQLite+summary slice);
2. close TCP, drain + finalize SQLite (prune, wal_checkpoint TRUNCATE);
3. generate final_summary.{json,txt[,html]};
4. summary mode without a final_summary.json is a HARD failure
   (TraceMLFinalizationError) and writes finalization_error.json.
"""

from __future__ import annotations

import logging
import os
import threading
import time
from typing import Optional, Set

from traceml_amd.aggregator.display.base import DisplayDriver, SummaryDisplayDriver
from traceml_amd.aggregator.sqlite_writer import SQLiteWriterSimple
from traceml_amd.aggregator.summary_service import FinalSummaryService
from traceml_amd.reporting.final import generate_summary
from traceml_amd.runtime.session import get_session_id, session_dir
from traceml_amd.runtime.settings import TraceMLSettings
from traceml_amd.sdk import protocol
from traceml_amd.telemetry.control import CONTROL_KEY, RANK_FINISHED, parse_control
from traceml_amd.telemetry.envelope import normalize_telemetry_envelope
from traceml_amd.transport.tcp import TCPServer
from traceml_amd.utils.atomic_io import atomic_write_json

logger = logging.getLogger(__name__)

UI_TICK_MIN_INTERVAL = 1.0
#: fraction of the finalize budget spent settling (waiting for late ranks);
#: the rest goes to SQLite finalize + summary generation
SETTLE_BUDGET_FRACTION = 0.75
SQLITE_FINALIZE_MIN_SEC = 5.0
SQLITE_FINALIZE_MAX_SEC = 60.0


class TraceMLFinalizationError(RuntimeError):
    pass


def _build_display(mode: str, dashboard_port: int = 8765) -> DisplayDriver:
    if mode == "cli":
        from traceml_amd.aggregator.display.cli import CLIDisplayDriver

        return CLIDisplayDriver()
    if mode == "dashboard":
        from traceml_amd.aggregator.display.dashboard import DashboardDisplayDriver

        return DashboardDisplayDriver(port=dashboard_port)
    return SummaryDisplayDriver()


class TraceMLAggregator:
    def __init__(self, settings: Optional[TraceMLSettings] = None) -> None:
        self.settings = settings or TraceMLSettings.from_env()
        sid = get_session_id(self.settings.session_id)
        self.session_dir = session_dir(self.settings.logs_dir, sid)
        self.db_path = protocol.sqlite_path(self.session_dir)
        self.server = TCPServer(
            bind_host=self.settings.aggregator_bind,
            port=self.settings.aggregator_port,
        )
        self.sqlite = SQLiteWriterSimple(self.db_path)
        self.display = _build_display(
            self.settings.mode, self.settings.dashboard_port
        )
        self.summary_service = FinalSummaryService(
            self.session_dir,
            self.db_path,
            sqlite_writer=self.sqlite,
            run_name=self.settings.run_name,
            html=self.settings.html_report,
        )
        self._loop_thread: Optional[threading.Thread] = None
        self._stop_loop = threading.Event()
        self._finished_ranks: Set[int] = set()
        self._seen_ranks: Set[int] = set()
        self._last_ui_tick = 0.0
        self._started = False
        self._stopped = False

    @property
    def port(self) -> Optional[int]:
        return self.server.port

    # -- lifecycle ----------------------------------------------------------

    def start(self) -> None:
        if self._started:
            return
        self._started = True
        os.makedirs(self.session_dir, exist_ok=True)
        self.server.start()
        self.sqlite.start()
        self.display.start()
        self._loop_thread = threading.Thread(
            target=self._loop, name="traceml-aggregator", daemon=True
        )
        self._loop_thread.start()

    # -- main loop ----------------------------------------------------------

    def _loop(self) -> None:
        while not self._stop_loop.is_set():
            self.server.wait_for_data(timeout=0.5)
            self._drain_tcp()
            try:
                self.summary_service.poll()
            except Exception:
                logger.debug("traceml_amd: summary poll failed", exc_info=True)
            now = time.time()
            if now - self._last_ui_tick >= UI_TICK_MIN_INTERVAL:
                self._last_ui_tick = now
                self.sqlite.force_flush(timeout=0.5)
                self.display.render_tick(self.db_path)

    def _drain_tcp(self) -> int:
        count = 0
        for payload in self.server.drain():
            # One hostile/corrupt payload must never kill the ingest loop
            # thread (it would silently stop ALL telemetry): drop it.
            try:
                control = parse_control(payload)
                if control is not None:
                    self._handle_control(control)
                    continue
                envelope = normalize_telemetry_envelope(payload)
                if envelope is None:
                    continue
                meta = envelope["meta"]
                rank = meta.get("global_rank")
                if isinstance(rank, int):
                    self._seen_ranks.add(rank)
                self.sqlite.ingest(envelope)
                count += 1
            except Exception:
                logger.debug(
                    "traceml_amd: malformed payload dropped", exc_info=True
                )
        return count

    def _handle_control(self, control: dict) -> None:
        if control.get(CONTROL_KEY) == RANK_FINISHED:
            meta = control.get("meta")
            rank = meta.get("global_rank") if isinstance(meta, dict) else None
            if isinstance(rank, int):
                self._finished_ranks.add(rank)

    # -- finalization -------------------------------------------------------

    def _expected_ranks(self) -> Optional[int]:
        if self.settings.expected_ranks:
            return int(self.settings.expected_ranks)
        return None

    def _settle_end_of_run_telemetry(self, budget_sec: float) -> None:
        """Keep ingesting until all expected ranks said rank_finished or the
        budget ends (reference: trace_aggregator.py:440-499)."""
        start = time.time()
        deadline = start + budget_sec
        expected = self._expected_ranks()
        #: with no expected-rank contract and nothing ever received, a short
        #: grace beats burning the whole settle budget on an empty run
        empty_grace_deadline = start + min(3.0, budget_sec)
        while time.time() < deadline:
            self.server.wait_for_data(timeout=0.25)
            self._drain_tcp()
            if expected is not None:
                if len(self._finished_ranks) >= expected:
                    break
            elif self._seen_ranks:
                if self._finished_ranks >= self._seen_ranks:
                    break
            elif time.time() >= empty_grace_deadline:
                break
        self._drain_tcp()

    def stop(self) -> None:
        if self._stopped or not self._started:
            return
        self._stopped = True
        total_budget = max(5.0, float(self.settings.finalize_timeout))
        sqlite_slice = min(
            SQLITE_FINALIZE_MAX_SEC,
            max(SQLITE_FINALIZE_MIN_SEC, total_budget * (1 - SETTLE_BUDGET_FRACTION)),
        )
        settle_budget = max(1.0, total_budget - sqlite_slice)

        self._settle_end_of_run_telemetry(settle_budget)
        self._stop_loop.set()
        if self._loop_thread is not None:
            self._loop_thread.join(timeout=3.0)
        self.server.stop()
        self.display.stop()
        self.sqlite.finalize(budget_sec=sqlite_slice)

        warning: Optional[dict] = None
        expected = self._expected_ranks()
        if expected is not None and len(self._finished_ranks) < expected:
            warning = {
                "kind": "missing_rank_finished",
                "expected_ranks": expected,
                "finished_ranks": sorted(self._finished_ranks),
                "seen_ranks": sorted(self._seen_ranks),
            }

        try:
            generate_summary(
                self.db_path,
                self.session_dir,
                run_name=self.settings.run_name,
                html=self.settings.html_report,
            )
        except Exception as exc:
            atomic_write_json(
                os.path.join(self.session_dir, "finalization_error.json"),
                {"error": repr(exc), "warning": warning},
            )
            if self.settings.mode == "summary":
                raise TraceMLFinalizationError(
                    f"final summary generation failed: {exc!r}"
                ) from exc
            return

        if warning is not None:
            atomic_write_json(
                os.path.join(self.session_dir, "finalization_warning.json"), warning
            )

        if self.settings.mode == "summary" and not os.path.exists(
            protocol.summary_json_path(self.session_dir)
        ):
            raise TraceMLFinalizationError(
                "summary mode requires final_summary.json but it was not written"
            )
