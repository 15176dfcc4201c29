"""Per-rank runtime agent: one daemon sampler thread.

Tick = every sampler's ``sample()`` + one publisher flush (a single TCP
frame). Stop = final drain pass, last publish, then a ``rank_finished``
control message so the aggregator can settle deterministically
(reference: runtime/runtime.py:41-246).
"""

from __future__ import annotations

import logging
import os
import threading
import time
from typing import List, Optional

from traceml_amd.runtime.exporter import TelemetryExporter
from traceml_amd.runtime.identity import RuntimeIdentity, resolve_runtime_identity
from traceml_amd.runtime.registry import build_samplers
from traceml_amd.runtime.sender import TelemetryPublisher
from traceml_amd.runtime.session import get_session_id, rank_dir_name, session_dir
from traceml_amd.runtime.settings import TraceMLSettings
from traceml_amd.telemetry.control import build_rank_finished
from traceml_amd.transport.tcp import TCPClient

logger = logging.getLogger(__name__)


class TraceMLRuntime:
    def __init__(
        self,
        settings: TraceMLSettings,
        identity: Optional[RuntimeIdentity] = None,
    ) -> None:
        self.settings = settings
        self.identity = identity or resolve_runtime_identity()
        self._thread: Optional[threading.Thread] = None
        self._stop = threading.Event()
        self._started = False
        self._samplers: List = []

        session_id = get_session_id(settings.session_id)
        rank_data_dir = None
        if settings.logs_dir:
            rank_data_dir = os.path.join(
                session_dir(settings.logs_dir, session_id),
                rank_dir_name(self.identity.local_rank),
                "data",
            )

        client = TCPClient(settings.aggregator_host, settings.aggregator_port)
        self._exporter = TelemetryExporter(client)
        self._publisher = TelemetryPublisher(
            self.identity, self._exporter, data_dir=rank_data_dir
        )
        for spec, sampler, db in build_samplers(self.identity, settings.mode):
            self._samplers.append(sampler)
            self._publisher.attach_sampler(spec.name, db)

    def start(self) -> None:
        if self._started:
            return
        self._started = True
        self._exporter.start()
        self._thread = threading.Thread(
            target=self._sampler_loop, name="traceml-sampler", daemon=True
        )
        self._thread.start()

    def _tick(self) -> None:
        for sampler in self._samplers:
            sampler.sample()
        self._publisher.publish()

    def _sampler_loop(self) -> None:
        interval = max(0.1, float(self.settings.interval))
        while not self._stop.wait(timeout=interval):
            try:
                self._tick()
            except Exception:
                logger.debug("traceml_amd: tick failed", exc_info=True)

    def stop(self) -> None:
        if not self._started:
            return
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5.0)
        # Settle the RCCL rank-stats exchange BEFORE the final sampler drain:
        # waits bounded for in-flight gathers so none are pending when the
        # process group is later destroyed (teardown-hang guard).
        try:
            from traceml_amd.parallel.rank_stats import (
                disable_rank_stats_exchange,
            )

            disable_rank_stats_exchange(shutdown_timeout_sec=3.0)
        except Exception:
            logger.debug("traceml_amd: rank-stats shutdown failed", exc_info=True)
        # Final drain: give every sampler a last chance + publish + control.
        try:
            for sampler in self._samplers:
                sampler.on_stop()
            self._publisher.publish()
            self._publisher.send_control(
                build_rank_finished(self.identity.to_meta())
            )
        except Exception:
            logger.debug("traceml_amd: final drain failed", exc_info=True)
        self._publisher.close()
        self._exporter.stop()
        self._started = False

    def tick_once_for_tests(self) -> None:
        self._tick()
