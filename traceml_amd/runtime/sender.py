"""Telemetry publisher: per-tick collection of incremental payloads from every
sampler's database, flushed to disk writers and batched to the exporter as
ONE TCP frame per tick (reference: runtime/sender.py:19-180)."""

from __future__ import annotations

from typing import List, Optional

from traceml_amd.database.sender import DBIncrementalSender
from traceml_amd.database.writer import DatabaseWriter
from traceml_amd.runtime.exporter import TelemetryExporter
from traceml_amd.runtime.identity import RuntimeIdentity


class TelemetryPublisher:
    def __init__(
        self,
        identity: RuntimeIdentity,
        exporter: TelemetryExporter,
        data_dir: Optional[str] = None,
    ) -> None:
        self._identity_meta = identity.to_meta()
        self._exporter = exporter
        self._senders: List[DBIncrementalSender] = []
        self._writers: List[DatabaseWriter] = []
        self._data_dir = data_dir

    def attach_sampler(self, sampler_name: str, database) -> None:
        self._senders.append(DBIncrementalSender(sampler_name, database))
        if self._data_dir:
            self._writers.append(
                DatabaseWriter(sampler_name, database, self._data_dir)
            )

    def publish(self) -> int:
        for writer in self._writers:
            writer.flush()
        payloads = []
        for sender in self._senders:
            payload = sender.collect_payload(self._identity_meta)
            if payload is not None:
                payloads.append(payload)
        if payloads:
            self._exporter.send_batch(payloads)
        return len(payloads)

    def send_control(self, payload: dict) -> None:
        self._exporter.send_batch([payload])

    def close(self) -> None:
        for writer in self._writers:
            writer.close()
