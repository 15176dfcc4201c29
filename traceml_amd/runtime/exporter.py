"""Telemetry exporter: dedicated send thread with a bounded drop-oldest queue,
isolating the sampler loop from a slow or unreachable aggregator
(reference: runtime/exporter.py:25-191)."""

from __future__ import annotations

import logging
import threading
import time
from collections import deque
from typing import List, Optional

from traceml_amd.transport.tcp import TCPClient

logger = logging.getLogger(__name__)

QUEUE_MAX = 2048
FINAL_DRAIN_BUDGET_SEC = 2.0


class TelemetryExporter:
    def __init__(self, client: TCPClient) -> None:
        self._client = client
        self._queue: deque = deque()
        self._lock = threading.Lock()
        self._wake = threading.Event()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.dropped = 0

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self._loop, name="traceml-exporter", daemon=True
        )
        self._thread.start()

    def send_batch(self, payloads: List[dict]) -> None:
        if not payloads:
            return
        with self._lock:
            while len(self._queue) >= QUEUE_MAX:
                self._queue.popleft()
                self.dropped += 1
            self._queue.append(payloads)
        self._wake.set()

    def _take(self) -> Optional[List[dict]]:
        with self._lock:
            if self._queue:
                return self._queue.popleft()
        return None

    def _loop(self) -> None:
        while not self._stop.is_set():
            batch = self._take()
            if batch is None:
                self._wake.wait(timeout=0.5)
                self._wake.clear()
                continue
            self._client.send_batch(batch)

    def stop(self) -> None:
        deadline = time.time() + FINAL_DRAIN_BUDGET_SEC
        while time.time() < deadline:
            batch = self._take()
            if batch is None:
                break
            self._client.send_batch(batch)
        self._stop.set()
        self._wake.set()
        if self._thread is not None:
            self._thread.join(timeout=2.0)
        self._client.close()
