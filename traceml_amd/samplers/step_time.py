"""Step-time sampler: ordered, non-blocking GPU resolution + per-step aggregation.

Each tick drains the step-time handoff queue and resolves batches strictly
oldest-first. A batch whose GPU stamps have not landed yet blocks everything
behind it (ordering guarantee for the analyzer's step alignment); it is
requeued and retried next tick. Resolution is a plain pinned-host-memory
read per ring stamp — no hipEventQuery, no synchronize
(reference semantics: samplers/step_time_sampler.py:92-160).

Aggregated row per step: ``{timestamp, step, events: {name: {duration_ms,
cpu_ms, gpu_ms, n_calls, device, is_gpu}}}``. ``duration_ms`` uses the GPU
clock except for ``dataloader_next`` and ``step_time`` which stay CPU wall
(reference: step_time_sampler.py:31-37,67-77).
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

from traceml_amd.core import event_names, timing
from traceml_amd.database.database import Database
from traceml_amd.samplers.base import BaseSampler

TABLE = "step_time_samples"

#: Batches older than this with unresolved GPU stamps are shipped CPU-only
#: (ring wrap / lost stamps must not wedge the pipeline forever).
STALE_BATCH_SEC = 30.0


def aggregate_batch(batch: timing.StepTimeBatch) -> dict:
    events: Dict[str, dict] = {}
    for event in batch.events:
        agg = events.get(event.name)
        if agg is None:
            agg = {
                "duration_ms": 0.0,
                "cpu_ms": 0.0,
                "gpu_ms": None,
                "n_calls": 0,
                "device": event.device,
                "is_gpu": False,
            }
            events[event.name] = agg
        cpu_ms = event.cpu_ms or 0.0
        agg["cpu_ms"] += cpu_ms
        agg["n_calls"] += 1
        if event.gpu_ms is not None:
            agg["gpu_ms"] = (agg["gpu_ms"] or 0.0) + event.gpu_ms
            agg["is_gpu"] = True
            agg["device"] = event.device
        if event.name in event_names.CPU_CLOCK_PREFERRED:
            agg["duration_ms"] += cpu_ms
        elif event.gpu_ms is not None:
            agg["duration_ms"] += event.gpu_ms
        else:
            agg["duration_ms"] += cpu_ms
    return {
        "timestamp": batch.flushed_at,
        "step": batch.step,
        "events": events,
    }


class StepTimeSampler(BaseSampler):
    name = "step_time"

    def __init__(self, database: Database) -> None:
        super().__init__(database)
        self._pending: List[timing.StepTimeBatch] = []

    def _sample(self) -> None:
        self._pending.extend(timing.drain_step_time_queue())
        now = time.time()
        emitted = 0
        while self._pending:
            batch = self._pending[0]
            unresolved = [e for e in batch.events if not e.try_resolve()]
            if unresolved:
                if now - batch.flushed_at < STALE_BATCH_SEC:
                    break  # ordered: retry this and everything behind next tick
                # Stale: abandon the GPU side for the stuck events.
                import logging

                logging.getLogger(__name__).warning(
                    "traceml_amd: step %d GPU stamps unresolved after %.0fs "
                    "(%s) — shipping CPU-only",
                    batch.step,
                    now - batch.flushed_at,
                    ",".join(e.name.split(":")[-1] for e in unresolved),
                )
                for event in unresolved:
                    event.gpu_start = event.gpu_end = None
                    event._gpu_done = True
            self._pending.pop(0)
            self.database.add_record(TABLE, aggregate_batch(batch))
            emitted += 1
            if emitted >= 512:
                break

    def on_stop(self) -> None:
        # Final drain: give in-flight stamps one synchronous chance to land.
        from traceml_amd.core import gpu_timer

        backend = None
        try:
            backend = gpu_timer.get_backend()
        except Exception:
            backend = None
        if backend is not None:
            try:
                backend.synchronize_resolution()
            except Exception:
                pass
        self.sample()

    def pending_batches(self) -> int:
        return len(self._pending) + timing.queue_depth()
