"""Step-memory sampler: drain-all of the per-step watermark queue
(reference: samplers/step_memory_sampler.py:33)."""

from __future__ import annotations

from traceml_amd.core import step_memory
from traceml_amd.samplers.base import BaseSampler

TABLE = "step_memory_samples"


class StepMemorySampler(BaseSampler):
    name = "step_memory"

    def _sample(self) -> None:
        for event in step_memory.drain_step_memory_queue():
            self.database.add_record(
                TABLE,
                {
                    "timestamp": event.timestamp,
                    "step": event.step,
                    "peak_allocated_bytes": event.peak_allocated_bytes,
                    "peak_reserved_bytes": event.peak_reserved_bytes,
                    "device_capacity_bytes": event.device_capacity_bytes,
                    "device": event.device,
                    "active_peak_bytes": event.active_peak_bytes,
                    "alloc_retries": event.alloc_retries,
                    "segments": event.segments,
                },
            )
