"""Runtime-environment sampler: drains the one-shot topology/strategy queue
(reference: samplers/runtime_environment_sampler.py:20-33)."""

from __future__ import annotations

from traceml_amd.runtime import environment
from traceml_amd.samplers.base import BaseSampler

TABLE = "runtime_environment"


class RuntimeEnvironmentSampler(BaseSampler):
    name = "runtime_environment"

    def _sample(self) -> None:
        for info in environment.drain_runtime_environment():
            self.database.add_record(TABLE, info.to_row())
