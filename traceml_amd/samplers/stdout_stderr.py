"""Stdout/stderr sampler: drains the stream-capture queue. Only rank 0's
lines go over the wire (others stay in the per-rank log file; reference:
samplers/stdout_stderr_sampler.py:55-76)."""

from __future__ import annotations

from traceml_amd.runtime.stdout_capture import get_active_capture
from traceml_amd.samplers.base import BaseSampler

TABLE = "stdout_stderr"


class StdoutStderrSampler(BaseSampler):
    name = "stdout_stderr"

    def _sample(self) -> None:
        capture = get_active_capture()
        if capture is None:
            return
        for row in capture.drain():
            self.database.add_record(TABLE, row)
