"""Single step-end flush entry point (reference: utils/flush_buffers.py:15)."""

from __future__ import annotations

from traceml_amd.core.timing import flush_step_time_buffer


def flush_step_events(step: int) -> None:
    flush_step_time_buffer(step)
