"""Process-local trace session state.

* ``TraceSessionState``: the step counter. Every ``trace_step`` context exit
  advances it (gradient-accumulation micro-batches count as steps, matching
  reference runtime/state.py:31).
* ``RecordingState``: RECORDING → DRAINING → COMPLETE lifecycle driven by the
  optional ``--trace-max-steps`` budget (reference runtime/state.py:100-146).
  In DRAINING the instrumentation stops recording new events but samplers
  keep resolving and shipping what is buffered; COMPLETE is reached when the
  queues drain.
"""

from __future__ import annotations

import threading
from typing import Optional

RECORDING = "recording"
DRAINING = "draining"
COMPLETE = "complete"


class TraceSessionState:
    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._step = 0

    @property
    def current_step(self) -> int:
        return self._step

    def advance(self) -> int:
        with self._lock:
            self._step += 1
            return self._step

    def reset_for_tests(self) -> None:
        with self._lock:
            self._step = 0


class RecordingState:
    def __init__(self, max_steps: Optional[int] = None) -> None:
        self._lock = threading.Lock()
        self._max_steps = max_steps if (max_steps or 0) > 0 else None
        self._phase = RECORDING
        self._flushed_steps = 0

    @property
    def phase(self) -> str:
        return self._phase

    def set_max_steps(self, max_steps: Optional[int]) -> None:
        with self._lock:
            self._max_steps = max_steps if (max_steps or 0) > 0 else None

    def should_record_trace_events(self) -> bool:
        return self._phase == RECORDING

    def mark_trace_step_flushed(self) -> None:
        with self._lock:
            self._flushed_steps += 1
            if self._max_steps is not None and self._flushed_steps >= self._max_steps:
                if self._phase == RECORDING:
                    self._phase = DRAINING

    def mark_drained(self) -> None:
        with self._lock:
            if self._phase == DRAINING:
                self._phase = COMPLETE

    def reset_for_tests(self) -> None:
        with self._lock:
            self._phase = RECORDING
            self._flushed_steps = 0
            self._max_steps = None


_session_state = TraceSessionState()
_recording_state = RecordingState()


def session_state() -> TraceSessionState:
    return _session_state


def recording_state() -> RecordingState:
    return _recording_state
