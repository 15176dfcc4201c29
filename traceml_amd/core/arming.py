"""Tracing arming gate + per-phase thread-local enables.

Global monkeypatches are installed once per process but stay inert unless
BOTH the process-wide armed flag is set (by ``init()``) AND the per-phase
thread-local enable is set (by ``trace_step``'s auto-timers). The fast-path
bail in every patch is therefore two cheap reads
(reference behavior: runtime/arming.py:17, forward_auto_timer_patch.py:51-56).
"""

from __future__ import annotations

import threading

_armed = False
_armed_lock = threading.Lock()


def set_tracing_armed(value: bool) -> None:
    global _armed
    with _armed_lock:
        _armed = bool(value)


def is_tracing_armed() -> bool:
    return _armed


class _PhaseFlags(threading.local):
    def __init__(self) -> None:
        self.in_step = False
        self.forward_enabled = False
        self.backward_enabled = False
        self.h2d_enabled = False
        self.optimizer_enabled = False
        # Re-entrancy depth for the outermost-forward-only rule.
        self.forward_depth = 0
        # Identity set of modules whose forward should be timed (the
        # trace_step target model + its DDP/FSDP unwrapped inner module).
        self.forward_targets: tuple[int, ...] = ()


_flags = _PhaseFlags()


def phase_flags() -> _PhaseFlags:
    return _flags


# Process-wide step-open indicator (NOT thread-local): consumers that run on
# other threads — the DDP comm hook fires on the autograd worker thread on
# GPU — need to know a step bracket is open. Single int mutated under the
# GIL by the (single) training thread's trace_step enter/exit.
_steps_open = 0


def mark_step_open(opened: bool) -> None:
    global _steps_open
    _steps_open += 1 if opened else -1
    if _steps_open < 0:
        _steps_open = 0


def any_step_open() -> bool:
    return _steps_open > 0
