"""Optimizer-step timing via torch's global optimizer pre/post hooks.

Uses ``torch.optim.optimizer.register_optimizer_step_pre_hook`` /
``register_optimizer_step_post_hook`` (global, any optimizer instance).
Auto mode only; manual mode uses ``wrap_optimizer``
(reference: instrumentation/hooks/optimizer_hooks.py:17-103).
"""

from __future__ import annotations

import threading
from typing import Optional

from traceml_amd.core import event_names
from traceml_amd.core.arming import is_tracing_armed, phase_flags
from traceml_amd.core.timing import TimeEvent, close_event, open_event

_installed = False
_pre_handle = None
_post_handle = None


class _OpenEvents(threading.local):
    def __init__(self) -> None:
        self.current: Optional[TimeEvent] = None


_open = _OpenEvents()


def _pre_hook(optimizer, args, kwargs):
    flags = phase_flags()
    if not (flags.optimizer_enabled and is_tracing_armed()):
        return
    if _open.current is not None:
        return
    _open.current = open_event(event_names.OPTIMIZER)


def _post_hook(optimizer, args, kwargs):
    event = _open.current
    if event is None:
        return
    _open.current = None
    close_event(event)


def ensure_optimizer_timing_installed() -> None:
    global _installed, _pre_handle, _post_handle
    if _installed:
        return
    from torch.optim.optimizer import (
        register_optimizer_step_post_hook,
        register_optimizer_step_pre_hook,
    )

    _pre_handle = register_optimizer_step_pre_hook(_pre_hook)
    _post_handle = register_optimizer_step_post_hook(_post_hook)
    _installed = True


def remove_optimizer_time_hooks() -> None:
    global _installed
    if not _installed:
        return
    for handle in (_pre_handle, _post_handle):
        try:
            handle.remove()
        except Exception:
            pass
    _installed = False


def abandon_open_optimizer_event() -> None:
    _open.current = None
