"""Backward-pass timing.

Patches ``torch.Tensor.backward`` AND ``torch.autograd.backward`` (either may
be the user's entry point). A thread-local re-entrancy flag prevents double
counting when ``Tensor.backward`` delegates to ``autograd.backward``
(reference: instrumentation/patches/backward_patch.py:27-78).
"""

from __future__ import annotations

import threading

from traceml_amd.core import event_names
from traceml_amd.core.arming import is_tracing_armed, phase_flags
from traceml_amd.core.timing import close_event, open_event

_original_tensor_backward = None
_original_autograd_backward = None
_patched = False


class _Reentry(threading.local):
    def __init__(self) -> None:
        self.depth = 0


_reentry = _Reentry()


def _should_time() -> bool:
    flags = phase_flags()
    return flags.backward_enabled and is_tracing_armed() and _reentry.depth == 0


def patch_backward() -> None:
    global _original_tensor_backward, _original_autograd_backward, _patched
    if _patched:
        return
    import torch

    _original_tensor_backward = torch.Tensor.backward
    _original_autograd_backward = torch.autograd.backward

    def tensor_backward(self, *args, **kwargs):
        if not _should_time():
            return _original_tensor_backward(self, *args, **kwargs)
        _reentry.depth += 1
        event = open_event(event_names.BACKWARD)
        try:
            return _original_tensor_backward(self, *args, **kwargs)
        finally:
            close_event(event)
            _reentry.depth -= 1

    def autograd_backward(*args, **kwargs):
        if not _should_time():
            return _original_autograd_backward(*args, **kwargs)
        _reentry.depth += 1
        event = open_event(event_names.BACKWARD)
        try:
            return _original_autograd_backward(*args, **kwargs)
        finally:
            close_event(event)
            _reentry.depth -= 1

    torch.Tensor.backward = tensor_backward
    torch.autograd.backward = autograd_backward
    _patched = True


def unpatch_backward() -> None:
    global _patched
    if not _patched:
        return
    import torch

    torch.Tensor.backward = _original_tensor_backward
    torch.autograd.backward = _original_autograd_backward
    _patched = False
