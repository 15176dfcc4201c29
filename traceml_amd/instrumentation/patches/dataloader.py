"""DataLoader fetch timing.

Patches ``torch.utils.data.DataLoader.__iter__`` so every ``next()`` on the
returned iterator is timed as ``dataloader_next`` (CPU wall clock only — a
host-side wait). Events are recorded whenever tracing is armed; the fetch
for step N typically happens before ``trace_step`` is entered, and lands in
step N's buffer because flush happens at step end
(reference behavior: instrumentation/patches/dataloader_patch.py:9-39,
call-stack note SURVEY §3.3).
"""

from __future__ import annotations

import time

from traceml_amd.core import event_names
from traceml_amd.core.arming import is_tracing_armed
from traceml_amd.core.timing import TimeEvent, record_event

_original_iter = None
_patched = False


class _TimedDataLoaderIter:
    __slots__ = ("_inner",)

    def __init__(self, inner) -> None:
        self._inner = inner

    def __iter__(self):
        return self

    def __next__(self):
        if not is_tracing_armed():
            return next(self._inner)
        cpu_start = time.time()
        value = next(self._inner)
        record_event(
            TimeEvent(
                name=event_names.DATALOADER,
                device="cpu",
                cpu_start=cpu_start,
                cpu_end=time.time(),
            )
        )
        return value

    def __len__(self):
        return len(self._inner)

    def __getattr__(self, name):
        return getattr(self._inner, name)


def patch_dataloader() -> None:
    global _original_iter, _patched
    if _patched:
        return
    from torch.utils.data import DataLoader

    _original_iter = DataLoader.__iter__

    def __iter__(self):
        inner = _original_iter(self)
        if not is_tracing_armed():
            return inner
        return _TimedDataLoaderIter(inner)

    DataLoader.__iter__ = __iter__
    _patched = True


def unpatch_dataloader() -> None:
    global _patched
    if not _patched:
        return
    from torch.utils.data import DataLoader

    DataLoader.__iter__ = _original_iter
    _patched = False
