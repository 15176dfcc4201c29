"""Manual/selective-mode wrappers (reference: sdk/wrappers.py:16-365).

Per-instance proxies that time exactly the wrapped object, refusing
double-instrumentation when the corresponding auto patch is active.
"""

from __future__ import annotations

import time
from typing import Any

from traceml_amd.core import event_names
from traceml_amd.core.arming import is_tracing_armed
from traceml_amd.core.timing import TimeEvent, close_event, open_event, record_event
from traceml_amd.instrumentation.h2d_filter import should_time_h2d
from traceml_amd.sdk import initial


def _refuse_if_auto(patch_field: str, what: str) -> None:
    config = initial.get_active_config()
    if config is not None and not config.noop and getattr(config, patch_field):
        raise RuntimeError(
            f"traceml_amd: {what} is already auto-instrumented "
            f"(init mode={config.mode!r}); don't wrap it manually too"
        )


def wrap_dataloader_fetch(obj: Any) -> Any:
    """Wrap an iterable/dataloader so each fetch is timed as dataloader_next."""
    _refuse_if_auto("patch_dataloader", "DataLoader fetch")

    class _Wrapped:
        def __init__(self, inner):
            self._inner = inner

        def __iter__(self):
            it = iter(self._inner)
            while True:
                if not is_tracing_armed():
                    try:
                        yield next(it)
                    except StopIteration:
                        return
                    continue
                cpu_start = time.time()
                try:
                    value = next(it)
                except StopIteration:
                    return
                record_event(
                    TimeEvent(
                        name=event_names.DATALOADER,
                        device="cpu",
                        cpu_start=cpu_start,
                        cpu_end=time.time(),
                    )
                )
                yield value

        def __len__(self):
            return len(self._inner)

        def __getattr__(self, name):
            return getattr(self._inner, name)

    return _Wrapped(obj)


def wrap_forward(model):
    """Wrap a model's forward so each call is timed as forward_time."""
    _refuse_if_auto("patch_forward", "forward")
    original_forward = model.forward

    def forward(*args, **kwargs):
        if not is_tracing_armed():
            return original_forward(*args, **kwargs)
        event = open_event(event_names.FORWARD)
        try:
            return original_forward(*args, **kwargs)
        finally:
            close_event(event)

    model.forward = forward
    return model


def wrap_backward(loss: Any) -> Any:
    """Wrap a loss tensor so ``loss.backward()`` is timed as backward_time."""
    _refuse_if_auto("patch_backward", "backward")

    class _WrappedLoss:
        def __init__(self, inner):
            object.__setattr__(self, "_inner", inner)

        def backward(self, *args, **kwargs):
            inner = object.__getattribute__(self, "_inner")
            if not is_tracing_armed():
                return inner.backward(*args, **kwargs)
            event = open_event(event_names.BACKWARD)
            try:
                return inner.backward(*args, **kwargs)
            finally:
                close_event(event)

        def __getattr__(self, name):
            return getattr(object.__getattribute__(self, "_inner"), name)

    return _WrappedLoss(loss)


def wrap_optimizer(optimizer: Any) -> Any:
    """Wrap optimizer.step() so each call is timed as optimizer_step."""
    config = initial.get_active_config()
    if config is not None and not config.noop and config.mode == "auto":
        raise RuntimeError(
            "traceml_amd: optimizer steps are already timed by the global "
            "optimizer hooks in auto mode; don't wrap the optimizer too"
        )
    original_step = optimizer.step

    def step(*args, **kwargs):
        if not is_tracing_armed():
            return original_step(*args, **kwargs)
        event = open_event(event_names.OPTIMIZER)
        try:
            return original_step(*args, **kwargs)
        finally:
            close_event(event)

    optimizer.step = step
    return optimizer


class _WrappedH2D:
    """Callable proxy for a `.to(device)`-style transfer helper."""

    def __init__(self, fn):
        self._fn = fn

    def __call__(self, tensor, *args, **kwargs):
        if not is_tracing_armed() or not should_time_h2d(tensor, args, kwargs):
            return self._fn(tensor, *args, **kwargs)
        event = open_event(event_names.H2D)
        try:
            return self._fn(tensor, *args, **kwargs)
        finally:
            close_event(event)


def wrap_h2d(obj: Any) -> Any:
    """Wrap a transfer callable (e.g. ``lambda t: t.to('cuda')``)."""
    _refuse_if_auto("patch_h2d", "h2d")
    if callable(obj):
        return _WrappedH2D(obj)
    raise TypeError("wrap_h2d expects a callable transfer function")
