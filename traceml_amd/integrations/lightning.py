"""PyTorch Lightning integration (reference: integrations/lightning.py:109-425).

Lightning owns its loop, so this callback uses MANUAL mode and times the
phases itself: it wraps ``pl_module.forward`` and the batch-to-device
transfer, opens timed regions in the backward/optimizer hooks, and brackets
each batch with the step envelope. ``init()`` configures traceml in manual
mode (no global patches — Lightning's internals would double-count).
"""

from __future__ import annotations

import logging
from typing import Any, Optional

from traceml_amd.core import event_names
from traceml_amd.core.timing import TimeEvent, close_event, open_event
from traceml_amd.core.arming import is_tracing_armed

logger = logging.getLogger(__name__)


def init(**kwargs):
    """Selective mode (reference: integrations/lightning.py:108-125): the
    callback owns forward/backward/optimizer timing, but the GLOBAL
    DataLoader patch supplies the input-wait stream and the Tensor.to patch
    supplies H2D — without them the summary reports INCOMPLETE DATA for
    the dataloader signal."""
    import traceml_amd

    kwargs.setdefault("mode", "custom")
    kwargs.setdefault("patch_dataloader", True)
    kwargs.setdefault("patch_h2d", True)
    kwargs.setdefault("patch_forward", False)
    kwargs.setdefault("patch_backward", False)
    config = traceml_amd.init(**kwargs)
    from traceml_amd.integrations._capability import warn_if_missing_streams

    warn_if_missing_streams("lightning", config)
    return config


def _lightning_callback_base():
    try:
        from lightning.pytorch.callbacks import Callback

        return Callback
    except Exception:
        try:
            from pytorch_lightning.callbacks import Callback

            return Callback
        except Exception:
            return object


_CallbackBase = _lightning_callback_base()


class TraceMLCallback(_CallbackBase):  # type: ignore[misc]
    """Manual-mode phase owner for Lightning training loops."""

    def __init__(self) -> None:
        self._step_ctx = None
        self._open_events: dict = {}
        self._wrapped_forward = False

    # -- helpers ------------------------------------------------------------

    def _open(self, key: str, name: str) -> None:
        if not is_tracing_armed() or key in self._open_events:
            return
        self._open_events[key] = open_event(name)

    def _close(self, key: str) -> None:
        event = self._open_events.pop(key, None)
        if event is not None:
            close_event(event)

    # -- step bracket --------------------------------------------------------

    def on_train_batch_start(self, trainer, pl_module, batch, batch_idx):
        from traceml_amd.sdk.instrumentation import trace_step

        if self._step_ctx is not None:  # self-heal leaked bracket
            try:
                self._step_ctx.__exit__(None, None, None)
            except Exception:
                pass
        self._wrap_forward_once(pl_module)
        self._step_ctx = trace_step(pl_module)
        self._step_ctx.__enter__()

    def on_train_batch_end(self, trainer, pl_module, outputs, batch, batch_idx):
        if self._step_ctx is not None:
            ctx, self._step_ctx = self._step_ctx, None
            ctx.__exit__(None, None, None)

    # -- phase timing ---------------------------------------------------------

    def _wrap_forward_once(self, pl_module) -> None:
        if self._wrapped_forward:
            return
        self._wrapped_forward = True
        original = pl_module.forward

        def forward(*args: Any, **kwargs: Any):
            if not is_tracing_armed():
                return original(*args, **kwargs)
            event = open_event(event_names.FORWARD)
            try:
                return original(*args, **kwargs)
            finally:
                close_event(event)

        pl_module.forward = forward

    def on_before_backward(self, trainer, pl_module, loss):
        self._open("backward", event_names.BACKWARD)

    def on_after_backward(self, trainer, pl_module):
        self._close("backward")

    def on_before_optimizer_step(self, trainer, pl_module, optimizer):
        self._open("optimizer", event_names.OPTIMIZER)

    def on_before_zero_grad(self, trainer, pl_module, optimizer):
        self._close("optimizer")

    def on_train_end(self, trainer, pl_module):
        for key in list(self._open_events):
            self._close(key)


def wrap_batch_to_device(transfer_fn):
    """Wrap a LightningModule.transfer_batch_to_device-style callable so the
    move is timed as the H2D phase."""

    def wrapped(batch, device, dataloader_idx=0):
        if not is_tracing_armed():
            return transfer_fn(batch, device, dataloader_idx)
        event = open_event(event_names.H2D)
        try:
            return transfer_fn(batch, device, dataloader_idx)
        finally:
            close_event(event)

    return wrapped
