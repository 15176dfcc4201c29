"""HuggingFace Trainer integration (reference: integrations/huggingface.py:27-211).

``TraceMLTrainerCallback`` is a pure bracket: it enters ``trace_step`` on
``on_step_begin`` and exits on ``on_step_end``, so gradient-accumulation
micro-batches fold into one traced step (HF fires step callbacks at
optimizer-step granularity). A leaked context (exception paths inside the
Trainer) is self-healed at the next begin. ``TraceMLTrainer`` is a Trainer
subclass with the callback pre-installed.
"""

from __future__ import annotations

import logging
from typing import Optional

logger = logging.getLogger(__name__)


def init(**kwargs):
    """Convenience: auto-mode traceml init for HF scripts."""
    import traceml_amd

    kwargs.setdefault("mode", "auto")
    config = traceml_amd.init(**kwargs)
    from traceml_amd.integrations._capability import warn_if_missing_streams

    warn_if_missing_streams("huggingface", config)
    return config


try:
    from transformers import TrainerCallback

    _HAVE_TRANSFORMERS = True
except Exception:  # pragma: no cover
    TrainerCallback = object
    _HAVE_TRANSFORMERS = False


class TraceMLTrainerCallback(TrainerCallback):
    """Bracket each optimizer step with trace_step(model)."""

    def __init__(self) -> None:
        self._ctx = None
        self._model = None

    def _close_leaked(self) -> None:
        if self._ctx is not None:
            try:
                self._ctx.__exit__(None, None, None)
            except Exception:
                logger.debug("traceml_amd: leaked trace_step close failed",
                             exc_info=True)
            self._ctx = None

    def on_train_begin(self, args, state, control, model=None, **kwargs):
        self._model = model

    def on_step_begin(self, args, state, control, model=None, **kwargs):
        from traceml_amd.sdk.instrumentation import trace_step

        self._close_leaked()  # self-heal
        self._ctx = trace_step(model if model is not None else self._model)
        self._ctx.__enter__()

    def on_step_end(self, args, state, control, **kwargs):
        if self._ctx is not None:
            ctx, self._ctx = self._ctx, None
            ctx.__exit__(None, None, None)

    def on_train_end(self, args, state, control, **kwargs):
        self._close_leaked()


def TraceMLTrainer(*args, **kwargs):
    """Trainer factory with the TraceML callback pre-installed."""
    if not _HAVE_TRANSFORMERS:
        raise ImportError("transformers is required for TraceMLTrainer")
    from transformers import Trainer

    callbacks = list(kwargs.pop("callbacks", None) or [])
    if not any(isinstance(c, TraceMLTrainerCallback) for c in callbacks):
        callbacks.append(TraceMLTrainerCallback())
    kwargs["callbacks"] = callbacks
    return Trainer(*args, **kwargs)
