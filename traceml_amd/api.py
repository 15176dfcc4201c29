"""Public API facade (reference: api.py:15-403).

Lazy, torch-free until used; 10 public symbols re-exported by
``traceml_amd/__init__.py``.
"""

from __future__ import annotations

from typing import Any, ContextManager, Optional


def trace_step(model=None) -> ContextManager[None]:
    """Bracket one training step. Usage::

        with traceml_amd.trace_step(model):
            optimizer.zero_grad()
            loss = model(x).loss
            loss.backward()
            optimizer.step()
    """
    from traceml_amd.sdk.instrumentation import trace_step as _trace_step

    return _trace_step(model)


def init(**kwargs):
    """Initialize traceml_amd for this process (see sdk.initial.init)."""
    from traceml_amd.sdk.initial import init as _init

    return _init(**kwargs)


def start(**kwargs):
    """Alias for init() (reference api.py:345)."""
    return init(**kwargs)


def summary(timeout_sec: float = 30.0) -> dict:
    from traceml_amd.sdk.summary_client import summary as _summary

    return _summary(timeout_sec=timeout_sec)


def final_summary(
    wait: bool = True, timeout_sec: float = 60.0
) -> Optional[dict]:
    from traceml_amd.sdk.summary_client import final_summary as _final_summary

    return _final_summary(wait=wait, timeout_sec=timeout_sec)


def wrap_dataloader_fetch(obj: Any) -> Any:
    from traceml_amd.sdk.wrappers import wrap_dataloader_fetch as _wrap

    return _wrap(obj)


def wrap_forward(model: Any) -> Any:
    from traceml_amd.sdk.wrappers import wrap_forward as _wrap

    return _wrap(model)


def wrap_backward(loss: Any) -> Any:
    from traceml_amd.sdk.wrappers import wrap_backward as _wrap

    return _wrap(loss)


def wrap_optimizer(optimizer: Any) -> Any:
    from traceml_amd.sdk.wrappers import wrap_optimizer as _wrap

    return _wrap(optimizer)


def wrap_h2d(obj: Any) -> Any:
    from traceml_amd.sdk.wrappers import wrap_h2d as _wrap

    return _wrap(obj)
