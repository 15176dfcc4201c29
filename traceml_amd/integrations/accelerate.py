"""HuggingFace Accelerate integration.

Accelerate scripts own their loop, so this is a thin bracket helper plus an
auto-mode init (the global patches work under Accelerate's wrappers — the
forward target unwrap handles the DDP module Accelerate builds).
"""

from __future__ import annotations


def init(**kwargs):
    import traceml_amd

    kwargs.setdefault("mode", "auto")
    config = traceml_amd.init(**kwargs)
    from traceml_amd.integrations._capability import warn_if_missing_streams

    warn_if_missing_streams("accelerate", config)
    return config


def trace_step(model):
    """Bracket one accelerate training step (alias for the SDK bracket; the
    model may be the accelerate-prepared wrapper, which is unwrapped via its
    ``.module`` chain for forward targeting)."""
    from traceml_amd.sdk.instrumentation import trace_step as _trace_step

    return _trace_step(model)
