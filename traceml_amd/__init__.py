"""traceml_amd — MI355X-native training-step profiler.

A brand-new AMD-native observability framework with the capability set of
traceopt-ai/traceml (reference: /root/reference/src/traceml_ai/__init__.py:20-32):
it answers "why is my PyTorch training slow?" by timing the training-step
phases (dataloader / H2D / forward / backward / optimizer / DDP-comm) and
sampling system/process telemetry, shipping everything to an out-of-process
aggregator that persists to SQLite, runs rule-based diagnosis, and writes
``final_summary.{json,txt}``.

The measurement layer is MI355X-native by design:

* GPU phase timing uses a pooled ``hipEvent_t`` C++/HIP extension plus a
  hand-written CDNA4 ring-stamp kernel reading ``s_memrealtime`` (gfx950)
  for sub-microsecond on-device timestamps resolved by plain host reads of
  pinned memory (no ``hipEventQuery`` syscall on the hot resolve path).
* System telemetry comes from **amdsmi** (GPU util / VRAM / temp / power) —
  no NVML anywhere.
* Per-rank step stats are all-gathered over **RCCL on xGMI** for low-latency
  rank-skew / straggler diagnosis, and the DDP gradient all-reduce is timed
  explicitly as a first-class ``ddp_comm`` phase.

Import is lazy and torch-free until a public symbol is touched
(mirrors reference ``__init__.py:50`` lazy ``__getattr__`` behavior).
"""

from typing import TYPE_CHECKING

from traceml_amd.version import __version__

_PUBLIC = {
    "init": ("traceml_amd.api", "init"),
    "start": ("traceml_amd.api", "start"),
    "trace_step": ("traceml_amd.api", "trace_step"),
    "summary": ("traceml_amd.api", "summary"),
    "final_summary": ("traceml_amd.api", "final_summary"),
    "wrap_dataloader_fetch": ("traceml_amd.api", "wrap_dataloader_fetch"),
    "wrap_forward": ("traceml_amd.api", "wrap_forward"),
    "wrap_backward": ("traceml_amd.api", "wrap_backward"),
    "wrap_optimizer": ("traceml_amd.api", "wrap_optimizer"),
    "wrap_h2d": ("traceml_amd.api", "wrap_h2d"),
    "deep_profile": ("traceml_amd.sdk.deep_profile", "deep_profile"),
}

__all__ = ["__version__", *list(_PUBLIC)]


def __getattr__(name: str):
    if name in _PUBLIC:
        import importlib

        module_name, attr = _PUBLIC[name]
        module = importlib.import_module(module_name)
        value = getattr(module, attr)
        globals()[name] = value
        return value
    raise AttributeError(f"module 'traceml_amd' has no attribute {name!r}")


def __dir__():
    return sorted(set(globals()) | set(_PUBLIC))


if TYPE_CHECKING:  # pragma: no cover
    from traceml_amd.api import (  # noqa: F401
        final_summary,
        init,
        start,
        summary,
        trace_step,
        wrap_backward,
        wrap_dataloader_fetch,
        wrap_forward,
        wrap_h2d,
        wrap_optimizer,
    )
