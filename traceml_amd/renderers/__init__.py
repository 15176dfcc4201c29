"""View-model builders shared by every display surface (CLI / web
dashboard / HTML report) — SQLite → plain dicts, no UI toolkit imports
(reference: renderers/ ~3.1k LoC of per-surface view models)."""

from traceml_amd.renderers.views import (
    live_view,
    memory_view,
    step_time_view,
    system_view,
)
from traceml_amd.renderers.comm import render_comm
from traceml_amd.renderers.process import render_process
from traceml_amd.renderers.step_memory import render_step_memory
from traceml_amd.renderers.step_time import render_step_time
from traceml_amd.renderers.system import render_system

__all__ = [
    "live_view",
    "step_time_view",
    "memory_view",
    "system_view",
    "render_step_time",
    "render_step_memory",
    "render_system",
    "render_process",
    "render_comm",
]
