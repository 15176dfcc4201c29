"""View-model builders shared by every display surface (CLI / web
dashboard / HTML report) — SQLite → plain dicts, no UI toolkit imports
(reference: renderers/ ~3.1k LoC of per-surface view models)."""

from traceml_amd.renderers.views import (
    live_view,
    memory_view,
    step_time_view,
    system_view,
)

__all__ = ["live_view", "step_time_view", "memory_view", "system_view"]
