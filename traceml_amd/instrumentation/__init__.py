"""Monkeypatch + hook instrumentation layer.

Patches are installed once per process by ``init()`` and stay inert until
the process is armed AND the current thread is inside ``trace_step``
(two cheap reads on the bail path). Reference layout:
instrumentation/patches/* and instrumentation/hooks/* in traceml_ai.
"""

from traceml_amd.instrumentation.patches.dataloader import (
    patch_dataloader,
    unpatch_dataloader,
)
from traceml_amd.instrumentation.patches.forward import (
    patch_forward,
    unpatch_forward,
)
from traceml_amd.instrumentation.patches.backward import (
    patch_backward,
    unpatch_backward,
)
from traceml_amd.instrumentation.patches.h2d import patch_h2d, unpatch_h2d
from traceml_amd.instrumentation.hooks.optimizer_hooks import (
    ensure_optimizer_timing_installed,
    remove_optimizer_time_hooks,
)

__all__ = [
    "patch_dataloader",
    "unpatch_dataloader",
    "patch_forward",
    "unpatch_forward",
    "patch_backward",
    "unpatch_backward",
    "patch_h2d",
    "unpatch_h2d",
    "ensure_optimizer_timing_installed",
    "remove_optimizer_time_hooks",
]
