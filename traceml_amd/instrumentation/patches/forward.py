"""Forward-pass timing: outermost call of the trace_step target model only.

Patches ``nn.Module.__call__``. The fast bail path is two thread-local reads
(armed check is a module-global read, phase flag a TLS read); only when the
called module's identity is in the ``forward_targets`` set registered by
``trace_step`` — the user model plus its DDP ``.module`` / FSDP
``_fsdp_wrapped_module`` unwraps — is the call timed, and nested calls of a
target are skipped via a depth counter
(reference behavior: instrumentation/patches/forward_auto_timer_patch.py:34-77).
"""

from __future__ import annotations

from traceml_amd.core import event_names
from traceml_amd.core.arming import is_tracing_armed, phase_flags
from traceml_amd.core.timing import close_event, open_event

_original_call = None
_patched = False


def forward_target_ids(model) -> tuple:
    """Identity set: the wrapper and its unwrapped inner module(s)."""
    ids = [id(model)]
    inner = getattr(model, "module", None)  # DDP / DataParallel
    if inner is not None and hasattr(inner, "forward"):
        ids.append(id(inner))
    fsdp_inner = getattr(model, "_fsdp_wrapped_module", None)  # FSDP1
    if fsdp_inner is not None:
        ids.append(id(fsdp_inner))
    return tuple(ids)


def patch_forward() -> None:
    global _original_call, _patched
    if _patched:
        return
    import torch.nn as nn

    _original_call = nn.Module.__call__

    def __call__(self, *args, **kwargs):
        flags = phase_flags()
        if not (flags.forward_enabled and is_tracing_armed()):
            return _original_call(self, *args, **kwargs)
        if flags.forward_depth > 0 or id(self) not in flags.forward_targets:
            return _original_call(self, *args, **kwargs)
        flags.forward_depth += 1
        event = open_event(event_names.FORWARD)
        try:
            return _original_call(self, *args, **kwargs)
        finally:
            close_event(event)
            flags.forward_depth -= 1

    nn.Module.__call__ = __call__
    _patched = True


def unpatch_forward() -> None:
    global _patched
    if not _patched:
        return
    import torch.nn as nn

    nn.Module.__call__ = _original_call
    _patched = False
