"""Host→device transfer timing: patches ``torch.Tensor.to``.

Only calls passing the ``should_time_h2d`` filter inside an armed step are
timed (reference: instrumentation/patches/h2d_patch.py:66-93). GPU-side
duration comes from ring stamps bracketing the copy on the current stream,
so async (`non_blocking=True`) copies are timed on the device clock, not by
host return time.
"""

from __future__ import annotations

from traceml_amd.core import event_names
from traceml_amd.core.arming import is_tracing_armed, phase_flags
from traceml_amd.core.timing import close_event, open_event
from traceml_amd.instrumentation.h2d_filter import should_time_h2d

_original_to = None
_original_cuda = None
_patched = False


def patch_h2d() -> None:
    global _original_to, _original_cuda, _patched
    if _patched:
        return
    import torch

    _original_to = torch.Tensor.to
    _original_cuda = torch.Tensor.cuda

    def to(self, *args, **kwargs):
        flags = phase_flags()
        if not (flags.h2d_enabled and is_tracing_armed()):
            return _original_to(self, *args, **kwargs)
        if not should_time_h2d(self, args, kwargs):
            return _original_to(self, *args, **kwargs)
        event = open_event(event_names.H2D)
        try:
            return _original_to(self, *args, **kwargs)
        finally:
            close_event(event)

    def cuda(self, device=None, non_blocking=False, **kwargs):
        flags = phase_flags()
        if not (flags.h2d_enabled and is_tracing_armed()):
            return _original_cuda(self, device, non_blocking, **kwargs)
        # .cuda() == .to("cuda:<device>"): same filter semantics
        if isinstance(self, torch.nn.Parameter) or self.device.type == "cuda":
            return _original_cuda(self, device, non_blocking, **kwargs)
        event = open_event(event_names.H2D)
        try:
            return _original_cuda(self, device, non_blocking, **kwargs)
        finally:
            close_event(event)

    torch.Tensor.to = to
    torch.Tensor.cuda = cuda
    _patched = True


def unpatch_h2d() -> None:
    global _patched
    if not _patched:
        return
    import torch

    torch.Tensor.to = _original_to
    torch.Tensor.cuda = _original_cuda
    _patched = False
