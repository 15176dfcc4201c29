"""GPU timestamp backends for phase timing.

Reference equivalent: pooled ``torch.cuda.Event`` timing
(utils/cuda_event_pool.py:25-52, utils/timing.py:68-93). The MI355X-native
design replaces it with the ``_traceml_hip`` C++/HIP extension, which offers
two on-device clocks:

* **Ring stamps** (primary): a hand-written CDNA4 kernel reads
  ``s_memrealtime`` (the constant ~100 MHz real-time counter on gfx950) and
  writes ``{ticks, seq}`` into a slot of a pinned host ring buffer with a
  system-scope release. Recording is one tiny async kernel launch on the
  current HIP stream; *resolution is a plain host memory read* — strictly
  cheaper than ``hipEventQuery`` and consistent across streams, which plain
  hipEvent pairs are not.
* **HIP event pool** (secondary): classic ``hipEvent_t`` pairs
  (``hipEventCreateWithFlags``/``Record``/``Query``/``ElapsedTime``), used to
  cross-validate the ring clock and to time DDP communication buckets.

On a machine without a GPU the backend is ``None`` and all timing falls back
to the CPU wall clock (the analyzer's clock-selection rule handles this).
On a GPU machine the native extension is REQUIRED: ``preflight_check()``
(run once from ``init()``) raises loudly when it cannot load, BEFORE any
training step; after that gate the hot path fails open to the CPU clock
(telemetry must never crash training — reference architecture.md:54-59).
Set ``TRACEML_AMD_GPU_TIMER=off`` to explicitly disable GPU timing, or
``=torch`` to use torch.cuda.Event (debug only).
"""

from __future__ import annotations

import os
import threading
from typing import Optional

_lock = threading.Lock()
_backend: Optional["RingStampBackend"] = None
_backend_resolved = False


class GpuTimerUnavailable(RuntimeError):
    pass


def _torch():
    import torch

    return torch


class RingStampBackend:
    """Primary backend: CDNA4 s_memrealtime ring stamps via _traceml_hip."""

    name = "hip_ring"

    def __init__(self) -> None:
        from traceml_amd.ops import hip_ext

        self._ext = hip_ext.load_extension()
        torch = _torch()
        device = torch.cuda.current_device()
        self._ext.init(device, int(os.environ.get("TRACEML_AMD_RING_SLOTS", "65536")))
        self._torch = torch
        self._device = device
        # torch.cuda.current_stream() builds a Python Stream object per call
        # (~5-10 µs); the private raw-stream hook returns the bare HIP
        # stream handle (the same shortcut torch.compile's generated code
        # uses). mark() runs ~12x per traced step — this is the bracket's
        # single hottest line.
        self._raw_stream = getattr(torch._C, "_cuda_getCurrentRawStream", None)

    def mark(self) -> int:
        if self._raw_stream is not None:
            return self._ext.ring_mark(self._raw_stream(self._device))
        stream = self._torch.cuda.current_stream().cuda_stream
        return self._ext.ring_mark(stream)

    def ready(self, handle: int) -> bool:
        return self._ext.ring_ready(handle)

    def elapsed_ms(self, start: int, end: int) -> Optional[float]:
        ms = self._ext.ring_elapsed_ms(start, end)
        return None if ms < 0.0 else ms

    def release(self, handle: int) -> None:  # ring slots recycle themselves
        pass

    def synchronize_resolution(self) -> None:
        """Block until in-flight stamps land (drain/stop path only)."""
        self._torch.cuda.synchronize()


class TorchEventBackend:
    """Debug-only fallback using torch.cuda.Event (still HIP on ROCm, but
    not the native extension path). Enabled only via TRACEML_AMD_GPU_TIMER=torch."""

    name = "torch_event"

    def __init__(self) -> None:
        torch = _torch()
        self._torch = torch
        self._pool: list = []
        self._pool_lock = threading.Lock()
        self._max_pool = 2000

    def mark(self):
        with self._pool_lock:
            event = self._pool.pop() if self._pool else None
        if event is None:
            event = self._torch.cuda.Event(enable_timing=True)
        event.record()
        return event

    def ready(self, handle) -> bool:
        return handle.query()

    def elapsed_ms(self, start, end) -> Optional[float]:
        if not (start.query() and end.query()):
            return None
        return float(start.elapsed_time(end))

    def release(self, handle) -> None:
        with self._pool_lock:
            if len(self._pool) < self._max_pool:
                self._pool.append(handle)

    def synchronize_resolution(self) -> None:
        self._torch.cuda.synchronize()


def get_backend():
    """Return the process GPU-timer backend, or None when no GPU is present.

    Raises GpuTimerUnavailable on a GPU machine where the native extension
    cannot be loaded (unless an explicit override is set).
    """
    global _backend, _backend_resolved
    if _backend_resolved:
        return _backend
    with _lock:
        if _backend_resolved:
            return _backend
        mode = os.environ.get("TRACEML_AMD_GPU_TIMER", "native").lower()
        if mode == "off":
            _backend = None
            _backend_resolved = True
            return None
        try:
            import torch

            has_gpu = torch.cuda.is_available()
        except Exception:
            has_gpu = False
        if not has_gpu:
            _backend = None
            _backend_resolved = True
            return None
        if mode == "torch":
            _backend = TorchEventBackend()
        else:
            try:
                _backend = RingStampBackend()
            except Exception as exc:
                raise GpuTimerUnavailable(
                    "traceml_amd: GPU present but the native _traceml_hip "
                    "extension could not be loaded. Build it with "
                    "`python -m traceml_amd.ops.build_ext` (gfx950). "
                    f"Underlying error: {exc!r}"
                ) from exc
        _backend_resolved = True
        return _backend


def get_backend_or_none():
    """Fail-open variant for hot/sampler paths: an unavailable native timer
    degrades to None (CPU clock) instead of raising. The loud check lives in
    the init() preflight (sdk/initial.py)."""
    try:
        return get_backend()
    except GpuTimerUnavailable:
        return None


def preflight_check() -> None:
    """Verify the native extension is loadable when a GPU is visible.

    Called once from ``init()`` so a GPU machine with a stale/missing ``.so``
    refuses (or warns+no-ops, per the fail-open ladder) BEFORE training
    starts — never at first mark inside the user's hot loop. Deliberately
    does not bind a device: ranks may ``torch.cuda.set_device`` after
    ``init()``, so full backend construction stays lazy at first mark.
    """
    mode = os.environ.get("TRACEML_AMD_GPU_TIMER", "native").lower()
    if mode in ("off", "torch"):
        return
    try:
        import torch

        if not torch.cuda.is_available():
            return
    except Exception:
        return
    try:
        from traceml_amd.ops import hip_ext

        hip_ext.load_extension()
    except Exception as exc:
        raise GpuTimerUnavailable(
            "traceml_amd: GPU present but the native _traceml_hip extension "
            "could not be loaded. Build it with "
            "`python -m traceml_amd.ops.build_ext` (gfx950). "
            f"Underlying error: {exc!r}"
        ) from exc


def reset_backend_for_tests() -> None:
    global _backend, _backend_resolved
    with _lock:
        _backend = None
        _backend_resolved = False
