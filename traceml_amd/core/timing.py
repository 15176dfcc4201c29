"""Dual-clock timing core.

Every traced phase produces a ``TimeEvent`` carrying a CPU wall-clock pair
(always) and a GPU timestamp-handle pair (when a GPU backend is active:
ring-stamp sequence numbers from the CDNA4 ``s_memrealtime`` kernel). Events
are buffered per step and flushed as one ``StepTimeBatch`` at step end; GPU
resolution happens later, non-blocking, in the sampler thread (no
``synchronize()`` anywhere on the hot path).

Reference contract mirrored: utils/timing.py:46-267 (TimeEvent/try_resolve,
step buffer + bounded queue with drop-oldest + warn, timed_region).
"""

from __future__ import annotations

import logging
import threading
import time
from collections import deque
from contextlib import contextmanager
from dataclasses import dataclass, field
from typing import Deque, List, Optional

from traceml_amd.core import event_names, gpu_timer
from traceml_amd.core.arming import is_tracing_armed

logger = logging.getLogger(__name__)

STEP_TIME_QUEUE_MAX = 2048

#: Phases that never record GPU timestamps (host-side waits).
_CPU_ONLY_EVENTS = frozenset({event_names.DATALOADER})


@dataclass(slots=True)
class TimeEvent:
    name: str
    device: str
    cpu_start: float
    cpu_end: Optional[float] = None
    gpu_start: Optional[int] = None
    gpu_end: Optional[int] = None
    gpu_ms: Optional[float] = None
    _gpu_done: bool = field(default=False, repr=False)

    @property
    def cpu_ms(self) -> Optional[float]:
        if self.cpu_end is None:
            return None
        return (self.cpu_end - self.cpu_start) * 1000.0

    @property
    def has_gpu(self) -> bool:
        return self.gpu_start is not None and self.gpu_end is not None

    def try_resolve(self) -> bool:
        """Non-blocking resolution of the GPU pair.

        Returns True when this event needs no further work (no GPU pair, or
        the GPU elapsed time has been computed). Never blocks: a pending
        stamp just returns False and is retried on the next sampler tick.
        """
        if self._gpu_done or not self.has_gpu:
            return True
        backend = gpu_timer.get_backend_or_none()
        if backend is None:
            # GPU handles without a backend: device gone, drop the GPU side.
            self._gpu_done = True
            return True
        if not (backend.ready(self.gpu_start) and backend.ready(self.gpu_end)):
            return False
        ms = backend.elapsed_ms(self.gpu_start, self.gpu_end)
        if ms is not None:
            self.gpu_ms = max(0.0, ms)
        backend.release(self.gpu_start)
        backend.release(self.gpu_end)
        self._gpu_done = True
        return True


@dataclass
class StepTimeBatch:
    step: int
    events: List[TimeEvent]
    flushed_at: float


# ---------------------------------------------------------------------------
# Step buffer + bounded handoff queue (instrumentation thread -> sampler)
# ---------------------------------------------------------------------------

_buffer_lock = threading.Lock()
_step_buffer: List[TimeEvent] = []

_queue_lock = threading.Lock()
_step_time_queue: Deque[StepTimeBatch] = deque()
_dropped_batches = 0


_warned_timer_unavailable = False


def _gpu_mark_if_enabled(name: str) -> Optional[int]:
    if name in _CPU_ONLY_EVENTS:
        return None
    # Fail-open here: the loud check for a GPU machine with a missing native
    # extension happens ONCE at init() preflight (sdk/initial.py), never
    # inside the user's hot loop — telemetry must not crash training
    # (reference principle: docs/developer_guide/architecture.md:54-59).
    # The preflight is what keeps the native-path signal trustworthy; by the
    # time a mark runs, an unavailable timer only degrades to the CPU clock
    # with a single warning.
    global _warned_timer_unavailable
    try:
        backend = gpu_timer.get_backend()
    except gpu_timer.GpuTimerUnavailable as exc:
        if not _warned_timer_unavailable:
            _warned_timer_unavailable = True
            logger.warning(
                "traceml_amd: GPU timing disabled for this run "
                "(falling back to CPU wall clock): %s",
                exc,
            )
        return None
    if backend is None:
        return None
    try:
        return backend.mark()
    except Exception:
        logger.debug("traceml_amd: GPU mark failed", exc_info=True)
        return None


def _device_string(has_gpu_mark: bool) -> str:
    return "cuda" if has_gpu_mark else "cpu"


def record_event(event: TimeEvent) -> None:
    """Append a completed TimeEvent to the current step buffer."""
    with _buffer_lock:
        _step_buffer.append(event)


def open_event(name: str) -> TimeEvent:
    """Start a TimeEvent: CPU wall stamp + optional GPU stamp, no buffering yet."""
    gpu_start = _gpu_mark_if_enabled(name)
    return TimeEvent(
        name=name,
        device=_device_string(gpu_start is not None),
        cpu_start=time.time(),
        gpu_start=gpu_start,
    )


def close_event(event: TimeEvent) -> None:
    """Finish a TimeEvent (CPU + GPU end stamps) and buffer it."""
    event.cpu_end = time.time()
    if event.gpu_start is not None:
        event.gpu_end = _gpu_mark_if_enabled(event.name)
        if event.gpu_end is None:
            # Unpaired start: abandon the GPU side rather than resolve garbage.
            event.gpu_start = None
            event.device = "cpu"
    record_event(event)


@contextmanager
def timed_region(name: str):
    """Bracket a phase. Cheap no-op unless tracing is armed."""
    if not is_tracing_armed():
        yield
        return
    event = open_event(name)
    try:
        yield
    finally:
        close_event(event)


# Guarded by _summary_lock: written at flush time on the training thread,
# read by the rank-stats launch path (today also the training thread, but a
# sampler-side reader must never see a half-updated dict).
_summary_lock = threading.Lock()
_last_cpu_summary: dict = {}

#: event name -> rank-stats summary key (CPU wall clock, available at flush
#: time without waiting for GPU resolution; feeds the RCCL rank-stats gather)
_SUMMARY_KEYS = {
    event_names.DATALOADER: "input_ms",
    event_names.FORWARD: "forward_ms",
    event_names.BACKWARD: "backward_ms",
    event_names.OPTIMIZER: "optimizer_ms",
    event_names.STEP_TIME: "step_ms",
    event_names.DDP_COMM: "ddp_comm_ms",
}


def last_step_cpu_summary() -> dict:
    with _summary_lock:
        return dict(_last_cpu_summary)


def flush_step_time_buffer(step: int) -> None:
    """Move the per-step buffer into the bounded handoff queue as one batch."""
    global _dropped_batches
    with _buffer_lock:
        if not _step_buffer:
            return
        events = list(_step_buffer)
        _step_buffer.clear()
    # the CPU summary feeds the RCCL rank-stats gather; skip the work
    # entirely when no exchange is active (single-process runs)
    from traceml_amd.parallel.rank_stats import get_active_exchange

    if get_active_exchange() is not None:
        summary: dict = {}
        for event in events:
            key = _SUMMARY_KEYS.get(event.name)
            if key is not None and event.cpu_end is not None:
                summary[key] = summary.get(key, 0.0) + (
                    (event.cpu_end - event.cpu_start) * 1000.0
                )
        with _summary_lock:
            _last_cpu_summary.clear()
            _last_cpu_summary.update(summary)
    batch = StepTimeBatch(step=step, events=events, flushed_at=time.time())
    with _queue_lock:
        if len(_step_time_queue) >= STEP_TIME_QUEUE_MAX:
            _step_time_queue.popleft()
            _dropped_batches += 1
            if _dropped_batches in (1, 100, 1000):
                logger.warning(
                    "traceml_amd: step-time queue full, dropped %d batch(es)",
                    _dropped_batches,
                )
        _step_time_queue.append(batch)


def drain_step_time_queue(max_batches: Optional[int] = None) -> List[StepTimeBatch]:
    """Sampler-side: remove and return up to max_batches oldest batches."""
    out: List[StepTimeBatch] = []
    with _queue_lock:
        while _step_time_queue and (max_batches is None or len(out) < max_batches):
            out.append(_step_time_queue.popleft())
    return out


def queue_depth() -> int:
    with _queue_lock:
        return len(_step_time_queue)


def clear_for_tests() -> None:
    global _dropped_batches
    with _buffer_lock:
        _step_buffer.clear()
    with _queue_lock:
        _step_time_queue.clear()
    _dropped_batches = 0
