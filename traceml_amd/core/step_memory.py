"""Per-step HIP caching-allocator watermarks.

On ROCm, ``torch.cuda.reset_peak_memory_stats`` / ``max_memory_allocated`` /
``max_memory_reserved`` read the HIP caching allocator directly (the torch
"cuda" namespace IS HIP on a ROCm build — there is no CUDA anywhere in the
stack). Semantics mirrored from the reference (utils/step_memory.py:34-114):
watermarks are reset at step start and read at step end; on a machine
without a GPU the values are ``None`` (null means "not measured", never 0).

The device-capacity field is carried alongside so pressure diagnosis scales
to the MI355X's 288 GB HBM3E without hard-coded capacities anywhere else.
"""

from __future__ import annotations

import threading
import time
from collections import deque
from dataclasses import dataclass
from typing import Deque, List, Optional

STEP_MEMORY_QUEUE_MAX = 4096
#: allocator-churn stats (memory_stats dict) are read every Nth step —
#: cumulative counters make the sparse delta lossless for retries
CHURN_SAMPLE_EVERY = 16


@dataclass
class StepMemoryEvent:
    step: int
    timestamp: float
    peak_allocated_bytes: Optional[int]
    peak_reserved_bytes: Optional[int]
    device_capacity_bytes: Optional[int]
    device: Optional[str]
    #: HIP caching-allocator churn stats (torch.cuda.memory_stats — the
    #: native allocator on ROCm): per-step peak of ACTIVE bytes (allocated
    #: minus freed-but-cached), the number of blocking cudaMalloc retries
    #: this step (cache thrash: the allocator had to flush + re-reserve),
    #: and the live segment count (fragmentation proxy).
    active_peak_bytes: Optional[int] = None
    alloc_retries: Optional[int] = None
    segments: Optional[int] = None


_queue_lock = threading.Lock()
_queue: Deque[StepMemoryEvent] = deque()

#: most recent per-step peak-allocated watermark (bytes); feeds the RCCL
#: rank-stats gather so peers see each other's memory pressure. None until
#: the first GPU-measured step (null ≠ 0).
_last_peak_alloc: Optional[int] = None


def last_peak_alloc_bytes() -> Optional[int]:
    return _last_peak_alloc


def _cuda():
    try:
        import torch

        if torch.cuda.is_available():
            return torch.cuda
    except Exception:
        pass
    return None


class StepMemoryTracker:
    """Reset-at-start / record-at-end peak watermark tracker for one model."""

    def __init__(self, model=None) -> None:
        self._cuda = _cuda()
        self._device_index: Optional[int] = None
        self._capacity: Optional[int] = None
        self._last_retries: Optional[int] = None
        if self._cuda is not None:
            try:
                self._device_index = self._cuda.current_device()
                props = self._cuda.get_device_properties(self._device_index)
                self._capacity = int(props.total_memory)
            except Exception:
                self._cuda = None

    def reset(self) -> None:
        if self._cuda is None:
            return
        try:
            self._cuda.reset_peak_memory_stats(self._device_index)
        except Exception:
            pass

    def record(self, step: int) -> None:
        peak_alloc: Optional[int] = None
        peak_reserved: Optional[int] = None
        device: Optional[str] = None
        active_peak: Optional[int] = None
        retries_delta: Optional[int] = None
        segments: Optional[int] = None
        if self._cuda is not None:
            # ONE stats read per step: torch.cuda.max_memory_allocated and
            # max_memory_reserved each build the full ~100-entry allocator
            # stats dict internally (~40 µs apiece on a busy allocator), so
            # both peaks are read from a single memory_stats() call — and
            # the churn fields reuse the SAME dict on their sampled steps.
            try:
                stats = self._cuda.memory_stats(self._device_index)
                peak_alloc = int(stats.get("allocated_bytes.all.peak", 0))
                peak_reserved = int(stats.get("reserved_bytes.all.peak", 0))
                device = f"cuda:{self._device_index}"
            except Exception:
                stats = None
                peak_alloc = peak_reserved = None
            # Churn stats every CHURN_SAMPLE_EVERY steps: retries are
            # cumulative, so the delta over the sparse read still captures
            # EVERY retry in between; active-peak/segments are trend
            # signals where sparse is enough.
            if stats is not None and (
                step % CHURN_SAMPLE_EVERY == 0 or self._last_retries is None
            ):
                active_peak = stats.get("active_bytes.all.peak")
                segments = stats.get("segment.all.current")
                retries_total = stats.get("num_alloc_retries")
                if retries_total is not None:
                    if self._last_retries is not None:
                        retries_delta = int(retries_total - self._last_retries)
                    self._last_retries = int(retries_total)
        event = StepMemoryEvent(
            step=step,
            timestamp=time.time(),
            peak_allocated_bytes=peak_alloc,
            peak_reserved_bytes=peak_reserved,
            device_capacity_bytes=self._capacity,
            device=device,
            active_peak_bytes=active_peak,
            alloc_retries=retries_delta,
            segments=segments,
        )
        if peak_alloc is not None:
            global _last_peak_alloc
            _last_peak_alloc = peak_alloc
        with _queue_lock:
            if len(_queue) >= STEP_MEMORY_QUEUE_MAX:
                _queue.popleft()
            _queue.append(event)


def drain_step_memory_queue() -> List[StepMemoryEvent]:
    with _queue_lock:
        out = list(_queue)
        _queue.clear()
    return out


def clear_for_tests() -> None:
    global _last_peak_alloc
    with _queue_lock:
        _queue.clear()
    _last_peak_alloc = None
