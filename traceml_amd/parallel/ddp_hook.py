"""Explicit DDP gradient-allreduce timing: a communication hook that turns
the reference's unmeasured "residual" proxy into a measured ``ddp_comm``
phase (SURVEY §2.0 / BASELINE.json north star).

Per bucket: a device timestamp (CDNA4 ring stamp) is recorded on the
autograd/default stream right before the allreduce launches (RCCL's stream
waits on it, so the stamp lower-bounds comm start), and the end stamp lands
on a dedicated side stream that waits on the collective's work — i.e. it
fires in stream order immediately after RCCL completes that bucket on the
device. ``s_memrealtime`` is globally consistent across streams, so the
pair is directly subtractable — something plain event pairs on different
streams do not guarantee.

DDP semantics are untouched: the hook returns the ProcessGroup's own
future (pre-divided gradients), exactly like the default allreduce hook.
On CPU (gloo) the collective is run synchronously and timed on the CPU
clock, which keeps multi-process CPU tests meaningful.
"""

from __future__ import annotations

import logging
import time
from typing import Optional

from traceml_amd.core import event_names
from traceml_amd.core.arming import any_step_open, is_tracing_armed
from traceml_amd.core.timing import TimeEvent, record_event

logger = logging.getLogger(__name__)


class _CommTimerState:
    def __init__(self, process_group=None) -> None:
        import torch
        import torch.distributed as dist

        self.pg = process_group if process_group is not None else dist.group.WORLD
        self.world_size = dist.get_world_size(self.pg)
        self.use_gpu = torch.cuda.is_available()
        self.side_stream = torch.cuda.Stream() if self.use_gpu else None


def _should_time() -> bool:
    # NOT the thread-local in_step: on GPU this hook runs on the autograd
    # worker thread, which never sees the training thread's TLS flags.
    return is_tracing_armed() and any_step_open()


def timed_allreduce_hook(state: _CommTimerState, bucket):
    import torch
    import torch.distributed as dist

    tensor = bucket.buffer()
    tensor.div_(state.world_size)

    if not _should_time():
        work = dist.all_reduce(tensor, group=state.pg, async_op=True)
        return work.get_future().then(lambda fut: fut.value()[0])

    if state.use_gpu:
        from traceml_amd.core import gpu_timer

        backend = gpu_timer.get_backend_or_none()
        event = TimeEvent(
            name=event_names.DDP_COMM,
            device="cuda",
            cpu_start=time.time(),
        )
        if backend is not None:
            event.gpu_start = backend.mark()  # current (autograd) stream
        work = dist.all_reduce(tensor, group=state.pg, async_op=True)
        if backend is not None and event.gpu_start is not None:
            with torch.cuda.stream(state.side_stream):
                work.wait()  # side stream waits on the RCCL stream
                event.gpu_end = backend.mark()  # fires right after comm completes
        event.cpu_end = time.time()
        record_event(event)
        return work.get_future().then(lambda fut: fut.value()[0])

    # CPU/gloo path: synchronous, CPU-clocked.
    cpu_start = time.time()
    dist.all_reduce(tensor, group=state.pg)
    record_event(
        TimeEvent(
            name=event_names.DDP_COMM,
            device="cpu",
            cpu_start=cpu_start,
            cpu_end=time.time(),
        )
    )
    fut = torch.futures.Future()
    fut.set_result(tensor)
    return fut


def attach_ddp_comm_timing(ddp_model, process_group=None) -> Optional[_CommTimerState]:
    """Register the timing comm hook on a DistributedDataParallel model.

    Idempotent: DDP allows exactly one comm hook per model, and both an
    explicit call (bench/user) and trace_step's auto-attach may race to be
    first — a second attach returns the existing state silently."""
    existing = getattr(ddp_model, "_traceml_comm_timer_state", None)
    if existing is not None:
        return existing
    try:
        state = _CommTimerState(process_group)
        ddp_model.register_comm_hook(state, timed_allreduce_hook)
        ddp_model._traceml_comm_timer_state = state
        return state
    except Exception:
        logger.warning("traceml_amd: could not attach DDP comm timing", exc_info=True)
        return None
