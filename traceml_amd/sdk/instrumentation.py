"""``trace_step``: the per-step bracket (reference: sdk/instrumentation.py:127-260).

Enter: publish runtime environment once, reset HIP-allocator watermarks,
open the ``step_time`` envelope event (CPU wall + device ring stamp), arm
the per-thread phase enables + forward-target identity set, ensure
optimizer hooks (auto mode).

Exit (finally): close the envelope, disarm phase flags, advance the step
counter (gradient-accumulation micro-steps count as steps), record memory
watermarks, flush the step's events as one batch, notify the recording
budget, and kick the RCCL rank-stats exchange when active.
"""

from __future__ import annotations

import functools
import time
import time as _time


from traceml_amd.core import event_names
from traceml_amd.core.arming import is_tracing_armed, phase_flags
from traceml_amd.core.flush import flush_step_events
from traceml_amd.core.step_memory import StepMemoryTracker
from traceml_amd.core.timing import TimeEvent, close_event, open_event, record_event
from traceml_amd.instrumentation.hooks.optimizer_hooks import (
    ensure_optimizer_timing_installed,
)
from traceml_amd.instrumentation.patches.forward import forward_target_ids
from traceml_amd.runtime import environment, state
from traceml_amd.runtime.identity import resolve_runtime_identity
from traceml_amd.sdk import initial

# Hot-path caches: identity, the memory tracker and the forward-target set
# are process-stable; the per-step bracket must stay at a few microseconds
# of Python (class-based context manager — the generator form costs ~2x).
_cached_identity = None
_cached_mem_tracker = None
_cached_targets: dict = {}

# self-overhead accounting: time spent inside the bracket's own enter/exit
# (not user code), published via the process sampler
_self_cost_sec = 0.0
_self_steps = 0


class _NoopStep:
    __slots__ = ("_advance",)

    def __init__(self, advance: bool) -> None:
        self._advance = advance

    def __enter__(self):
        return None

    def __exit__(self, *exc):
        if self._advance:
            state.session_state().advance()
        return False


class _TraceStep:
    __slots__ = ("_model", "_config", "_recording", "_mem", "_event", "_t_enter")

    def __init__(self, model, config, recording) -> None:
        self._model = model
        self._config = config
        self._recording = recording

    def __enter__(self):
        global _cached_identity, _cached_mem_tracker
        self._t_enter = _time.perf_counter()
        model = self._model
        config = self._config
        if _cached_identity is None:
            _cached_identity = resolve_runtime_identity()
        environment.publish_runtime_environment_once(_cached_identity, model)
        if _cached_mem_tracker is None:
            _cached_mem_tracker = StepMemoryTracker(model)
        self._mem = _cached_mem_tracker
        self._mem.reset()

        flags = phase_flags()
        flags.in_step = True
        from traceml_amd.core import arming as _arming

        _arming.mark_step_open(True)
        if model is not None:
            key = id(model)
            targets = _cached_targets.get(key)
            if targets is None:
                targets = forward_target_ids(model)
                if len(_cached_targets) > 64:
                    _cached_targets.clear()
                _cached_targets[key] = targets
                _maybe_attach_ddp_timing(model, config)
            flags.forward_targets = targets
            flags.forward_enabled = config.patch_forward
        else:
            flags.forward_targets = ()
            flags.forward_enabled = False
        flags.backward_enabled = config.patch_backward
        flags.h2d_enabled = config.patch_h2d
        flags.optimizer_enabled = config.mode == "auto"
        if config.auto_optimizer_hooks:
            ensure_optimizer_timing_installed()
        # account the bookkeeping BEFORE the stamp launch: under deep
        # pipelining a kernel launch can block on queue backpressure for a
        # full step's worth of time — that is the stream's natural
        # submission throttle (paid somewhere in any run), not our cost.
        # Device-side stamp cost is measured separately (bench + rocprof).
        global _self_cost_sec
        _self_cost_sec += _time.perf_counter() - self._t_enter
        self._event = open_event(event_names.STEP_TIME)
        return None

    def __exit__(self, *exc):
        t_exit = _time.perf_counter()
        close_event(self._event)
        from traceml_amd.core import arming as _arming

        _arming.mark_step_open(False)
        flags = phase_flags()
        flags.in_step = False
        flags.forward_enabled = False
        flags.backward_enabled = False
        flags.h2d_enabled = False
        flags.optimizer_enabled = False
        flags.forward_targets = ()
        step = state.session_state().advance()
        self._mem.record(step)
        flush_step_events(step)
        self._recording.mark_trace_step_flushed()
        _kick_rank_stats(step)
        global _self_cost_sec, _self_steps
        _self_cost_sec += _time.perf_counter() - t_exit
        _self_steps += 1
        return False


def trace_step(model=None):
    config = initial.get_active_config()
    recording = state.recording_state()
    if (
        config is None
        or config.noop
        or not is_tracing_armed()
        or not recording.should_record_trace_events()
    ):
        return _NoopStep(advance=config is not None and not config.noop)
    return _TraceStep(model, config, recording)


def _maybe_attach_ddp_timing(model, config) -> None:
    """First trace_step of a DDP-wrapped model: auto-attach the ddp_comm
    timing hook (settings.ddp_comm_timing, on by default). Runs before the
    first backward, which is when DDP still accepts a comm hook."""
    if not getattr(config.settings, "ddp_comm_timing", True):
        return
    try:
        from torch.nn.parallel import DistributedDataParallel

        if not isinstance(model, DistributedDataParallel):
            return
        from traceml_amd.parallel.ddp_hook import attach_ddp_comm_timing

        attach_ddp_comm_timing(model)
    except Exception:
        pass  # user may have registered their own comm hook — keep theirs


def _kick_rank_stats(step: int) -> None:
    try:
        from traceml_amd.parallel.rank_stats import get_active_exchange

        exchange = get_active_exchange()
        if exchange is not None:
            exchange.on_step_flushed(step)
    except Exception:
        pass


def trace_time(name: str):
    """Decorator: time a function as a custom region inside the step."""

    def decorator(fn):
        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            if not is_tracing_armed():
                return fn(*args, **kwargs)
            cpu_start = time.time()
            try:
                return fn(*args, **kwargs)
            finally:
                record_event(
                    TimeEvent(
                        name=f"_traceml_user:{name}",
                        device="cpu",
                        cpu_start=cpu_start,
                        cpu_end=time.time(),
                    )
                )

        return wrapper

    return decorator


def self_overhead_us_per_step():
    """Mean microseconds the trace_step bracket itself spent per step
    (enter+exit bookkeeping; excludes user code and async GPU stamps)."""
    if _self_steps == 0:
        return None
    return _self_cost_sec / _self_steps * 1e6
