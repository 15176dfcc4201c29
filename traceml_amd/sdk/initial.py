"""``init()``: instrumentation policy + runtime startup with the fail-open ladder.

Modes (reference: sdk/initial.py:14-641):
* ``auto``   — install all global patches; trace_step arms them per step.
* ``manual`` — install nothing; the user times phases with ``wrap_*``.
* ``custom`` — selective: per-patch booleans decide.

Once-per-process: a second ``init()`` with a different policy warns and
returns the existing config. ``disabled=True`` (or env TRACEML_DISABLED=1)
yields an inert no-op config. Runtime startup is skipped when the executor
already registered a RuntimeHandle; otherwise the aggregator is probed over
TCP first and failures follow ``on_missing_aggregator`` (warn → fail-open
no-op, raise → RuntimeError).
"""

from __future__ import annotations

import os
import sys
import threading
from dataclasses import dataclass, field
from typing import Optional

from traceml_amd.core.arming import set_tracing_armed
from traceml_amd.runtime import lifecycle
from traceml_amd.runtime.settings import TraceMLSettings

VALID_MODES = ("auto", "manual", "custom")
VALID_ON_MISSING = ("warn", "raise")


@dataclass
class TraceMLInitConfig:
    mode: str = "auto"
    patch_dataloader: bool = True
    patch_forward: bool = True
    patch_backward: bool = True
    patch_h2d: bool = True
    disabled: bool = False
    noop: bool = False
    settings: TraceMLSettings = field(default_factory=TraceMLSettings)

    @property
    def auto_optimizer_hooks(self) -> bool:
        return self.mode == "auto" and not self.noop


_config_lock = threading.Lock()
_active_config: Optional[TraceMLInitConfig] = None
_warned_reinit = False


def get_active_config() -> Optional[TraceMLInitConfig]:
    return _active_config


def _noop_config(settings: TraceMLSettings) -> TraceMLInitConfig:
    set_tracing_armed(False)
    return TraceMLInitConfig(
        mode="auto",
        patch_dataloader=False,
        patch_forward=False,
        patch_backward=False,
        patch_h2d=False,
        disabled=True,
        noop=True,
        settings=settings,
    )


def _build_config(
    mode: str,
    patch_dataloader: Optional[bool],
    patch_forward: Optional[bool],
    patch_backward: Optional[bool],
    patch_h2d: Optional[bool],
    settings: TraceMLSettings,
) -> TraceMLInitConfig:
    if mode not in VALID_MODES:
        raise ValueError(f"traceml_amd.init: invalid mode {mode!r}; use one of {VALID_MODES}")
    overrides = dict(
        patch_dataloader=patch_dataloader,
        patch_forward=patch_forward,
        patch_backward=patch_backward,
        patch_h2d=patch_h2d,
    )
    if mode == "auto":
        for key, value in overrides.items():
            if value is False:
                # Selective disable inside auto is allowed but noted.
                pass
        resolved = {k: (True if v is None else bool(v)) for k, v in overrides.items()}
    elif mode == "manual":
        for key, value in overrides.items():
            if value:
                raise ValueError(
                    f"traceml_amd.init: mode='manual' conflicts with {key}=True "
                    "(manual mode uses wrap_* helpers, not global patches)"
                )
        resolved = {k: False for k in overrides}
    else:  # custom
        resolved = {k: bool(v) for k, v in overrides.items()}
    return TraceMLInitConfig(mode=mode, settings=settings, **resolved)


def _apply_requested_patches(config: TraceMLInitConfig) -> None:
    from traceml_amd import instrumentation as instr

    if config.patch_dataloader:
        instr.patch_dataloader()
    if config.patch_forward:
        instr.patch_forward()
    if config.patch_backward:
        instr.patch_backward()
    if config.patch_h2d:
        instr.patch_h2d()
    set_tracing_armed(True)


def _start_runtime_for_init(
    settings: TraceMLSettings,
    connect_timeout_sec: float,
    connect_retry_interval_sec: float,
    on_missing_aggregator: str,
) -> None:
    if lifecycle.get_active_runtime_handle() is not None:
        return  # executor (traceml-amd run) already started it
    reachable = lifecycle.wait_for_aggregator(
        settings.aggregator_host,
        settings.aggregator_port,
        timeout=connect_timeout_sec,
        retry_interval=connect_retry_interval_sec,
    )
    if not reachable:
        raise RuntimeError(
            "traceml_amd: aggregator not reachable at "
            f"{settings.aggregator_host}:{settings.aggregator_port} "
            f"after {connect_timeout_sec}s (start one with `traceml-amd serve` "
            "or launch via `traceml-amd run`)"
        )
    lifecycle.start_runtime(settings, fail_open=False)


def init(
    *,
    mode: str = "auto",
    patch_dataloader: Optional[bool] = None,
    patch_forward: Optional[bool] = None,
    patch_backward: Optional[bool] = None,
    patch_h2d: Optional[bool] = None,
    disabled: Optional[bool] = None,
    ui_mode: Optional[str] = None,
    interval: Optional[float] = None,
    logs_dir: Optional[str] = None,
    enable_logging: Optional[bool] = None,
    session_id: Optional[str] = None,
    aggregator_host: Optional[str] = None,
    aggregator_port: Optional[int] = None,
    connect_timeout_sec: float = 10.0,
    connect_retry_interval_sec: float = 0.25,
    on_missing_aggregator: Optional[str] = None,
    _source: str = "user",
) -> TraceMLInitConfig:
    global _active_config, _warned_reinit
    with _config_lock:
        if _active_config is not None:
            if not _warned_reinit:
                _warned_reinit = True
                print(
                    "[TraceML-AMD] init() called more than once in this process; "
                    "keeping the first configuration",
                    file=sys.stderr,
                )
            return _active_config

        settings = TraceMLSettings.from_env()
        if ui_mode is not None:
            settings.mode = ui_mode
        if interval is not None:
            settings.interval = float(interval)
        if logs_dir is not None:
            settings.logs_dir = logs_dir
        if enable_logging is not None:
            settings.enable_logging = bool(enable_logging)
        if session_id is not None:
            settings.session_id = session_id
        if aggregator_host is not None:
            settings.aggregator_host = aggregator_host
        if aggregator_port is not None:
            settings.aggregator_port = int(aggregator_port)

        env_disabled = os.environ.get("TRACEML_DISABLED", "").strip().lower() in (
            "1",
            "true",
            "yes",
        )
        if disabled or (disabled is None and (env_disabled or settings.disabled)):
            config = _noop_config(settings)
            _active_config = config
            return config

        on_missing = on_missing_aggregator or "warn"
        if on_missing not in VALID_ON_MISSING:
            raise ValueError(
                f"traceml_amd.init: on_missing_aggregator must be one of {VALID_ON_MISSING}"
            )

        config = _build_config(
            mode, patch_dataloader, patch_forward, patch_backward, patch_h2d, settings
        )

        try:
            # Loud native-timer check happens HERE (before any training step),
            # so a GPU box with a missing .so either refuses or runs no-op —
            # it never kills a training step from inside the profiler
            # (reference fail-open principle: architecture.md:54-59).
            from traceml_amd.core import gpu_timer

            gpu_timer.preflight_check()
            _start_runtime_for_init(
                settings,
                connect_timeout_sec,
                connect_retry_interval_sec,
                on_missing,
            )
        except Exception as exc:
            if on_missing == "raise":
                raise
            # Fail-open ladder: warn once to stderr, store a no-op config.
            print(
                f"[TraceML-AMD] disabled for this run: {exc}",
                file=sys.stderr,
            )
            config = _noop_config(settings)
            _active_config = config
            return config

        _apply_requested_patches(config)
        if settings.trace_max_steps:
            from traceml_amd.runtime import state

            state.recording_state().set_max_steps(settings.trace_max_steps)
        _active_config = config
        return config


def reset_for_tests() -> None:
    """Tear down patches + config so each test starts clean."""
    global _active_config, _warned_reinit
    from traceml_amd import instrumentation as instr
    from traceml_amd.core import step_memory, timing
    from traceml_amd.runtime import environment, state

    with _config_lock:
        set_tracing_armed(False)
        instr.unpatch_dataloader()
        instr.unpatch_forward()
        instr.unpatch_backward()
        instr.unpatch_h2d()
        instr.remove_optimizer_time_hooks()
        timing.clear_for_tests()
        step_memory.clear_for_tests()
        environment.reset_for_tests()
        from traceml_amd.sdk import instrumentation as _sdk_instr

        _sdk_instr._cached_identity = None
        _sdk_instr._cached_mem_tracker = None
        _sdk_instr._cached_targets.clear()
        _sdk_instr._self_cost_sec = 0.0
        _sdk_instr._self_steps = 0
        state.session_state().reset_for_tests()
        state.recording_state().reset_for_tests()
        _active_config = None
        _warned_reinit = False
