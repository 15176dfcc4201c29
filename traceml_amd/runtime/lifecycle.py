"""Runtime + aggregator lifecycle: in-process start/stop with fail-open.

* ``start_runtime``: builds + starts a TraceMLRuntime for this process. With
  ``fail_open=True`` any startup failure yields a ``NoOpRuntime`` and a
  single warning — telemetry must never crash training
  (reference: runtime/lifecycle.py:31-296).
* ``start_aggregator``: starts an in-process aggregator (used by `serve`,
  Ray, and tests; the launcher normally spawns it as its own process).
* A process-global ``RuntimeHandle`` lets the executor pre-start the runtime
  so a later ``traceml.init()`` in user code skips runtime startup.
"""

from __future__ import annotations

import atexit
import logging
import sys
import threading
import time
from typing import Optional

from traceml_amd.runtime.runtime import TraceMLRuntime
from traceml_amd.runtime.settings import TraceMLSettings
from traceml_amd.transport.tcp import probe_tcp

logger = logging.getLogger(__name__)


class NoOpRuntime:
    """Fail-open stand-in: every runtime operation is a no-op."""

    def start(self) -> None:
        pass

    def stop(self) -> None:
        pass


class RuntimeHandle:
    def __init__(self, runtime) -> None:
        self.runtime = runtime
        self._stopped = False
        self._lock = threading.Lock()

    def stop(self) -> None:
        with self._lock:
            if self._stopped:
                return
            self._stopped = True
        try:
            self.runtime.stop()
        except Exception:
            logger.debug("traceml_amd: runtime stop failed", exc_info=True)
        _clear_active_runtime_handle(self)


class AggregatorHandle:
    def __init__(self, aggregator) -> None:
        self.aggregator = aggregator
        self._stopped = False

    @property
    def port(self):
        return self.aggregator.port

    def stop(self) -> None:
        if self._stopped:
            return
        self._stopped = True
        try:
            self.aggregator.stop()
        except Exception:
            logger.debug("traceml_amd: aggregator stop failed", exc_info=True)


_handle_lock = threading.Lock()
_active_runtime_handle: Optional[RuntimeHandle] = None


def get_active_runtime_handle() -> Optional[RuntimeHandle]:
    with _handle_lock:
        return _active_runtime_handle


def _set_active_runtime_handle(handle: RuntimeHandle) -> None:
    global _active_runtime_handle
    with _handle_lock:
        _active_runtime_handle = handle


def _clear_active_runtime_handle(handle: RuntimeHandle) -> None:
    global _active_runtime_handle
    with _handle_lock:
        if _active_runtime_handle is handle:
            _active_runtime_handle = None


def wait_for_aggregator(
    host: str,
    port: int,
    timeout: float = 10.0,
    retry_interval: float = 0.25,
) -> bool:
    deadline = time.time() + max(0.0, timeout)
    while True:
        if probe_tcp(host, port, timeout=min(1.0, max(0.1, retry_interval))):
            return True
        if time.time() >= deadline:
            return False
        time.sleep(retry_interval)


def start_runtime(
    settings: Optional[TraceMLSettings] = None,
    fail_open: bool = True,
    register_atexit: bool = True,
) -> RuntimeHandle:
    settings = settings or TraceMLSettings.from_env()
    if settings.enable_logging and settings.logs_dir:
        try:
            from traceml_amd.runtime.session import get_session_id, session_dir
            from traceml_amd.utils.loggers import setup_error_logger

            setup_error_logger(
                session_dir(settings.logs_dir, get_session_id(settings.session_id))
            )
        except Exception:
            pass
    try:
        runtime = TraceMLRuntime(settings)
        runtime.start()
    except Exception as exc:
        if not fail_open:
            raise
        print(
            f"[TraceML-AMD] runtime startup failed, telemetry disabled: {exc!r}",
            file=sys.stderr,
        )
        handle = RuntimeHandle(NoOpRuntime())
        _set_active_runtime_handle(handle)
        return handle
    handle = RuntimeHandle(runtime)
    _set_active_runtime_handle(handle)
    if register_atexit:
        atexit.register(handle.stop)
    return handle


def start_aggregator(
    settings: Optional[TraceMLSettings] = None,
) -> AggregatorHandle:
    from traceml_amd.aggregator.aggregator import TraceMLAggregator

    settings = settings or TraceMLSettings.from_env()
    aggregator = TraceMLAggregator(settings)
    aggregator.start()
    return AggregatorHandle(aggregator)
