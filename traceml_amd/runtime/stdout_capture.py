"""Per-rank stdout/stderr capture (reference: runtime/stdout_stderr_capture.py).

Tees sys.stdout/sys.stderr line-wise into a bounded in-memory queue (drained
by the stdout_stderr sampler) and a per-rank log file, without altering what
the user sees.
"""

from __future__ import annotations

import sys
import threading
import time
from collections import deque
from typing import Deque, List, Optional, TextIO

QUEUE_MAX = 2000


class _TeeStream:
    def __init__(self, inner: TextIO, stream_name: str, capture) -> None:
        self._inner = inner
        self._name = stream_name
        self._capture = capture
        self._buffer = ""

    def write(self, text: str) -> int:
        result = self._inner.write(text)
        self._buffer += text
        while "\n" in self._buffer:
            line, self._buffer = self._buffer.split("\n", 1)
            self._capture._record(self._name, line)
        return result

    def flush(self) -> None:
        self._inner.flush()

    def __getattr__(self, name):
        return getattr(self._inner, name)


class StreamCapture:
    def __init__(self, log_path: Optional[str] = None) -> None:
        self._lock = threading.Lock()
        self._queue: Deque[dict] = deque()
        self._log_file: Optional[TextIO] = None
        self._log_path = log_path
        self._installed = False
        self._orig_stdout: Optional[TextIO] = None
        self._orig_stderr: Optional[TextIO] = None

    def _record(self, stream: str, line: str) -> None:
        row = {"timestamp": time.time(), "stream": stream, "line": line[:4096]}
        with self._lock:
            if len(self._queue) >= QUEUE_MAX:
                self._queue.popleft()
            self._queue.append(row)
            if self._log_file is not None:
                try:
                    self._log_file.write(f"[{stream}] {line}\n")
                except OSError:
                    pass

    def install(self) -> None:
        if self._installed:
            return
        if self._log_path:
            import os

            os.makedirs(os.path.dirname(self._log_path), exist_ok=True)
            self._log_file = open(self._log_path, "a", encoding="utf-8")
        self._orig_stdout = sys.stdout
        self._orig_stderr = sys.stderr
        sys.stdout = _TeeStream(sys.stdout, "stdout", self)
        sys.stderr = _TeeStream(sys.stderr, "stderr", self)
        self._installed = True

    def uninstall(self) -> None:
        if not self._installed:
            return
        sys.stdout = self._orig_stdout
        sys.stderr = self._orig_stderr
        if self._log_file is not None:
            try:
                self._log_file.close()
            except OSError:
                pass
            self._log_file = None
        self._installed = False

    def drain(self) -> List[dict]:
        with self._lock:
            out = list(self._queue)
            self._queue.clear()
        return out


_active: Optional[StreamCapture] = None


def install_stream_capture(log_path: Optional[str] = None) -> StreamCapture:
    global _active
    if _active is None:
        _active = StreamCapture(log_path)
        _active.install()
    return _active


def get_active_capture() -> Optional[StreamCapture]:
    return _active


def reset_for_tests() -> None:
    global _active
    if _active is not None:
        _active.uninstall()
        _active = None
