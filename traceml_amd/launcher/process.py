"""Child-process management: process groups, graceful termination, TCP
readiness, stderr tail capture (reference: launcher/process.py:34-283)."""

from __future__ import annotations

import os
import signal
import socket
import subprocess
import threading
import time
from typing import IO, List, Optional

STDERR_TAIL_BYTES = 64 * 1024


class StderrTailCapture:
    """Tee a child's stderr to ours while keeping the last 64 KiB for crash
    reports (reference: launcher/process.py:34-167)."""

    def __init__(self, stream: IO[bytes], echo: bool = True) -> None:
        self._stream = stream
        self._echo = echo
        self._tail = bytearray()
        self._lock = threading.Lock()
        self._thread = threading.Thread(target=self._pump, daemon=True)
        self._thread.start()

    def _pump(self) -> None:
        try:
            for chunk in iter(lambda: self._stream.read1(16384), b""):
                if self._echo:
                    try:
                        os.write(2, chunk)
                    except OSError:
                        pass
                with self._lock:
                    self._tail.extend(chunk)
                    if len(self._tail) > STDERR_TAIL_BYTES:
                        del self._tail[: len(self._tail) - STDERR_TAIL_BYTES]
        except (OSError, ValueError):
            pass

    def tail(self) -> str:
        with self._lock:
            return self._tail.decode("utf-8", errors="replace")

    def join(self, timeout: float = 1.0) -> None:
        self._thread.join(timeout)


def spawn_process_group(
    cmd: List[str], env: Optional[dict] = None, capture_stderr: bool = False
):
    """Spawn a child in its own process group so the whole tree can be
    signalled together."""
    kwargs = {}
    if capture_stderr:
        kwargs["stderr"] = subprocess.PIPE
    proc = subprocess.Popen(
        cmd,
        env=env,
        start_new_session=True,
        **kwargs,
    )
    tail = StderrTailCapture(proc.stderr) if capture_stderr else None
    return proc, tail


def terminate_process_group(
    proc: subprocess.Popen, grace_sec: float = 10.0
) -> Optional[int]:
    """SIGTERM the group, escalate to SIGKILL after the grace period."""
    if proc.poll() is not None:
        return proc.returncode
    try:
        os.killpg(proc.pid, signal.SIGTERM)
    except (ProcessLookupError, PermissionError):
        return proc.poll()
    deadline = time.time() + grace_sec
    while time.time() < deadline:
        if proc.poll() is not None:
            return proc.returncode
        time.sleep(0.1)
    try:
        os.killpg(proc.pid, signal.SIGKILL)
    except (ProcessLookupError, PermissionError):
        pass
    try:
        proc.wait(timeout=5.0)
    except subprocess.TimeoutExpired:
        pass
    return proc.poll()


def wait_for_tcp_listen(
    host: str, port: int, timeout: float = 30.0, proc: Optional[subprocess.Popen] = None
) -> bool:
    """Wait until something accepts on host:port; abort early if proc died."""
    deadline = time.time() + timeout
    while time.time() < deadline:
        if proc is not None and proc.poll() is not None:
            return False
        try:
            with socket.create_connection((host, port), timeout=0.5):
                return True
        except OSError:
            time.sleep(0.2)
    return False
