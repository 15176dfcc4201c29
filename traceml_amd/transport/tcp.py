"""Length-prefixed msgpack TCP transport — the rank↔aggregator process boundary.

* ``TCPServer``: accept thread + one reader thread per client; each frame is
  a 4-byte big-endian length prefix followed by a msgpack document (either a
  single payload or a list batch). Decoded payloads land in a thread-safe
  queue; ``wait_for_data`` lets the aggregator loop block event-driven.
* ``TCPClient``: best-effort — ``send_batch`` never raises; on failure it
  drops the batch, closes the socket and reconnects lazily on the next send
  (reference: transport/tcp_transport.py:21-262).

Frames above ``MAX_FRAME_BYTES`` are rejected (malformed/hostile peer guard).
"""

from __future__ import annotations

import logging
import socket
import struct
import threading
from collections import deque
from typing import Any, List, Optional

from traceml_amd.transport import codec

logger = logging.getLogger(__name__)

LENGTH_PREFIX = struct.Struct(">I")
MAX_FRAME_BYTES = 64 * 1024 * 1024
#: Ingest queue bound — drop-oldest like every other queue in the pipeline
#: (step-time handoff, SQLite ingest, stdout capture); if the aggregator
#: loop stalls, fast clients must not grow host memory without limit.
SERVER_QUEUE_MAX = 100_000


class TCPServer:
    def __init__(self, bind_host: str = "127.0.0.1", port: int = 0) -> None:
        self._bind_host = bind_host
        self._requested_port = port
        self._sock: Optional[socket.socket] = None
        self._accept_thread: Optional[threading.Thread] = None
        self._client_threads: List[threading.Thread] = []
        self._stop = threading.Event()
        self._queue_lock = threading.Lock()
        self._queue: deque = deque()
        self._data_event = threading.Event()
        self.port: Optional[int] = None
        self.dropped = 0
        self._last_drop_warn = 0.0

    def start(self) -> None:
        sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        sock.bind((self._bind_host, self._requested_port))
        sock.listen(64)
        sock.settimeout(0.25)
        self._sock = sock
        self.port = sock.getsockname()[1]
        self._accept_thread = threading.Thread(
            target=self._accept_loop, name="traceml-tcp-accept", daemon=True
        )
        self._accept_thread.start()

    def _accept_loop(self) -> None:
        while not self._stop.is_set():
            try:
                client, _addr = self._sock.accept()
            except socket.timeout:
                continue
            except OSError:
                break
            client.settimeout(0.5)
            t = threading.Thread(
                target=self._handle_client,
                args=(client,),
                name="traceml-tcp-client",
                daemon=True,
            )
            t.start()
            # prune finished reader threads so long runs with reconnecting
            # clients don't accumulate dead references
            self._client_threads = [
                th for th in self._client_threads if th.is_alive()
            ]
            self._client_threads.append(t)

    def _handle_client(self, client: socket.socket) -> None:
        buffer = bytearray()
        try:
            while not self._stop.is_set():
                try:
                    chunk = client.recv(256 * 1024)
                except socket.timeout:
                    continue
                except OSError:
                    break
                if not chunk:
                    break
                buffer.extend(chunk)
                self._drain_frames(buffer)
        finally:
            try:
                client.close()
            except OSError:
                pass

    def _drain_frames(self, buffer: bytearray) -> None:
        while True:
            if len(buffer) < LENGTH_PREFIX.size:
                return
            (length,) = LENGTH_PREFIX.unpack_from(buffer, 0)
            if length > MAX_FRAME_BYTES:
                logger.warning("traceml_amd: oversized frame (%d bytes), dropping client buffer", length)
                buffer.clear()
                return
            if len(buffer) < LENGTH_PREFIX.size + length:
                return
            frame = bytes(buffer[LENGTH_PREFIX.size : LENGTH_PREFIX.size + length])
            del buffer[: LENGTH_PREFIX.size + length]
            try:
                payload = codec.decode(frame)
            except Exception:
                logger.debug("traceml_amd: malformed frame dropped", exc_info=True)
                continue
            items = payload if isinstance(payload, list) else [payload]
            with self._queue_lock:
                self._queue.extend(items)
                overflow = len(self._queue) - SERVER_QUEUE_MAX
                if overflow > 0:
                    for _ in range(overflow):
                        self._queue.popleft()
                    self.dropped += overflow
                    dropped = self.dropped
                else:
                    dropped = 0
            if dropped:
                import time as _time

                now = _time.time()
                if now - self._last_drop_warn > 10.0:
                    self._last_drop_warn = now
                    logger.warning(
                        "traceml_amd: aggregator ingest queue full, "
                        "dropped %d payload(s) total (oldest-first)",
                        dropped,
                    )
            self._data_event.set()

    def wait_for_data(self, timeout: float) -> bool:
        got = self._data_event.wait(timeout)
        if got:
            self._data_event.clear()
        return got

    def drain(self, max_items: Optional[int] = None) -> List[Any]:
        out: List[Any] = []
        with self._queue_lock:
            while self._queue and (max_items is None or len(out) < max_items):
                out.append(self._queue.popleft())
        return out

    def stop(self) -> None:
        self._stop.set()
        if self._sock is not None:
            try:
                self._sock.close()
            except OSError:
                pass
        if self._accept_thread is not None:
            self._accept_thread.join(timeout=2.0)


class TCPClient:
    def __init__(self, host: str, port: int, connect_timeout: float = 2.0) -> None:
        self._host = host
        self._port = port
        self._connect_timeout = connect_timeout
        self._sock: Optional[socket.socket] = None
        self._lock = threading.Lock()

    def _connect_locked(self) -> bool:
        if self._sock is not None:
            return True
        try:
            sock = socket.create_connection(
                (self._host, self._port), timeout=self._connect_timeout
            )
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            self._sock = sock
            return True
        except OSError:
            self._sock = None
            return False

    def connect(self) -> bool:
        with self._lock:
            return self._connect_locked()

    @property
    def connected(self) -> bool:
        return self._sock is not None

    def send_batch(self, payloads: List[Any]) -> bool:
        """One frame per tick. Never raises; False = dropped."""
        if not payloads:
            return True
        try:
            frame = codec.encode_batch(payloads)
        except Exception:
            logger.debug("traceml_amd: encode failed, batch dropped", exc_info=True)
            return False
        data = LENGTH_PREFIX.pack(len(frame)) + frame
        with self._lock:
            if not self._connect_locked():
                return False
            try:
                self._sock.sendall(data)
                return True
            except OSError:
                try:
                    self._sock.close()
                except OSError:
                    pass
                self._sock = None  # lazy reconnect on next send
                return False

    def close(self) -> None:
        with self._lock:
            if self._sock is not None:
                try:
                    self._sock.close()
                except OSError:
                    pass
                self._sock = None


def probe_tcp(host: str, port: int, timeout: float = 1.0) -> bool:
    try:
        with socket.create_connection((host, port), timeout=timeout):
            return True
    except OSError:
        return False
