"""Async durable SQLite history, bounded by per-identity retention pruning.

One writer thread owns the connection: WAL journal + synchronous=NORMAL,
ingest queue (50k, drop-oldest), flush every 0.5 s or 20k rows, periodic
retention prune keeping the newest N rows per (table, global_rank) via a
ROW_NUMBER window, ``force_flush`` barriers for read-after-write surfaces,
and ``finalize`` = drain → final prune → ``wal_checkpoint(TRUNCATE)`` →
close (reference: aggregator/sqlite_writer.py:116-651).
"""

from __future__ import annotations

import logging
import os
import sqlite3
import threading
import time
from collections import deque
from typing import List, Optional

from traceml_amd.aggregator.writers import build_all_writers
from traceml_amd.reporting.config import RETENTION_ROWS_PER_IDENTITY

logger = logging.getLogger(__name__)

QUEUE_MAX = 50_000
FLUSH_INTERVAL_SEC = 0.5
FLUSH_ROW_THRESHOLD = 20_000
PRUNE_INTERVAL_SEC = 30.0


class _FlushBarrier:
    def __init__(self) -> None:
        self.event = threading.Event()


class SQLiteWriterSimple:
    def __init__(self, db_path: str) -> None:
        self.db_path = db_path
        self._writers = build_all_writers()
        self._queue: deque = deque()
        self._queue_lock = threading.Lock()
        self._wake = threading.Event()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._conn: Optional[sqlite3.Connection] = None
        self._last_prune = 0.0
        self._conn_lock = threading.Lock()
        self.dropped = 0
        self._prunable_tables: List[str] = []

    # -- lifecycle ----------------------------------------------------------

    def start(self) -> None:
        os.makedirs(os.path.dirname(os.path.abspath(self.db_path)), exist_ok=True)
        self._thread = threading.Thread(
            target=self._loop, name="traceml-sqlite", daemon=True
        )
        self._thread.start()

    def _open(self) -> None:
        # finalize()/_prune() may run on the caller's thread after the writer
        # thread stops; every connection use is serialized by _conn_lock.
        conn = sqlite3.connect(self.db_path, check_same_thread=False)
        conn.execute("PRAGMA journal_mode=WAL")
        conn.execute("PRAGMA synchronous=NORMAL")
        for writer in self._writers:
            writer.init_schema(conn)
            self._prunable_tables.extend(writer.sql_tables())
        conn.commit()
        self._conn = conn

    # -- ingest -------------------------------------------------------------

    def ingest(self, envelope: dict) -> None:
        with self._queue_lock:
            while len(self._queue) >= QUEUE_MAX:
                evicted = self._queue.popleft()
                if isinstance(evicted, _FlushBarrier):
                    # Never strand a force_flush() waiter on its full timeout:
                    # release it immediately (durability is degraded under
                    # overload anyway — rows ahead of it were just dropped).
                    evicted.event.set()
                else:
                    self.dropped += 1
            self._queue.append(envelope)
        self._wake.set()

    def force_flush(self, timeout: float = 5.0) -> bool:
        """Barrier: returns once everything queued before the call is durable."""
        barrier = _FlushBarrier()
        with self._queue_lock:
            self._queue.append(barrier)
        self._wake.set()
        return barrier.event.wait(timeout)

    # -- writer thread ------------------------------------------------------

    def _loop(self) -> None:
        try:
            self._open()
        except Exception:
            logger.exception("traceml_amd: sqlite open failed")
            return
        while not self._stop.is_set():
            self._wake.wait(timeout=FLUSH_INTERVAL_SEC)
            self._wake.clear()
            try:
                self._flush_once()
            except Exception:
                logger.debug("traceml_amd: sqlite flush failed", exc_info=True)
            now = time.time()
            if now - self._last_prune > PRUNE_INTERVAL_SEC:
                self._last_prune = now
                try:
                    self._prune()
                except Exception:
                    logger.debug("traceml_amd: sqlite prune failed", exc_info=True)
        # final drain happens in finalize()

    def _take_batch(self) -> List:
        with self._queue_lock:
            items = list(self._queue)
            self._queue.clear()
        return items

    def _flush_once(self) -> int:
        items = self._take_batch()
        if not items:
            return 0
        conn = self._conn
        inserted = 0
        barriers: List[_FlushBarrier] = []
        by_table: dict = {}
        for item in items:
            if isinstance(item, _FlushBarrier):
                barriers.append(item)
                continue
            sampler = item.get("meta", {}).get("sampler")
            for writer in self._writers:
                if not writer.accepts_sampler(sampler):
                    continue
                for sql_table, row in writer.build_rows(item):
                    by_table.setdefault((writer, sql_table), []).append(row)
        if by_table:
            with self._conn_lock:
                with conn:
                    for (writer, sql_table), rows in by_table.items():
                        writer.insert_rows(conn, sql_table, rows)
                        inserted += len(rows)
        for barrier in barriers:
            barrier.event.set()
        return inserted

    def _prune(self) -> None:
        conn = self._conn
        with self._conn_lock, conn:
            for table in self._prunable_tables:
                conn.execute(
                    f"""
                    DELETE FROM {table} WHERE id IN (
                      SELECT id FROM (
                        SELECT id, ROW_NUMBER() OVER (
                          PARTITION BY global_rank ORDER BY id DESC
                        ) AS rn FROM {table}
                      ) WHERE rn > ?
                    )
                    """,
                    (RETENTION_ROWS_PER_IDENTITY,),
                )

    # -- finalize -----------------------------------------------------------

    def finalize(self, budget_sec: float = 10.0) -> None:
        deadline = time.time() + budget_sec
        self._stop.set()
        self._wake.set()
        if self._thread is not None:
            self._thread.join(timeout=max(0.5, deadline - time.time()))
        if self._conn is None:
            return
        try:
            while True:
                if self._flush_once() == 0:
                    with self._queue_lock:
                        if not self._queue:
                            break
                if time.time() > deadline:
                    break
            self._prune()
            with self._conn_lock:
                self._conn.execute("PRAGMA wal_checkpoint(TRUNCATE)")
                self._conn.commit()
        except Exception:
            logger.debug("traceml_amd: sqlite finalize failed", exc_info=True)
        finally:
            try:
                self._conn.close()
            except Exception:
                pass
            self._conn = None
