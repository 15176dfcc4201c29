"""Bounded in-memory tables, one per (sampler, table) pair.

Each table is a deque with maxlen (drop-oldest) plus a monotonically
increasing append counter so the incremental sender can detect new rows in
O(1) and survive eviction (reference: database/database.py:7-34).
"""

from __future__ import annotations

import threading
from collections import deque
from typing import Dict, List, Optional

DEFAULT_TABLE_MAXLEN = 3000


class _Table:
    __slots__ = ("rows", "append_count", "maxlen")

    def __init__(self, maxlen: int) -> None:
        self.rows: deque = deque(maxlen=maxlen)
        self.append_count = 0
        self.maxlen = maxlen


class Database:
    def __init__(self, maxlen: int = DEFAULT_TABLE_MAXLEN) -> None:
        self._lock = threading.Lock()
        self._tables: Dict[str, _Table] = {}
        self._maxlen = maxlen

    def add_record(self, table: str, row: dict) -> None:
        with self._lock:
            t = self._tables.get(table)
            if t is None:
                t = _Table(self._maxlen)
                self._tables[table] = t
            t.rows.append(row)
            t.append_count += 1

    def add_records(self, table: str, rows: List[dict]) -> None:
        if not rows:
            return
        with self._lock:
            t = self._tables.get(table)
            if t is None:
                t = _Table(self._maxlen)
                self._tables[table] = t
            t.rows.extend(rows)
            t.append_count += len(rows)

    def table_names(self) -> List[str]:
        with self._lock:
            return list(self._tables)

    def append_count(self, table: str) -> int:
        with self._lock:
            t = self._tables.get(table)
            return 0 if t is None else t.append_count

    def rows_since(self, table: str, last_seen_count: int) -> tuple:
        """Return (new_rows, current_append_count). Rows evicted past the
        deque window are silently skipped (bounded memory beats completeness
        for live telemetry; the msgpack disk writer is the durable path)."""
        with self._lock:
            t = self._tables.get(table)
            if t is None:
                return ([], last_seen_count)
            count = t.append_count
            new = count - last_seen_count
            if new <= 0:
                return ([], count)
            available = len(t.rows)
            take = min(new, available)
            rows = list(t.rows)[-take:] if take else []
            return (rows, count)

    def tail(self, table: str, n: Optional[int] = None) -> List[dict]:
        with self._lock:
            t = self._tables.get(table)
            if t is None:
                return []
            rows = list(t.rows)
            return rows if n is None else rows[-n:]
