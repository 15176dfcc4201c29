"""Per-rank msgpack disk backup: length-prefixed rows, one file per table.

Layout: ``<logs>/<session>/r<rank>/data/<sampler>/<table>.msgpack``
(read back by ``traceml-amd inspect``; reference: database/database_writer.py:27-143).
Writes are throttled (flush every N rows) and best-effort.
"""

from __future__ import annotations

import os
import struct
import threading
from typing import Dict, Optional

from traceml_amd.database.database import Database
from traceml_amd.transport import codec

_LEN = struct.Struct(">I")
FLUSH_EVERY = 100


class DatabaseWriter:
    def __init__(self, sampler_name: str, database: Database, data_dir: str) -> None:
        self.sampler_name = sampler_name
        self.database = database
        self.data_dir = os.path.join(data_dir, sampler_name)
        self._cursors: Dict[str, int] = {}
        self._files: Dict[str, object] = {}
        self._pending = 0
        self._lock = threading.Lock()

    def _file_for(self, table: str):
        f = self._files.get(table)
        if f is None:
            os.makedirs(self.data_dir, exist_ok=True)
            f = open(os.path.join(self.data_dir, f"{table}.msgpack"), "ab")
            self._files[table] = f
        return f

    def flush(self) -> None:
        with self._lock:
            try:
                for table in self.database.table_names():
                    last = self._cursors.get(table, 0)
                    rows, count = self.database.rows_since(table, last)
                    self._cursors[table] = count
                    if not rows:
                        continue
                    f = self._file_for(table)
                    for row in rows:
                        blob = codec.encode(row)
                        f.write(_LEN.pack(len(blob)))
                        f.write(blob)
                    self._pending += len(rows)
                if self._pending >= FLUSH_EVERY:
                    for f in self._files.values():
                        f.flush()
                    self._pending = 0
            except Exception:
                pass

    def close(self) -> None:
        with self._lock:
            for f in self._files.values():
                try:
                    f.flush()
                    f.close()
                except Exception:
                    pass
            self._files.clear()


def read_msgpack_table(path: str) -> list:
    """Inspect helper: decode a length-prefixed msgpack table file."""
    rows = []
    with open(path, "rb") as f:
        data = f.read()
    offset = 0
    while offset + _LEN.size <= len(data):
        (length,) = _LEN.unpack_from(data, offset)
        offset += _LEN.size
        if offset + length > len(data):
            break
        rows.append(codec.decode(data[offset : offset + length]))
        offset += length
    return rows
