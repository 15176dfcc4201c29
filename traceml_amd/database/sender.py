"""Incremental sender: ships only rows appended since the last flush.

One cursor per table keyed on the Database's monotonic append counter, so
eviction from the bounded deque cannot desync the stream (reference:
database/database_sender.py:33-188).
"""

from __future__ import annotations

from typing import Dict, Optional

from traceml_amd.database.database import Database
from traceml_amd.telemetry.envelope import build_telemetry_envelope


class DBIncrementalSender:
    def __init__(self, sampler_name: str, database: Database) -> None:
        self.sampler_name = sampler_name
        self.database = database
        self._cursors: Dict[str, int] = {}

    def collect_payload(self, identity_meta: dict) -> Optional[dict]:
        tables = {}
        for table in self.database.table_names():
            last = self._cursors.get(table, 0)
            rows, count = self.database.rows_since(table, last)
            self._cursors[table] = count
            if rows:
                tables[table] = rows
        if not tables:
            return None
        return build_telemetry_envelope(identity_meta, self.sampler_name, tables)
