"""Projection-writer base: declarative table specs, identity columns,
additive schema migration, batch inserts."""

from __future__ import annotations

import json
import sqlite3
from typing import Dict, List, Optional, Sequence, Tuple

#: identity columns stamped on every projected row (from envelope meta)
IDENTITY_COLUMNS: List[Tuple[str, str]] = [
    ("global_rank", "INTEGER"),
    ("local_rank", "INTEGER"),
    ("world_size", "INTEGER"),
    ("local_world_size", "INTEGER"),
    ("node_rank", "INTEGER"),
    ("hostname", "TEXT"),
    ("pid", "INTEGER"),
]

IDENTITY_KEYS = [c for c, _ in IDENTITY_COLUMNS]


class ProjectionWriter:
    """One per sampler. Subclasses declare:

    * ``sampler``: envelope sampler name this writer accepts
    * ``tables``: {wire_table_name: (sql_table_name, [(col, sqltype), ...])}
    * optionally ``json_columns``: {sql_table: {col: wire_key}} for fields
      serialized as restricted JSON text.
    """

    sampler: str = ""
    tables: Dict[str, Tuple[str, List[Tuple[str, str]]]] = {}
    json_columns: Dict[str, Dict[str, str]] = {}

    def accepts_sampler(self, sampler: str) -> bool:
        return sampler == self.sampler

    def init_schema(self, conn: sqlite3.Connection) -> None:
        for sql_table, columns in self.tables.values():
            cols = ", ".join(
                [f"{name} {sqltype}" for name, sqltype in IDENTITY_COLUMNS + columns]
            )
            conn.execute(
                f"CREATE TABLE IF NOT EXISTS {sql_table} "
                f"(id INTEGER PRIMARY KEY AUTOINCREMENT, {cols})"
            )
            for name, sqltype in IDENTITY_COLUMNS + columns:
                self._ensure_column(conn, sql_table, name, sqltype)
            conn.execute(
                f"CREATE INDEX IF NOT EXISTS idx_{sql_table}_identity "
                f"ON {sql_table} (global_rank, id)"
            )

    @staticmethod
    def _ensure_column(
        conn: sqlite3.Connection, table: str, column: str, sqltype: str
    ) -> None:
        existing = {
            row[1] for row in conn.execute(f"PRAGMA table_info({table})")
        }
        if column not in existing:
            conn.execute(f"ALTER TABLE {table} ADD COLUMN {column} {sqltype}")

    def build_rows(self, envelope: dict) -> List[Tuple[str, dict]]:
        """envelope → [(sql_table, row_dict)]."""
        meta = envelope.get("meta", {})
        # identity values go through the same sanitizer as row values: a
        # malformed meta (dict-valued rank) must not poison a whole flush
        # batch at executemany time
        identity = {k: _plain(meta.get(k)) for k in IDENTITY_KEYS}
        out: List[Tuple[str, dict]] = []
        for wire_table, rows in envelope.get("body", {}).get("tables", {}).items():
            spec = self.tables.get(wire_table)
            if spec is None:
                continue
            sql_table, columns = spec
            json_cols = self.json_columns.get(sql_table, {})
            for row in rows:
                projected = dict(identity)
                for col, _sqltype in columns:
                    if col in json_cols:
                        projected[col] = _encode_json(row.get(json_cols[col]))
                    else:
                        projected[col] = _plain(row.get(col))
                out.append((sql_table, projected))
        return out

    def insert_rows(
        self, conn: sqlite3.Connection, sql_table: str, rows: Sequence[dict]
    ) -> None:
        if not rows:
            return
        columns = list(rows[0].keys())
        placeholders = ", ".join(["?"] * len(columns))
        sql = (
            f"INSERT INTO {sql_table} ({', '.join(columns)}) VALUES ({placeholders})"
        )
        conn.execute("SAVEPOINT traceml_ins")
        try:
            conn.executemany(sql, [[r.get(c) for c in columns] for r in rows])
        except sqlite3.Error:
            # one unbindable row must not drop the whole flush batch: undo
            # the partial executemany, then retry row-by-row and skip only
            # the poisoned ones
            conn.execute("ROLLBACK TO traceml_ins")
            for r in rows:
                try:
                    conn.execute(sql, [r.get(c) for c in columns])
                except sqlite3.Error:
                    continue
        finally:
            conn.execute("RELEASE traceml_ins")

    def sql_tables(self) -> List[str]:
        return [sql_table for sql_table, _ in self.tables.values()]


def _plain(value):
    if isinstance(value, (dict, list)):
        return _encode_json(value)
    return value


def _encode_json(value) -> Optional[str]:
    if value is None:
        return None
    try:
        return json.dumps(value, default=str)
    except (TypeError, ValueError):
        return None
