"""Generic single-table store on SQLite.

Replaces the reference's MySQL-backed repos (SqlDataBase in
ols_core/utils/repo_utils.py and the TaskTableRepo/ResTableRepo CRUD in
ols_core/taskMgr/utils/utils.py:29-267 / resourceMgr/utils/utils.py): same
operation surface (get/set item by an identify column, add/delete rows,
conditional queries), but embedded — one node needs no DB service.

Thread-safe: one connection per call site guarded by a lock; SQLite WAL
mode so the manager threads (schedule / release / interrupt) can interleave.
"""

from __future__ import annotations

import json
import os
import sqlite3
import threading
from typing import Any, Dict, List, Optional


class SqlTableRepo:
    """A single named table with declared columns.

    Columns are stored as TEXT/INTEGER/REAL; Python values that are dicts
    or lists are JSON-encoded transparently on write and decoded on read
    is left to the caller (the reference stores JSON strings explicitly,
    and we keep that behaviour for parity).
    """

    def __init__(self, path: str, table: str, columns: Dict[str, str],
                 primary_key: Optional[str] = None):
        self._path = path
        self._table = table
        self._columns = dict(columns)
        self._pk = primary_key
        self._lock = threading.RLock()
        if path != ":memory:":
            os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        self._conn = sqlite3.connect(path, check_same_thread=False)
        self._conn.execute("PRAGMA journal_mode=WAL")
        cols = ", ".join(
            f'"{name}" {ctype}' + (" PRIMARY KEY" if name == primary_key else "")
            for name, ctype in columns.items()
        )
        with self._lock:
            self._conn.execute(f'CREATE TABLE IF NOT EXISTS "{table}" ({cols})')
            # migrate pre-existing databases: a schema that gained
            # columns since the table was created must ALTER, or every
            # write to a new column fails ("no such column")
            have = {r[1] for r in self._conn.execute(
                f'PRAGMA table_info("{table}")')}
            for name, ctype in columns.items():
                if name not in have:
                    self._conn.execute(
                        f'ALTER TABLE "{table}" ADD COLUMN "{name}" {ctype}')
            self._conn.commit()

    # -- row ops ---------------------------------------------------------
    def add_item(self, item: Dict[str, Any]) -> bool:
        names = list(item.keys())
        ph = ", ".join("?" for _ in names)
        collist = ", ".join(f'"{n}"' for n in names)
        with self._lock:
            try:
                self._conn.execute(
                    f'INSERT INTO "{self._table}" ({collist}) VALUES ({ph})',
                    [item[n] for n in names])
            except sqlite3.IntegrityError:
                return False    # duplicate primary key
            self._conn.commit()
        return True

    def upsert_item(self, identify_name: str, item: Dict[str, Any]) -> bool:
        if self.has_item(identify_name, item[identify_name]):
            rest = {k: v for k, v in item.items() if k != identify_name}
            for k, v in rest.items():
                self.set_item_value(identify_name, item[identify_name], k, v)
            return True
        return self.add_item(item)

    def delete_item(self, identify_name: str, identify_value: Any) -> bool:
        with self._lock:
            self._conn.execute(
                f'DELETE FROM "{self._table}" WHERE "{identify_name}" = ?',
                (identify_value,))
            self._conn.commit()
        return True

    def has_item(self, identify_name: str, identify_value: Any) -> bool:
        with self._lock:
            cur = self._conn.execute(
                f'SELECT 1 FROM "{self._table}" WHERE "{identify_name}" = ? LIMIT 1',
                (identify_value,))
            return cur.fetchone() is not None

    # -- cell ops --------------------------------------------------------
    def get_item_value(self, identify_name: str, identify_value: Any,
                       item: str) -> Any:
        """Return one cell, or None when the row or the cell is NULL/absent."""
        with self._lock:
            cur = self._conn.execute(
                f'SELECT "{item}" FROM "{self._table}" WHERE "{identify_name}" = ?',
                (identify_value,))
            row = cur.fetchone()
        return None if row is None else row[0]

    def set_item_value(self, identify_name: str, identify_value: Any,
                       item: str, value: Any) -> bool:
        with self._lock:
            cur = self._conn.execute(
                f'UPDATE "{self._table}" SET "{item}" = ? WHERE "{identify_name}" = ?',
                (value, identify_value))
            self._conn.commit()
            return cur.rowcount > 0

    # -- queries ---------------------------------------------------------
    def get_all_rows(self) -> List[Dict[str, Any]]:
        with self._lock:
            cur = self._conn.execute(f'SELECT * FROM "{self._table}"')
            names = [d[0] for d in cur.description]
            return [dict(zip(names, r)) for r in cur.fetchall()]

    def get_rows_where(self, conditions: Dict[str, Any]) -> List[Dict[str, Any]]:
        keys = list(conditions.keys())
        clause = " AND ".join(f'"{k}" = ?' for k in keys)
        with self._lock:
            cur = self._conn.execute(
                f'SELECT * FROM "{self._table}" WHERE {clause}',
                [conditions[k] for k in keys])
            names = [d[0] for d in cur.description]
            return [dict(zip(names, r)) for r in cur.fetchall()]

    def get_column_not_none(self, column: str) -> List[Any]:
        """All non-NULL values of one column (reference: get_all_items_not_None)."""
        with self._lock:
            cur = self._conn.execute(
                f'SELECT "{column}" FROM "{self._table}" WHERE "{column}" IS NOT NULL')
            return [r[0] for r in cur.fetchall()]

    def count(self) -> int:
        with self._lock:
            cur = self._conn.execute(f'SELECT COUNT(*) FROM "{self._table}"')
            return int(cur.fetchone()[0])

    def clear(self) -> None:
        with self._lock:
            self._conn.execute(f'DELETE FROM "{self._table}"')
            self._conn.commit()

    def close(self) -> None:
        with self._lock:
            self._conn.close()


def json_dumps(obj: Any) -> str:
    return json.dumps(obj, ensure_ascii=False)


def json_loads(s: Optional[str], default: Any = None) -> Any:
    if s is None or s == "":
        return default
    return json.loads(s)
