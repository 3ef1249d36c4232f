"""SQLite base store.

Reference parity: infomesh/db.py:37-98 (SQLiteStore: WAL mode,
busy_timeout, schema-exec on init). Base class for the ledger, trust
store, feedback store, dedup DB, link graph, GDPR/DMCA stores.
"""
from __future__ import annotations

import sqlite3
import threading
from pathlib import Path


class SQLiteStore:
    """Thread-safe-ish SQLite wrapper: WAL, busy timeout, schema on init.

    Connections are per-instance; concurrent multi-process readers are
    supported through WAL (the reference's dashboard/API read pattern)."""

    SCHEMA: str = ""

    def __init__(self, path: str | Path = ":memory:"):
        self.path = str(path)
        if self.path != ":memory:":
            Path(self.path).parent.mkdir(parents=True, exist_ok=True)
        self._lock = threading.RLock()
        self.conn = sqlite3.connect(self.path, check_same_thread=False)
        self.conn.row_factory = sqlite3.Row
        cur = self.conn.cursor()
        cur.execute("PRAGMA busy_timeout=5000")
        if self.path != ":memory:":
            cur.execute("PRAGMA journal_mode=WAL")
        cur.execute("PRAGMA synchronous=NORMAL")
        if self.SCHEMA:
            self.conn.executescript(self.SCHEMA)
            self.conn.commit()

    def execute(self, sql: str, params: tuple = ()) -> sqlite3.Cursor:
        with self._lock:
            return self.conn.execute(sql, params)

    def executemany(self, sql: str, rows) -> sqlite3.Cursor:
        with self._lock:
            return self.conn.executemany(sql, rows)

    def commit(self) -> None:
        with self._lock:
            self.conn.commit()

    def close(self) -> None:
        with self._lock:
            try:
                self.conn.commit()
            except sqlite3.Error:
                pass
            self.conn.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
        return False
