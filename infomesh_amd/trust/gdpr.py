"""GDPR right-to-be-forgotten: durable deletion records that survive
re-crawls and snapshot imports.

Reference parity: infomesh/trust/gdpr.py (DeletionManager + distributed
deletion records; single-node build keeps the signed record log and
enforces it at index time and on snapshot import).
"""
from __future__ import annotations

import json
import time
from pathlib import Path

from ..db import SQLiteStore
from ..hashing import content_hash
from ..index.local_store import LocalStore, extract_domain
from .keys import KeyPair


class DeletionManager(SQLiteStore):
    SCHEMA = """
    CREATE TABLE IF NOT EXISTS deletions (
        id INTEGER PRIMARY KEY,
        subject TEXT NOT NULL,          -- url | domain:x | text_hash:h
        reason TEXT NOT NULL DEFAULT '',
        requester TEXT NOT NULL DEFAULT '',
        ts REAL NOT NULL,
        record TEXT NOT NULL DEFAULT '',
        applied_count INTEGER NOT NULL DEFAULT 0
    );
    CREATE UNIQUE INDEX IF NOT EXISTS idx_del_subject ON deletions(subject);
    """

    def __init__(self, store: LocalStore, kp: KeyPair | None = None,
                 path: str | Path = ":memory:"):
        super().__init__(path)
        self.store = store
        self.kp = kp

    def request_deletion(self, subject: str, reason: str = "",
                         requester: str = "") -> dict:
        ts = time.time()
        payload = {"subject": subject, "reason": reason,
                   "requester_hash": content_hash(requester)[:16], "ts": ts}
        record = dict(payload)
        if self.kp is not None:
            blob = json.dumps(payload, sort_keys=True).encode()
            record["node_id"] = self.kp.node_id
            record["public_key"] = self.kp.public.hex()
            record["signature"] = self.kp.sign(blob).hex()
        self.execute(
            "INSERT OR REPLACE INTO deletions"
            " (subject, reason, requester, ts, record) VALUES (?,?,?,?,?)",
            (subject, reason, content_hash(requester)[:16], ts,
             json.dumps(record)))
        self.commit()
        self._subj_cache = None
        self.enforce()
        return record

    def enforce(self) -> int:
        """Apply all deletion records against the store."""
        removed = 0
        for row in self.execute("SELECT id, subject FROM deletions").fetchall():
            s = row["subject"]
            if s.startswith("domain:"):
                n = self.store.delete_by_domain(s[7:])
            elif s.startswith("text_hash:"):
                cur = self.store.conn.execute(
                    "DELETE FROM documents WHERE text_hash=?", (s[10:],))
                self.store.conn.commit()
                n = cur.rowcount
            else:
                n = 1 if self.store.delete_by_url(s) else 0
            if n:
                self.execute(
                    "UPDATE deletions SET applied_count=applied_count+?"
                    " WHERE id=?", (n, row["id"]))
                removed += n
        self.commit()
        return removed

    def is_forgotten(self, url: str) -> bool:
        """Index-time guard: refuses re-indexing of deleted subjects.
        Runs on every index_document call — subjects cached in memory
        (deletion records change rarely)."""
        exact, domains = self._subjects()
        if not exact and not domains:
            return False
        if url in exact:
            return True
        return bool(domains) and extract_domain(url) in domains

    def _subjects(self):
        cached = getattr(self, "_subj_cache", None)
        if cached is None:
            exact, domains = set(), set()
            for row in self.execute("SELECT subject FROM deletions"):
                s = row["subject"]
                if s.startswith("domain:"):
                    domains.add(s[7:])
                else:
                    exact.add(s)
            cached = self._subj_cache = (exact, domains)
        return cached

    def export_records(self) -> list[dict]:
        return [json.loads(r["record"]) for r in
                self.execute("SELECT record FROM deletions").fetchall()
                if r["record"]]
