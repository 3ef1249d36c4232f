"""DMCA takedown handling.

Reference parity: infomesh/trust/dmca.py (signed takedown notices with
24 h compliance, TakedownManager + SQLite store; DHT propagation in the
reference becomes a local signed record log in the single-node build —
notices still verify so exported snapshots can carry them).
"""
from __future__ import annotations

import json
import time
from dataclasses import dataclass, asdict
from pathlib import Path

from ..db import SQLiteStore
from ..index.local_store import LocalStore
from .keys import KeyPair

COMPLIANCE_WINDOW_S = 24 * 3600.0


@dataclass
class TakedownNotice:
    url_pattern: str          # exact URL or domain:example.com
    reason: str
    claimant: str
    ts: float
    node_id: str = ""
    public_key: str = ""
    signature: str = ""

    def payload(self) -> bytes:
        return json.dumps({
            "url_pattern": self.url_pattern, "reason": self.reason,
            "claimant": self.claimant, "ts": self.ts,
            "node_id": self.node_id}, sort_keys=True).encode()

    def to_dict(self) -> dict:
        return asdict(self)


class TakedownManager(SQLiteStore):
    SCHEMA = """
    CREATE TABLE IF NOT EXISTS takedowns (
        id INTEGER PRIMARY KEY,
        url_pattern TEXT NOT NULL,
        reason TEXT NOT NULL,
        claimant TEXT NOT NULL,
        ts REAL NOT NULL,
        record TEXT NOT NULL,
        applied_at REAL,
        removed_count INTEGER NOT NULL DEFAULT 0
    );
    """

    def __init__(self, store: LocalStore, kp: KeyPair | None = None,
                 path: str | Path = ":memory:"):
        super().__init__(path)
        self.store = store
        self.kp = kp

    def file_notice(self, url_pattern: str, reason: str,
                    claimant: str) -> TakedownNotice:
        notice = TakedownNotice(url_pattern=url_pattern, reason=reason,
                                claimant=claimant, ts=time.time())
        if self.kp is not None:
            notice.node_id = self.kp.node_id
            notice.public_key = self.kp.public.hex()
            notice.signature = self.kp.sign(notice.payload()).hex()
        self.execute(
            "INSERT INTO takedowns (url_pattern, reason, claimant, ts,"
            " record) VALUES (?,?,?,?,?)",
            (url_pattern, reason, claimant, notice.ts,
             json.dumps(notice.to_dict())))
        self.commit()
        self._pat_cache = None
        return notice

    @staticmethod
    def verify_notice(notice: TakedownNotice) -> bool:
        if not notice.signature:
            return False
        try:
            return KeyPair.verify(bytes.fromhex(notice.public_key),
                                  notice.payload(),
                                  bytes.fromhex(notice.signature))
        except ValueError:
            return False

    def apply_pending(self) -> int:
        """Remove matching docs for unapplied notices; returns count."""
        removed_total = 0
        for row in self.execute(
                "SELECT id, url_pattern FROM takedowns"
                " WHERE applied_at IS NULL").fetchall():
            pattern = row["url_pattern"]
            if pattern.startswith("domain:"):
                removed = self.store.delete_by_domain(pattern[7:])
            else:
                removed = 1 if self.store.delete_by_url(pattern) else 0
            self.execute(
                "UPDATE takedowns SET applied_at=?, removed_count=?"
                " WHERE id=?", (time.time(), removed, row["id"]))
            removed_total += removed
        self.commit()
        return removed_total

    def overdue(self, now: float | None = None) -> list[dict]:
        """Notices past the 24 h compliance window and still unapplied."""
        now = now or time.time()
        return [dict(r) for r in self.execute(
            "SELECT * FROM takedowns WHERE applied_at IS NULL AND ts < ?",
            (now - COMPLIANCE_WINDOW_S,)).fetchall()]

    def is_blocked(self, url: str) -> bool:
        """Would this URL be rejected at (re)index time? Runs on every
        index_document call — patterns are cached in memory (takedowns
        change rarely) instead of a table scan per page."""
        exact, domains = self._patterns()
        if not exact and not domains:
            return False
        if url in exact:
            return True
        if domains:
            from ..index.local_store import extract_domain
            return extract_domain(url) in domains
        return False

    def _patterns(self):
        cached = getattr(self, "_pat_cache", None)
        if cached is None:
            exact, domains = set(), set()
            for row in self.execute("SELECT url_pattern FROM takedowns"):
                p = row["url_pattern"]
                if p.startswith("domain:"):
                    domains.add(p[7:])
                else:
                    exact.add(p)
            cached = self._pat_cache = (exact, domains)
        return cached
