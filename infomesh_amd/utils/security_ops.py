"""Security operations: API-key manager + audit logger.

Reference parity: infomesh/security_ops.py (API-key manager, audit
logger) and parts of security_ext.py (key hashing, constant-time
comparison).
"""
from __future__ import annotations

import hmac
import json
import secrets
import time
from pathlib import Path

from ..db import SQLiteStore
from ..hashing import content_hash


class ApiKeyManager(SQLiteStore):
    SCHEMA = """
    CREATE TABLE IF NOT EXISTS api_keys (
        key_hash TEXT PRIMARY KEY,
        name TEXT NOT NULL,
        created_at REAL NOT NULL,
        revoked INTEGER NOT NULL DEFAULT 0,
        last_used REAL
    );
    """

    def create_key(self, name: str) -> str:
        """Returns the plaintext key ONCE; only its hash is stored."""
        key = "imk_" + secrets.token_urlsafe(32)
        self.execute("INSERT INTO api_keys (key_hash, name, created_at)"
                     " VALUES (?,?,?)",
                     (content_hash(key), name, time.time()))
        self.commit()
        return key

    def verify(self, key: str) -> bool:
        if not key:
            return False
        h = content_hash(key)
        row = self.execute(
            "SELECT key_hash, revoked FROM api_keys WHERE key_hash=?",
            (h,)).fetchone()
        if row is None or row["revoked"]:
            return False
        # constant-time double check (hash lookup already constant-ish)
        if not hmac.compare_digest(row["key_hash"], h):
            return False
        self.execute("UPDATE api_keys SET last_used=? WHERE key_hash=?",
                     (time.time(), h))
        self.commit()
        return True

    def revoke(self, name: str) -> int:
        cur = self.execute("UPDATE api_keys SET revoked=1 WHERE name=?",
                           (name,))
        self.commit()
        return cur.rowcount

    def list_keys(self) -> list[dict]:
        return [dict(r) for r in self.execute(
            "SELECT name, created_at, revoked, last_used FROM api_keys")]


class AuditLogger:
    """Append-only JSONL audit log with hash chaining."""

    def __init__(self, path: str | Path):
        self.path = Path(path)
        self.path.parent.mkdir(parents=True, exist_ok=True)
        self._prev = "genesis"
        if self.path.exists():
            try:
                last = self.path.read_text().strip().rsplit("\n", 1)[-1]
                self._prev = json.loads(last).get("hash", "genesis")
            except (json.JSONDecodeError, OSError):
                pass

    def log(self, event: str, **fields) -> dict:
        entry = {"ts": time.time(), "event": event, "prev": self._prev,
                 **fields}
        entry["hash"] = content_hash(json.dumps(entry, sort_keys=True))
        with open(self.path, "a") as f:
            f.write(json.dumps(entry) + "\n")
        self._prev = entry["hash"]
        return entry

    def verify(self) -> bool:
        prev = "genesis"
        try:
            for line in self.path.read_text().splitlines():
                entry = json.loads(line)
                h = entry.pop("hash")
                if entry.get("prev") != prev:
                    return False
                if content_hash(json.dumps(entry, sort_keys=True)) != h:
                    return False
                prev = h
        except (OSError, json.JSONDecodeError, KeyError):
            return False
        return True
