"""Cross-node credit aggregation for same-owner nodes.

Reference parity: infomesh/credits/sync.py (SHA-256 email-hash owner
announce, signed CreditSummary exchange, CreditSyncStore). Transport
here is artifact-based: nodes export signed summaries (files / the
admin API) and import peers' summaries — the single-node deployment's
equivalent of the reference's libp2p credit-sync rounds.
"""
from __future__ import annotations

import json
import time
from dataclasses import dataclass, asdict
from pathlib import Path

from ..db import SQLiteStore
from ..hashing import content_hash
from ..trust.keys import KeyPair
from .ledger import CreditLedger


def owner_hash(email: str) -> str:
    """Privacy-preserving owner identity (email never leaves the node)."""
    return content_hash(email.strip().lower())[:32]


@dataclass
class CreditSummary:
    node_id: str
    owner: str          # owner_hash
    balance: float
    n_entries: int
    ts: float
    public_key: str = ""
    signature: str = ""

    def payload(self) -> bytes:
        return json.dumps({
            "node_id": self.node_id, "owner": self.owner,
            "balance": round(self.balance, 6),
            "n_entries": self.n_entries, "ts": self.ts},
            sort_keys=True).encode()

    def to_dict(self) -> dict:
        return asdict(self)

    @classmethod
    def from_dict(cls, d: dict) -> "CreditSummary":
        return cls(**{k: d[k] for k in ("node_id", "owner", "balance",
                                        "n_entries", "ts", "public_key",
                                        "signature")})


def build_summary(ledger: CreditLedger, kp: KeyPair,
                  owner_email: str) -> CreditSummary:
    s = CreditSummary(node_id=kp.node_id, owner=owner_hash(owner_email),
                      balance=ledger.balance(),
                      n_entries=len(ledger.entries(10 ** 9)),
                      ts=time.time(), public_key=kp.public.hex())
    s.signature = kp.sign(s.payload()).hex()
    return s


def verify_summary(s: CreditSummary, max_age_s: float = 7 * 86400.0) -> bool:
    if time.time() - s.ts > max_age_s:
        return False
    try:
        pub = bytes.fromhex(s.public_key)
        if content_hash(pub)[:32] != s.node_id:
            return False
        return KeyPair.verify(pub, s.payload(),
                              bytes.fromhex(s.signature))
    except ValueError:
        return False


class CreditSyncStore(SQLiteStore):
    SCHEMA = """
    CREATE TABLE IF NOT EXISTS peer_summaries (
        node_id TEXT PRIMARY KEY,
        owner TEXT NOT NULL,
        balance REAL NOT NULL,
        n_entries INTEGER NOT NULL,
        ts REAL NOT NULL,
        record TEXT NOT NULL
    );
    CREATE INDEX IF NOT EXISTS idx_sync_owner ON peer_summaries(owner);
    """

    def ingest(self, summary: CreditSummary) -> bool:
        if not verify_summary(summary):
            return False
        row = self.execute("SELECT ts FROM peer_summaries WHERE node_id=?",
                           (summary.node_id,)).fetchone()
        if row is not None and row["ts"] >= summary.ts:
            return False  # stale
        self.execute(
            "INSERT OR REPLACE INTO peer_summaries VALUES (?,?,?,?,?,?)",
            (summary.node_id, summary.owner, summary.balance,
             summary.n_entries, summary.ts,
             json.dumps(summary.to_dict())))
        self.commit()
        return True

    def owner_total(self, owner: str, local_balance: float = 0.0) -> float:
        row = self.execute(
            "SELECT COALESCE(SUM(balance),0) b FROM peer_summaries"
            " WHERE owner=?", (owner,)).fetchone()
        return float(row["b"]) + local_balance

    def export_dir(self, path: Path) -> int:
        path.mkdir(parents=True, exist_ok=True)
        n = 0
        for row in self.execute("SELECT node_id, record FROM peer_summaries"):
            (path / f"{row['node_id']}.json").write_text(row["record"])
            n += 1
        return n

    def import_dir(self, path: Path) -> int:
        n = 0
        for f in Path(path).glob("*.json"):
            try:
                s = CreditSummary.from_dict(json.loads(f.read_text()))
            except (json.JSONDecodeError, KeyError):
                continue
            if self.ingest(s):
                n += 1
        return n
