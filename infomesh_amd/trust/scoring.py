"""Trust scoring per node/domain.

Reference parity: infomesh/trust/scoring.py —
Trust = 0.15·uptime + 0.25·contribution + 0.40·audit_pass +
0.20·summary_quality (scoring.py:32-35); tiers trusted >= 0.8 / normal /
suspect / untrusted < 0.3 with isolation after 3 consecutive failures.
In the single-node deployment "peers" are GPU shard ranks and crawl
sources (domains); the domain trust signal feeds composite ranking.
"""
from __future__ import annotations

import time
from pathlib import Path

from ..db import SQLiteStore

W_UPTIME = 0.15
W_CONTRIBUTION = 0.25
W_AUDIT = 0.40
W_SUMMARY = 0.20

TIER_TRUSTED = 0.8
TIER_SUSPECT = 0.5
TIER_UNTRUSTED = 0.3
ISOLATION_FAILURES = 3


def tier_of(score: float, isolated: bool = False) -> str:
    if isolated:
        return "isolated"
    if score >= TIER_TRUSTED:
        return "trusted"
    if score >= TIER_SUSPECT:
        return "normal"
    if score >= TIER_UNTRUSTED:
        return "suspect"
    return "untrusted"


class TrustStore(SQLiteStore):
    SCHEMA = """
    CREATE TABLE IF NOT EXISTS trust (
        subject TEXT PRIMARY KEY,
        uptime REAL NOT NULL DEFAULT 0.5,
        contribution REAL NOT NULL DEFAULT 0.5,
        audit_pass REAL NOT NULL DEFAULT 0.5,
        summary_quality REAL NOT NULL DEFAULT 0.5,
        consecutive_failures INTEGER NOT NULL DEFAULT 0,
        isolated INTEGER NOT NULL DEFAULT 0,
        updated_at REAL NOT NULL
    );
    """

    def __init__(self, path: str | Path = ":memory:",
                 isolation_failures: int = ISOLATION_FAILURES):
        super().__init__(path)
        # config trust.isolation_failures (reference trust_system.py
        # isolates after 3 consecutive failed audits)
        self.isolation_failures = int(isolation_failures)

    def _get(self, subject: str):
        return self.execute(
            "SELECT * FROM trust WHERE subject=?", (subject,)).fetchone()

    def _ensure(self, subject: str) -> None:
        self.execute(
            "INSERT OR IGNORE INTO trust (subject, updated_at) VALUES (?,?)",
            (subject, time.time()))

    def update_component(self, subject: str, component: str,
                         value: float, ema: float = 0.3) -> None:
        assert component in ("uptime", "contribution", "audit_pass",
                             "summary_quality")
        self._ensure(subject)
        row = self._get(subject)
        old = float(row[component])
        new = (1 - ema) * old + ema * max(0.0, min(1.0, value))
        self.execute(
            f"UPDATE trust SET {component}=?, updated_at=? WHERE subject=?",
            (new, time.time(), subject))
        self.commit()

    def record_audit(self, subject: str, passed: bool) -> None:
        self._ensure(subject)
        row = self._get(subject)
        fails = 0 if passed else int(row["consecutive_failures"]) + 1
        isolated = 1 if fails >= self.isolation_failures else int(row["isolated"])
        if passed:
            isolated = 0
        self.execute(
            "UPDATE trust SET consecutive_failures=?, isolated=?,"
            " updated_at=? WHERE subject=?",
            (fails, isolated, time.time(), subject))
        self.commit()
        self.update_component(subject, "audit_pass", 1.0 if passed else 0.0)

    def isolate(self, subject: str) -> None:
        """Direct isolation (MaliciousNodeDetector verdicts: CRITICAL
        always isolates; HIGH isolates on repeat forgery) — bypasses
        the consecutive-failures ladder."""
        self._ensure(subject)
        self.execute(
            "UPDATE trust SET isolated=1, updated_at=? WHERE subject=?",
            (time.time(), subject))
        self.commit()

    def score(self, subject: str) -> float:
        row = self._get(subject)
        if row is None:
            return 0.5
        return (W_UPTIME * row["uptime"]
                + W_CONTRIBUTION * row["contribution"]
                + W_AUDIT * row["audit_pass"]
                + W_SUMMARY * row["summary_quality"])

    def tier(self, subject: str) -> str:
        row = self._get(subject)
        if row is None:
            return "normal"
        return tier_of(self.score(subject), bool(row["isolated"]))

    def trust_fn(self):
        """Closure for ranking (the trust_fn injected into
        rank_local_results)."""
        def fn(domain: str) -> float:
            return self.score(domain)
        return fn
