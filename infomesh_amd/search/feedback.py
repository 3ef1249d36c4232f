"""Implicit search feedback -> URL ranking boosts.

Reference parity: infomesh/search/feedback.py (fetch/skip/cite signals
into a SQLite FeedbackStore, aggregated into per-URL boosts).
"""
from __future__ import annotations

import math
import time
from pathlib import Path

from ..db import SQLiteStore

SIGNAL_WEIGHTS = {"fetch": 1.0, "cite": 2.0, "click": 0.5, "skip": -0.5}
BOOST_CAP = 0.2  # max additive ranking boost


class FeedbackStore(SQLiteStore):
    SCHEMA = """
    CREATE TABLE IF NOT EXISTS feedback (
        id INTEGER PRIMARY KEY,
        url TEXT NOT NULL,
        query TEXT NOT NULL DEFAULT '',
        signal TEXT NOT NULL,
        ts REAL NOT NULL
    );
    CREATE INDEX IF NOT EXISTS idx_feedback_url ON feedback(url);
    """

    def __init__(self, path: str | Path = ":memory:"):
        super().__init__(path)
        self._boost_cache: dict[str, tuple[float, float]] = {}

    def record(self, url: str, signal: str, query: str = "") -> None:
        if signal not in SIGNAL_WEIGHTS:
            raise ValueError(f"unknown signal {signal!r}")
        self.execute(
            "INSERT INTO feedback (url, query, signal, ts) VALUES (?,?,?,?)",
            (url, query, signal, time.time()))
        self.commit()
        self._boost_cache.pop(url, None)
        self._has_cache = True

    def has_signals(self) -> bool:
        """Cheap guard for hot paths: False until anything is recorded
        (cached; invalidated by record())."""
        cached = getattr(self, "_has_cache", None)
        if cached is None:
            row = self.execute(
                "SELECT EXISTS(SELECT 1 FROM feedback) AS e").fetchone()
            cached = self._has_cache = bool(row["e"])
        return cached

    def url_boost(self, url: str) -> float:
        """Saturating additive boost in [-CAP, +CAP]."""
        cached = self._boost_cache.get(url)
        if cached and time.time() - cached[1] < 60:
            return cached[0]
        raw = 0.0
        for row in self.execute(
                "SELECT signal, COUNT(*) AS c FROM feedback WHERE url=?"
                " GROUP BY signal", (url,)):
            raw += SIGNAL_WEIGHTS[row["signal"]] * row["c"]
        boost = BOOST_CAP * math.tanh(raw / 10.0)
        self._boost_cache[url] = (boost, time.time())
        return boost

    def top_urls(self, limit: int = 10) -> list[tuple[str, float]]:
        urls = [r["url"] for r in self.execute(
            "SELECT DISTINCT url FROM feedback").fetchall()]
        scored = [(u, self.url_boost(u)) for u in urls]
        scored.sort(key=lambda p: -p[1])
        return scored[:limit]

    def stats(self) -> dict:
        out = {}
        for row in self.execute(
                "SELECT signal, COUNT(*) AS c FROM feedback GROUP BY signal"):
            out[row["signal"]] = row["c"]
        return out
