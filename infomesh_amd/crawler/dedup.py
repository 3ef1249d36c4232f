"""3-layer crawl dedup: URL normalization, exact content hash, SimHash.

Reference parity: infomesh/crawler/dedup.py (URL normalization rules —
lowercase host, strip tracking params, sort query, trailing slash;
SQLite seen_urls table; SimHashIndex reloaded at startup).
"""
from __future__ import annotations

import time
from pathlib import Path
from urllib.parse import parse_qsl, urlencode, urlparse

from ..db import SQLiteStore
from ..hashing import content_hash
from .simhash import SimHashIndex, simhash

TRACKING_PARAMS = frozenset(
    "utm_source utm_medium utm_campaign utm_term utm_content gclid fbclid "
    "mc_cid mc_eid ref_src igshid".split())


def normalize_url(url: str) -> str:
    # urlsplit (NOT urlparse): urlparse splits ';params' out of the
    # last path segment, which made normalization non-idempotent for
    # paths containing ';' (found by the hypothesis suite).
    from urllib.parse import urlsplit, urlunsplit
    try:
        p = urlsplit(url.strip())
    except ValueError:
        return url
    host = (p.hostname or "").lower()
    if p.port and not ((p.scheme == "http" and p.port == 80)
                       or (p.scheme == "https" and p.port == 443)):
        host = f"{host}:{p.port}"
    query = urlencode(sorted(
        (k, v) for k, v in parse_qsl(p.query, keep_blank_values=True)
        if k.lower() not in TRACKING_PARAMS))
    path = p.path or "/"
    if path != "/":
        path = path.rstrip("/") or "/"
    return urlunsplit((p.scheme.lower(), host, path, query, ""))


class DeduplicatorDB(SQLiteStore):
    """Durable crawl-dedup store; SimHash fingerprints reload on boot."""

    SCHEMA = """
    CREATE TABLE IF NOT EXISTS seen_urls (
        url_hash TEXT PRIMARY KEY,
        url TEXT NOT NULL,
        content_hash TEXT NOT NULL DEFAULT '',
        simhash INTEGER NOT NULL DEFAULT 0,
        crawled_at REAL NOT NULL
    );
    CREATE INDEX IF NOT EXISTS idx_seen_content ON seen_urls(content_hash);
    """

    def __init__(self, path: str | Path = ":memory:"):
        super().__init__(path)
        self.simhash_index = SimHashIndex()
        for row in self.execute(
                "SELECT url_hash, simhash FROM seen_urls WHERE simhash != 0"):
            self.simhash_index.add(row["url_hash"], row["simhash"] & (2**64 - 1))

    def seen_url(self, url: str) -> bool:
        h = content_hash(normalize_url(url))
        return self.execute(
            "SELECT 1 FROM seen_urls WHERE url_hash=?", (h,)).fetchone() is not None

    def seen_content(self, text: str) -> bool:
        return self.execute(
            "SELECT 1 FROM seen_urls WHERE content_hash=?",
            (content_hash(text),)).fetchone() is not None

    def near_duplicate(self, text: str) -> tuple[bool, int]:
        fp = simhash(text)
        return self.simhash_index.find_near(fp) is not None, fp

    def record(self, url: str, text: str = "", fp: int | None = None) -> None:
        norm = normalize_url(url)
        h = content_hash(norm)
        if fp is None and text:
            fp = simhash(text)
        fp = fp or 0
        # store as signed for SQLite
        signed = fp - 2**64 if fp >= 2**63 else fp
        self.execute(
            "INSERT OR REPLACE INTO seen_urls VALUES (?,?,?,?,?)",
            (h, norm, content_hash(text) if text else "", signed, time.time()))
        self.commit()
        if fp:
            self.simhash_index.add(h, fp)

    def count(self) -> int:
        return int(self.execute(
            "SELECT COUNT(*) c FROM seen_urls").fetchone()["c"])
