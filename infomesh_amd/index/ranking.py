"""Composite result ranking.

Reference parity: infomesh/index/ranking.py —
score = 0.40·bm25 + 0.15·freshness + 0.10·trust + 0.15·authority
      + 0.15·title_match + 0.05·url_path  (ranking.py:24-31)
with exponential freshness decay (7-day half-life, floor 0.05,
ranking.py:33-41) and BM25 normalized per batch via saturation s/(s+k)
(ranking.py:86-101).
"""
from __future__ import annotations

import math
import time
from dataclasses import dataclass
from typing import Callable, Sequence
from urllib.parse import urlparse

from .local_store import SearchHit

W_BM25 = 0.40
W_FRESHNESS = 0.15
W_TRUST = 0.10
W_AUTHORITY = 0.15
W_TITLE = 0.15
W_URL_PATH = 0.05

FRESHNESS_HALF_LIFE_S = 7 * 86400.0
FRESHNESS_FLOOR = 0.05
BM25_SATURATION_K = 1.2


@dataclass
class ScoreBreakdown:
    bm25: float
    freshness: float
    trust: float
    authority: float
    title_match: float
    url_path: float

    @property
    def total(self) -> float:
        from ..utils.plugins import GLOBAL_PLUGINS
        custom = GLOBAL_PLUGINS.get_single("scorer")
        if custom is not None:
            return float(custom(self))
        return (W_BM25 * self.bm25 + W_FRESHNESS * self.freshness
                + W_TRUST * self.trust + W_AUTHORITY * self.authority
                + W_TITLE * self.title_match + W_URL_PATH * self.url_path)


def freshness_score(crawled_at: float, now: float | None = None) -> float:
    """Exponential decay with 7-day half-life, floored at 0.05."""
    if crawled_at <= 0:
        return FRESHNESS_FLOOR
    now = time.time() if now is None else now
    age = max(0.0, now - crawled_at)
    return max(FRESHNESS_FLOOR, 0.5 ** (age / FRESHNESS_HALF_LIFE_S))


def normalize_bm25(raw: float, batch_max: float) -> float:
    """Normalize by the per-batch max, then saturate s/(s+k)."""
    if batch_max <= 0:
        return 0.0
    s = max(0.0, raw) / batch_max
    return s / (s + BM25_SATURATION_K / (1.0 + BM25_SATURATION_K))


def _query_terms(query: str) -> list[str]:
    return [t for t in query.lower().split() if t]


def title_match_score(query: str, title: str) -> float:
    terms = _query_terms(query)
    if not terms or not title:
        return 0.0
    tl = title.lower()
    hits = sum(1 for t in terms if t in tl)
    score = hits / len(terms)
    if tl.startswith(terms[0]):
        score = min(1.0, score + 0.2)
    return score


def url_path_score(query: str, url: str) -> float:
    terms = _query_terms(query)
    if not terms:
        return 0.0
    try:
        path = (urlparse(url).path or "").lower()
    except ValueError:
        return 0.0
    # Short paths are better; term presence in the path is a bonus.
    depth = max(0, path.count("/") - 1)
    base = 1.0 / (1.0 + 0.3 * depth)
    hits = sum(1 for t in terms if t in path)
    return min(1.0, base * (0.5 + 0.5 * hits / len(terms)))


AuthorityFn = Callable[[str], float]
TrustFn = Callable[[str], float]


def rank_local_results(query: str, hits: Sequence[SearchHit],
                       authority_fn: AuthorityFn | None = None,
                       trust_fn: TrustFn | None = None,
                       now: float | None = None,
                       explain: bool = False,
                       boost_fn=None,
                       ) -> list[SearchHit] | list[tuple[SearchHit, ScoreBreakdown]]:
    """Composite-rank FTS hits in place (reference: ranking.py:241-285).
    boost_fn: implicit-feedback additive boost per url (capped by
    FeedbackStore; reference applies it in ranking, feedback.py)."""
    if not hits:
        return []
    batch_max = max(h.bm25 for h in hits)
    out = []
    for h in hits:
        bd = ScoreBreakdown(
            bm25=normalize_bm25(h.bm25, batch_max),
            freshness=freshness_score(h.crawled_at, now),
            trust=(trust_fn(h.domain) if trust_fn else 0.5),
            authority=(authority_fn(h.url) if authority_fn else 0.5),
            title_match=title_match_score(query, h.title),
            url_path=url_path_score(query, h.url),
        )
        h.score = bd.total + (boost_fn(h.url) if boost_fn else 0.0)
        out.append((h, bd))
    out.sort(key=lambda p: p[0].score, reverse=True)
    if explain:
        return out
    return [h for h, _ in out]


def log_sigmoid(x: float) -> float:
    """Squash an unbounded model logit into (0,1) for merging."""
    return 1.0 / (1.0 + math.exp(-x))
