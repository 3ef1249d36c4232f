"""Search-quality metrics and A/B ranking profiles.

Reference parity: infomesh/search/quality.py (NDCG/MRR, A/B ranking
profiles, result diversification, temporal hints, intent classifier —
intent lives in passage.classify_intent).
"""
from __future__ import annotations

import math
import re
from dataclasses import dataclass, field
from typing import Sequence


def dcg(relevances: Sequence[float]) -> float:
    return sum(rel / math.log2(i + 2) for i, rel in enumerate(relevances))


def ndcg(relevances: Sequence[float], k: int | None = None) -> float:
    rels = list(relevances)[:k] if k else list(relevances)
    ideal = sorted(relevances, reverse=True)[:len(rels)]
    denom = dcg(ideal)
    return dcg(rels) / denom if denom > 0 else 0.0


def mrr(ranked_relevant: Sequence[bool]) -> float:
    for i, rel in enumerate(ranked_relevant, start=1):
        if rel:
            return 1.0 / i
    return 0.0


@dataclass
class RankingProfile:
    name: str
    weights: dict[str, float]


DEFAULT_PROFILES = {
    "default": RankingProfile("default", {
        "bm25": 0.40, "freshness": 0.15, "trust": 0.10,
        "authority": 0.15, "title_match": 0.15, "url_path": 0.05}),
    "fresh": RankingProfile("fresh", {
        "bm25": 0.30, "freshness": 0.35, "trust": 0.10,
        "authority": 0.10, "title_match": 0.10, "url_path": 0.05}),
    "authoritative": RankingProfile("authoritative", {
        "bm25": 0.30, "freshness": 0.05, "trust": 0.20,
        "authority": 0.30, "title_match": 0.10, "url_path": 0.05}),
}


@dataclass
class ABTest:
    """Deterministic per-query assignment to ranking profiles + outcome
    tracking (reference: quality.py:59-157)."""
    profiles: tuple[str, str] = ("default", "fresh")
    outcomes: dict[str, list[float]] = field(default_factory=dict)

    def assign(self, query: str) -> str:
        h = hash(query) & 0xFFFF
        return self.profiles[h % len(self.profiles)]

    def record_outcome(self, profile: str, score: float) -> None:
        self.outcomes.setdefault(profile, []).append(score)

    def report(self) -> dict[str, float]:
        return {p: (sum(v) / len(v) if v else 0.0)
                for p, v in self.outcomes.items()}


_YEAR_RE = re.compile(r"\b(19|20)\d{2}\b")
_TEMPORAL_HINTS = ("latest", "recent", "today", "news", "current", "new")


def temporal_hint(query: str) -> str | None:
    """'fresh' | 'year:NNNN' | None (reference: quality.py:318)."""
    low = query.lower()
    m = _YEAR_RE.search(query)
    if m:
        return f"year:{m.group(0)}"
    if any(h in low.split() for h in _TEMPORAL_HINTS):
        return "fresh"
    return None


def diversify(results: Sequence, key_fn, max_per_key: int = 2) -> list:
    """Greedy diversification preserving order (quality.py:255)."""
    from collections import defaultdict
    counts = defaultdict(int)
    out, rest = [], []
    for r in results:
        k = key_fn(r)
        if counts[k] < max_per_key:
            out.append(r)
            counts[k] += 1
        else:
            rest.append(r)
    return out + rest
