"""Extended search utilities: batch search, summary cache, cross-shard
result validation.

Reference parity: infomesh/search/extended.py (batch_search, summary
cache) and infomesh/search/cross_validate.py (fabricated-result
detection via score deviation + snippet similarity).
"""
from __future__ import annotations

import statistics
from typing import Any, Callable, Sequence

from .cache import QueryCache
from .facets import jaccard


def batch_search(search_fn: Callable[[str], Any],
                 queries: Sequence[str]) -> list[Any]:
    """Run many queries through one search callable (the GPU engine
    batches internally; this is the orchestration-level fallback)."""
    return [search_fn(q) for q in queries]


class SummaryCache(QueryCache):
    """TTL cache for per-URL summaries (reference: extended.py:116)."""

    def __init__(self, max_entries: int = 500, ttl_s: float = 3600.0):
        super().__init__(max_entries, ttl_s)

    def get_summary(self, url: str, text_hash: str) -> str | None:
        return self.get(self.make_key(url, h=text_hash))

    def put_summary(self, url: str, text_hash: str, summary: str) -> None:
        self.put(self.make_key(url, h=text_hash), summary)


def cross_validate_results(result_lists: dict[str, list[dict]],
                           score_z_threshold: float = 3.0,
                           sim_threshold: float = 0.15) -> dict[str, list]:
    """Flag suspicious per-source results (reference:
    cross_validate.py:89-287): scores far above the cross-source
    distribution, or snippets with no lexical overlap with any other
    source's results for the same query."""
    all_scores = [float(r.get("score", 0.0))
                  for rs in result_lists.values() for r in rs]
    flagged: dict[str, list] = {}
    if len(all_scores) < 3:
        return flagged
    mean = statistics.mean(all_scores)
    stdev = statistics.pstdev(all_scores) or 1e-9
    sources = list(result_lists)
    for src, rs in result_lists.items():
        others_text = " ".join(
            (r.get("snippet", "") + " " + r.get("title", ""))
            for other in sources if other != src
            for r in result_lists[other])
        for r in rs:
            reasons = []
            z = (float(r.get("score", 0.0)) - mean) / stdev
            if z > score_z_threshold:
                reasons.append(f"score z={z:.1f}")
            snippet = r.get("snippet", "") + " " + r.get("title", "")
            if others_text and snippet.strip() and \
                    jaccard(snippet, others_text) < sim_threshold \
                    and len(sources) > 1:
                reasons.append("no cross-source overlap")
            if reasons:
                flagged.setdefault(src, []).append(
                    {"url": r.get("url"), "reasons": reasons})
    return flagged


def keyword_translate(query: str, mapping: dict[str, str] | None = None
                      ) -> str:
    """Tiny keyword translation hook (reference: extended.py:285)."""
    mapping = mapping or {}
    return " ".join(mapping.get(t.lower(), t) for t in query.split())
