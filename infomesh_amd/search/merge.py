"""Reciprocal Rank Fusion of result lists.

Reference parity: infomesh/search/merge.py:19-133 (RRF with k=60,
per-source weights, source labels fts/vector/hybrid). Used both for
CPU hybrid merge and for the host-side fusion of per-GPU-shard top-k
lists after the RCCL all-gather.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Sequence

from ..index.local_store import SearchHit

RRF_K = 60


@dataclass
class MergedHit:
    url: str
    title: str
    snippet: str
    score: float
    sources: list[str] = field(default_factory=list)
    doc_id: int = -1
    shard: int = -1
    per_source_rank: dict[str, int] = field(default_factory=dict)


def merge_results(result_lists: Sequence[Sequence[SearchHit]],
                  sources: Sequence[str] | None = None,
                  weights: Sequence[float] | None = None,
                  k: int = RRF_K,
                  limit: int = 10) -> list[MergedHit]:
    """RRF-merge N ranked lists keyed by URL.

    score(url) = Σ_lists w_i / (k + rank_i(url)), rank 1-based."""
    n = len(result_lists)
    if sources is None:
        sources = [f"list{i}" for i in range(n)]
    if weights is None:
        weights = [1.0] * n
    merged: dict[str, MergedHit] = {}
    for li, hits in enumerate(result_lists):
        for rank, hit in enumerate(hits, start=1):
            m = merged.get(hit.url)
            if m is None:
                m = MergedHit(url=hit.url, title=hit.title,
                              snippet=hit.snippet, score=0.0,
                              doc_id=hit.doc_id)
                merged[hit.url] = m
            m.score += weights[li] / (k + rank)
            m.sources.append(sources[li])
            m.per_source_rank[sources[li]] = rank
            if not m.title and hit.title:
                m.title = hit.title
            if not m.snippet and hit.snippet:
                m.snippet = hit.snippet
    out = sorted(merged.values(), key=lambda m: -m.score)
    for m in out:
        m.sources = sorted(set(m.sources))
    return out[:limit]
