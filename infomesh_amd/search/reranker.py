"""Reranker orchestration: reorder top-N candidates, failure-safe
passthrough.

Reference parity: infomesh/search/reranker.py (top-20 LLM rerank with
passthrough on failure) — the scoring model here is the in-process
bge-reranker-base cross-encoder on MFMA kernels (models/reranker.py)
instead of a prompted LLM.
"""
from __future__ import annotations

import logging
from typing import Any, Protocol, Sequence

log = logging.getLogger("infomesh.reranker")


class PairScorer(Protocol):
    def rerank(self, query: str, passages: list[str],
               keep: int = 10) -> list[tuple[int, float]]: ...


def rerank_results(query: str, results: Sequence[Any],
                   scorer: PairScorer | None,
                   top_n: int = 100, keep: int = 10,
                   text_of=None) -> list[Any]:
    """Rerank `results[:top_n]` by cross-encoder score; pass the input
    through unchanged on any failure (reference behavior)."""
    if scorer is None or not results:
        return list(results)[:keep]
    cands = list(results)[:top_n]
    if text_of is None:
        def text_of(r):
            return " ".join(filter(None, (
                getattr(r, "title", "") or (r.get("title") if isinstance(r, dict) else ""),
                getattr(r, "snippet", "") or (r.get("snippet") if isinstance(r, dict) else ""))))
    try:
        passages = [text_of(r) or " " for r in cands]
        order = scorer.rerank(query, passages, keep=keep)
        reranked = [cands[i] for i, _ in order if 0 <= i < len(cands)]
        for (i, score), r in zip(order, reranked):
            try:
                r.rerank_score = score
            except AttributeError:
                pass
        rest = [r for j, r in enumerate(cands)
                if j not in {i for i, _ in order}]
        return (reranked + rest)[:keep]
    except Exception as e:
        log.warning("rerank failed (%s); passthrough", e)
        return cands[:keep]
