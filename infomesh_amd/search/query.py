"""Query orchestration: local, hybrid, distributed.

Reference parity: infomesh/search/query.py —
- search_local: CJK preprocess → sanitize → FTS5 (2× limit) →
  sparse-result query expansion → composite ranking → passage-based
  snippet enhancement (query.py:82-241).
- search_hybrid: FTS + dense vector + RRF merge (query.py:244-319).
- search_distributed: local + shard fan-out with degraded local-only
  fallback (query.py:388-531) — here the fan-out is the intra-node GPU
  fabric (RCCL all-gather over xGMI) instead of libp2p peers.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Protocol, Sequence

from ..index.local_store import LocalStore, SearchHit
from ..index.ranking import rank_local_results, AuthorityFn, TrustFn
from .cjk import tokenize_query_cjk
from .merge import MergedHit, merge_results
from .nlp import expand_query, parse_query_filters, remove_stop_words
from .passage import select_best_passage, highlight


class DenseSearcher(Protocol):
    """A dense (vector) searcher — the GPU cosine-top-k engine, or a fake."""

    def search(self, query: str, limit: int = 10) -> list[SearchHit]: ...


class ShardFabric(Protocol):
    """The distributed query plane (parallel/query_plane.py), or a fake."""

    @property
    def world_size(self) -> int: ...

    def search(self, query: str, limit_per_shard: int = 20) -> list[SearchHit]: ...


@dataclass
class SearchResponse:
    query: str
    effective_query: str
    results: list[Any]           # SearchHit | MergedHit
    elapsed_ms: float
    mode: str = "local"
    expanded: list[str] = field(default_factory=list)
    degraded: bool = False
    total_candidates: int = 0


def preprocess_query(query: str, language: str = "en") -> str:
    q = tokenize_query_cjk(query.strip())
    q = remove_stop_words(q, language)
    return q


def search_local(store: LocalStore, query: str, limit: int = 10,
                 authority_fn: AuthorityFn | None = None,
                 boost_fn=None,
                 trust_fn: TrustFn | None = None,
                 enhance_snippets: bool = True) -> SearchResponse:
    t0 = time.perf_counter()
    pq = parse_query_filters(query)
    eff = preprocess_query(pq.text)
    hits = store.search(eff, limit=limit * 2, language=pq.language,
                        domain=pq.site, after=pq.after, before=pq.before)
    expanded: list[str] = []
    if len(hits) < max(3, limit // 2):
        # Sparse results → synonym expansion (query.py:136-156).
        for alt in expand_query(eff):
            expanded.append(alt)
            more = store.search(alt, limit=limit, language=pq.language,
                                domain=pq.site, after=pq.after, before=pq.before)
            seen = {h.url for h in hits}
            hits.extend(h for h in more if h.url not in seen)
            if len(hits) >= limit:
                break
    ranked = rank_local_results(eff, hits, authority_fn=authority_fn,
                                trust_fn=trust_fn, boost_fn=boost_fn)
    ranked = ranked[:limit]
    if enhance_snippets:
        for h in ranked:
            doc = store.get_document(h.doc_id)
            if doc is not None and doc.text:
                h.snippet = highlight(
                    select_best_passage(eff, doc.text), eff)
    return SearchResponse(
        query=query, effective_query=eff, results=ranked,
        elapsed_ms=(time.perf_counter() - t0) * 1e3, mode="local",
        expanded=expanded, total_candidates=len(hits))


def search_hybrid(store: LocalStore, dense: DenseSearcher | None,
                  query: str, limit: int = 10,
                  authority_fn: AuthorityFn | None = None,
                  trust_fn: TrustFn | None = None,
                  rrf_k: int = 60, boost_fn=None) -> SearchResponse:
    t0 = time.perf_counter()
    local = search_local(store, query, limit=limit * 2,
                         authority_fn=authority_fn, trust_fn=trust_fn,
                         enhance_snippets=False, boost_fn=boost_fn)
    lists: list[Sequence[SearchHit]] = [local.results]
    sources = ["fts"]
    weights = [1.0]
    if dense is not None:
        vec_hits = dense.search(local.effective_query, limit=limit * 2)
        lists.append(vec_hits)
        sources.append("vector")
        weights.append(1.0)
    merged = merge_results(lists, sources=sources, weights=weights,
                           k=rrf_k, limit=limit)
    _enhance_merged(store, local.effective_query, merged)
    return SearchResponse(
        query=query, effective_query=local.effective_query, results=merged,
        elapsed_ms=(time.perf_counter() - t0) * 1e3, mode="hybrid",
        total_candidates=sum(len(l) for l in lists))


def search_distributed(store: LocalStore, fabric: ShardFabric | None,
                       query: str, limit: int = 10,
                       dense: DenseSearcher | None = None,
                       limit_per_shard: int = 20,
                       authority_fn: AuthorityFn | None = None,
                       trust_fn: TrustFn | None = None) -> SearchResponse:
    """Local + shard fan-out; degrades to local/hybrid when the fabric
    is absent (reference behavior: query.py:471-490)."""
    t0 = time.perf_counter()
    if fabric is None or fabric.world_size <= 1:
        resp = search_hybrid(store, dense, query, limit=limit,
                             authority_fn=authority_fn, trust_fn=trust_fn)
        resp.mode = "distributed"
        resp.degraded = fabric is None
        resp.elapsed_ms = (time.perf_counter() - t0) * 1e3
        return resp
    eff = preprocess_query(parse_query_filters(query).text)
    shard_hits = fabric.search(eff, limit_per_shard=limit_per_shard)
    # Merge by URL keeping the best score (query.py:492-508).
    best: dict[str, SearchHit] = {}
    for h in shard_hits:
        cur = best.get(h.url)
        if cur is None or h.score > cur.score:
            best[h.url] = h
    results = sorted(best.values(), key=lambda h: -h.score)[:limit]
    return SearchResponse(
        query=query, effective_query=eff, results=results,
        elapsed_ms=(time.perf_counter() - t0) * 1e3, mode="distributed",
        total_candidates=len(shard_hits))


def _enhance_merged(store: LocalStore, eff_query: str,
                    merged: list[MergedHit]) -> None:
    for m in merged:
        if m.doc_id >= 0:
            doc = store.get_document(m.doc_id)
            if doc is not None and doc.text:
                m.snippet = highlight(
                    select_best_passage(eff_query, doc.text), eff_query)
