"""Search-layer CPU tests: ranking, merge, nlp, passage, cjk, cache, query."""
from __future__ import annotations

import time

from infomesh_amd.index.local_store import SearchHit
from infomesh_amd.index.ranking import (
    ScoreBreakdown, freshness_score, normalize_bm25, rank_local_results,
    title_match_score)
from infomesh_amd.search.cache import QueryCache
from infomesh_amd.search.cjk import (cjk_ratio, contains_cjk,
                                     recommend_tokenizer, tokenize_query_cjk)
from infomesh_amd.search.merge import merge_results
from infomesh_amd.search.nlp import (did_you_mean, edit_distance, expand_query,
                                     parse_query_filters, remove_stop_words,
                                     RelatedSearchTracker)
from infomesh_amd.search.passage import (classify_intent, highlight,
                                         score_passage, select_best_passage,
                                         split_passages)
from infomesh_amd.search.query import (search_distributed, search_hybrid,
                                       search_local)


def _hit(url, bm25=1.0, title="", age_s=0.0):
    return SearchHit(doc_id=0, url=url, title=title, snippet="", bm25=bm25,
                     crawled_at=time.time() - age_s)


# ---------------------------------------------------------------- ranking

def test_freshness_decay():
    now = time.time()
    assert freshness_score(now, now) == 1.0
    week = freshness_score(now - 7 * 86400, now)
    assert abs(week - 0.5) < 0.01
    assert freshness_score(now - 365 * 86400, now) == 0.05  # floor


def test_bm25_normalization_monotonic():
    assert normalize_bm25(10, 10) > normalize_bm25(5, 10) > normalize_bm25(0, 10)
    assert normalize_bm25(5, 0) == 0.0


def test_title_match():
    assert title_match_score("python tutorial", "Python Tutorial") > 0.9
    assert title_match_score("python", "Unrelated") == 0.0


def test_rank_orders_by_composite():
    fresh = _hit("https://a.com/fresh", bm25=5.0, title="query match", age_s=0)
    stale = _hit("https://a.com/stale", bm25=5.0, title="query match",
                 age_s=90 * 86400)
    ranked = rank_local_results("query match", [stale, fresh])
    assert ranked[0].url.endswith("fresh")
    assert ranked[0].score > ranked[1].score


def test_rank_explain_breakdown():
    out = rank_local_results("q", [_hit("https://a.com/x", 2.0)], explain=True)
    hit, bd = out[0]
    assert isinstance(bd, ScoreBreakdown)
    assert abs(hit.score - bd.total) < 1e-9


# ------------------------------------------------------------------ merge

def test_rrf_merge_overlap_boost():
    l1 = [_hit("https://a.com/1"), _hit("https://a.com/2")]
    l2 = [_hit("https://a.com/2"), _hit("https://a.com/3")]
    merged = merge_results([l1, l2], sources=["fts", "vector"])
    assert merged[0].url == "https://a.com/2"  # appears in both lists
    assert merged[0].sources == ["fts", "vector"]


def test_rrf_weights():
    l1 = [_hit("https://a.com/1")]
    l2 = [_hit("https://a.com/2")]
    merged = merge_results([l1, l2], weights=[0.1, 1.0])
    assert merged[0].url == "https://a.com/2"


# -------------------------------------------------------------------- nlp

def test_stop_words():
    assert remove_stop_words("what is the python tutorial") == "python tutorial"
    assert remove_stop_words("the the the") == "the the the"  # never empties


def test_expand_query():
    alts = expand_query("fast database")
    assert any("quick" in a for a in alts)


def test_edit_distance():
    assert edit_distance("kitten", "sitting") == 3
    assert edit_distance("abc", "abc") == 0
    assert edit_distance("a", "abcdefgh") > 3  # capped


def test_did_you_mean():
    vocab = {"python": 100, "tutorial": 50}
    assert did_you_mean("pythn tutorial", vocab) == "python tutorial"
    assert did_you_mean("python tutorial", vocab) is None


def test_parse_filters():
    pq = parse_query_filters("gpu kernels site:rocm.docs.amd.com lang:en after:2024-01-01")
    assert pq.text == "gpu kernels"
    assert pq.site == "rocm.docs.amd.com"
    assert pq.language == "en"
    assert pq.after is not None


def test_related_tracker():
    t = RelatedSearchTracker()
    t.record("python async")
    t.record("python tutorial")
    assert "python tutorial" in t.related("python guide")


# ---------------------------------------------------------------- passage

def test_split_and_score_passages():
    text = ("Python is a language. " * 10 +
            "The asyncio module handles events. " * 10)
    ps = split_passages(text)
    assert len(ps) >= 2
    assert score_passage(["asyncio", "events"], ps[-1].text) > \
        score_passage(["asyncio", "events"], ps[0].text)


def test_select_best_passage():
    text = "Intro sentence here. " * 20 + \
        "The secret answer is forty two. " + "Outro filler. " * 20
    best = select_best_passage("secret answer", text)
    assert "forty two" in best


def test_highlight():
    assert highlight("python is great", "python") == "<b>python</b> is great"


def test_intent():
    assert classify_intent("how do I sort a list") == "question"
    assert classify_intent("buy cheap gpu") == "transactional"
    assert classify_intent("github login") == "navigational"
    assert classify_intent("bm25 scoring function") == "informational"


# -------------------------------------------------------------------- cjk

def test_cjk_detect_and_expand():
    assert contains_cjk("日本語のテキスト")
    assert not contains_cjk("english only")
    assert cjk_ratio("日本語abc") > 0.4
    out = tokenize_query_cjk("東京タワー")
    assert "東京" in out.split()
    assert tokenize_query_cjk("plain english") == "plain english"


def test_recommend_tokenizer():
    assert recommend_tokenizer(["日本語のテキストです" * 3]) == "trigram"
    assert recommend_tokenizer(["english text"]) == "unicode61"


# ------------------------------------------------------------------ cache

def test_cache_lru_ttl():
    c = QueryCache(max_entries=2, ttl_s=100)
    c.put("a", 1)
    c.put("b", 2)
    c.put("c", 3)  # evicts a
    assert c.get("a") is None
    assert c.get("b") == 2
    assert c.stats()["entries"] == 2


def test_cache_ttl_expiry(monkeypatch):
    c = QueryCache(ttl_s=0.0)
    c.put("k", "v")
    time.sleep(0.01)
    assert c.get("k") is None


def test_cache_key_stable():
    k1 = QueryCache.make_key("q", limit=10, lang=None)
    k2 = QueryCache.make_key("q", limit=10)
    assert k1 == k2


# ------------------------------------------------------------------ query

def test_search_local_pipeline(seeded_store):
    resp = search_local(seeded_store, "what is the BM25 ranking function")
    assert resp.results
    assert resp.results[0].url == "https://en.wikipedia.org/wiki/Okapi_BM25"
    assert resp.elapsed_ms < 1000
    assert "<b>" in resp.results[0].snippet


def test_search_local_site_filter(seeded_store):
    resp = search_local(seeded_store, "documentation site:rocm.docs.amd.com")
    assert all(h.domain == "rocm.docs.amd.com" for h in resp.results)


class _FakeDense:
    def __init__(self, hits):
        self._hits = hits

    def search(self, query, limit=10):
        return self._hits[:limit]


def test_search_hybrid_merges_vector(seeded_store):
    vec = [_hit("https://rocm.docs.amd.com/", bm25=0.9, title="ROCm documentation")]
    resp = search_hybrid(seeded_store, _FakeDense(vec), "gpu computing rccl")
    assert resp.mode == "hybrid"
    urls = [m.url for m in resp.results]
    assert "https://rocm.docs.amd.com/" in urls


class _FakeFabric:
    """Dict-backed shard fan-out fake (the MockInfoMeshDHT analogue —
    reference tests/test_distributed.py:18-36)."""

    def __init__(self, world_size, hits):
        self._ws = world_size
        self._hits = hits

    @property
    def world_size(self):
        return self._ws

    def search(self, query, limit_per_shard=20):
        return self._hits


def test_search_distributed_degraded(seeded_store):
    resp = search_distributed(seeded_store, None, "python tutorial")
    assert resp.degraded
    assert resp.results


def test_search_distributed_merges_shards(seeded_store):
    h1 = _hit("https://a.com/1", 1.0)
    h1.score = 0.9
    h2 = _hit("https://a.com/1", 1.0)
    h2.score = 0.5  # duplicate URL, lower score
    h3 = _hit("https://b.com/2", 1.0)
    h3.score = 0.7
    resp = search_distributed(seeded_store, _FakeFabric(2, [h1, h2, h3]),
                              "anything")
    urls = [h.url for h in resp.results]
    assert urls == ["https://a.com/1", "https://b.com/2"]
    assert resp.results[0].score == 0.9


def test_shard_incremental_build_matches_bulk():
    """build() with existing docs (segment merge) must produce the same
    index as one bulk build of all docs: identical search results AND
    identical BM25 stats (avgdl/norms recomputed over the union)."""
    import numpy as np
    import torch
    from infomesh_amd.index.gpu_index import CpuShard

    rng = np.random.default_rng(7)
    docs = [rng.integers(0, 300, size=rng.integers(4, 25)).astype(np.int64)
            for _ in range(120)]
    g = torch.Generator().manual_seed(7)
    emb = torch.nn.functional.normalize(
        torch.randn(120, 16, generator=g), dim=-1).bfloat16()

    # incremental: 3 batches through add_document + build
    inc = CpuShard()
    start = 0
    for batch in (40, 50, 30):
        for i in range(start, start + batch):
            inc.add_document(1000 + i, docs[i], emb[i])
        inc.build()
        start += batch
    assert inc.n_docs == 120

    # bulk oracle
    bulk = CpuShard()
    for i in range(120):
        bulk.add_document(1000 + i, docs[i], emb[i])
    bulk.build()

    assert abs(inc.avgdl - bulk.avgdl) < 1e-9
    assert np.array_equal(inc.df, bulk.df)
    assert torch.equal(inc.doc_norm, bulk.doc_norm)
    queries = [np.array([5, 17, 40]), np.array([100, 2]),
               np.array([250])]
    qe = torch.nn.functional.normalize(
        torch.randn(3, 16, generator=g), dim=-1)
    hi = inc.search(queries, qe, k=10)
    hb = bulk.search(queries, qe, k=10)
    assert torch.equal(hi.bm25_ids, hb.bm25_ids)
    assert torch.allclose(hi.bm25_scores, hb.bm25_scores, atol=1e-5)
    assert torch.equal(hi.dense_ids, hb.dense_ids)


def test_bm25_term_ids_cjk_bigrams():
    """CJK text must produce GPU BM25 terms (bigrams), and a CJK query
    must retrieve a CJK doc through the shard."""
    import numpy as np
    from infomesh_amd.index.gpu_index import CpuShard, bm25_term_ids

    t = bm25_term_ids("量子计算机")
    # whole run matches \w+ (1 token) + 4 bigrams — consistent on both
    # the index and query side, so the extra run token is harmless
    assert len(t) == 5
    assert len(bm25_term_ids("mixed 量子 text")) == 2 + 1 + 1
    shard = CpuShard()
    shard.add_document(1, bm25_term_ids("量子计算机的研究进展"), None)
    shard.add_document(2, bm25_term_ids("classic english doc"), None)
    shard.build()
    hits = shard.search([bm25_term_ids("量子计算")], None, k=2)
    top = int(hits.bm25_ids[0, 0])
    assert top == 1


def test_bm25_term_ids_unicode_folding():
    """GPU tokenizer matches FTS5 unicode61 semantics: diacritic
    folding and non-Latin scripts."""
    import numpy as np
    from infomesh_amd.index.gpu_index import CpuShard, bm25_term_ids

    assert np.array_equal(bm25_term_ids("café"), bm25_term_ids("cafe"))
    assert np.array_equal(bm25_term_ids("Müller"), bm25_term_ids("muller"))
    ru = bm25_term_ids("квантовый компьютер")
    assert len(ru) == 2  # Cyrillic words are real tokens now
    shard = CpuShard()
    shard.add_document(1, bm25_term_ids("квантовый компьютер исследования"),
                       None)
    shard.add_document(2, bm25_term_ids("english only text"), None)
    shard.build()
    hits = shard.search([bm25_term_ids("квантовый")], None, k=2)
    assert int(hits.bm25_ids[0, 0]) == 1


def test_cpu_shard_fp8_mode():
    """fp8 embedding storage works on the CPU oracle too (upcast
    scoring), so the mode is testable without a GPU."""
    import numpy as np
    import torch
    from infomesh_amd.index.gpu_index import CpuShard
    rng = np.random.default_rng(3)
    docs = [rng.integers(0, 200, size=10).astype(np.int64)
            for _ in range(50)]
    g = torch.Generator().manual_seed(3)
    emb = torch.nn.functional.normalize(
        torch.randn(50, 32, generator=g), dim=-1)
    shard = CpuShard(emb_dtype="fp8")
    for i, d in enumerate(docs):
        shard.add_document(i, d, emb[i])
    shard.build()
    assert shard.embeddings.dtype == torch.float8_e4m3fn
    hits = shard.search([docs[7][:3]], emb[7:8], k=5)
    assert int(hits.dense_ids[0, 0]) == 7   # self-retrieval
    assert float(hits.dense_scores[0, 0]) > 0.9
