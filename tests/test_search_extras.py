"""Tests: rag, reranker orchestration, facets, feedback, explain,
formatter, quality, extended, summarizer engine + verify."""
from __future__ import annotations

import time

from infomesh_amd.index.local_store import SearchHit
from infomesh_amd.search.explain import explain_search, render_explanation
from infomesh_amd.search.extended import (SummaryCache, batch_search,
                                          cross_validate_results,
                                          keyword_translate)
from infomesh_amd.search.facets import (cluster_by_domain, compute_facets,
                                        jaccard, remove_near_duplicates)
from infomesh_amd.search.feedback import FeedbackStore
from infomesh_amd.search.formatter import format_json, format_text
from infomesh_amd.search.quality import (ABTest, diversify, mrr, ndcg,
                                         temporal_hint)
from infomesh_amd.search.rag import (chunk_results, extract_answer,
                                     extract_entities, format_rag_output,
                                     toxicity_filter)
from infomesh_amd.search.reranker import rerank_results
from infomesh_amd.summarizer.engine import (ExtractiveBackend,
                                            SummarizationEngine)
from infomesh_amd.summarizer.verify import (cross_validate_summaries,
                                            verify_summary)


def _hit(url, title="t", snippet="s", score=0.5, domain="", lang="en",
         age_s=0.0):
    h = SearchHit(doc_id=0, url=url, title=title, snippet=snippet,
                  bm25=1.0, language=lang,
                  domain=domain or url.split("/")[2],
                  crawled_at=time.time() - age_s)
    h.score = score
    return h


# -------------------------------------------------------------------- rag

def test_chunk_and_answer():
    results = [{"url": "https://a.com", "title": "Python",
                "text": "Filler sentence one here. " * 10 +
                        "The answer to the question is forty two exactly. " +
                        "More filler text follows. " * 10}]
    chunks = chunk_results("answer question forty", results)
    assert chunks and "forty two" in chunks[0].text
    answer, conf = extract_answer("what is the answer to the question",
                                  results)
    assert "forty two" in answer and conf > 0


def test_entities_and_toxicity():
    text = ("Advanced Micro Devices builds the MI355X accelerator. "
            "Advanced Micro Devices is based in Santa Clara.")
    ents = extract_entities(text)
    assert any("Advanced Micro Devices" in e for e in ents)
    assert toxicity_filter("normal text about computers")


def test_format_rag_output():
    out = format_rag_output("gpu kernels", [
        {"url": "https://a.com", "title": "A",
         "text": "GPU kernels run on compute units. " * 5}],
        answer_mode=True)
    assert out.chunks and out.answer


# --------------------------------------------------------------- reranker

class _FlipScorer:
    def rerank(self, query, passages, keep=10):
        order = list(range(len(passages)))[::-1]
        return [(i, float(len(passages) - n)) for n, i in enumerate(order)][:keep]


class _BrokenScorer:
    def rerank(self, *a, **k):
        raise RuntimeError("boom")


def test_rerank_orders_and_passthrough():
    hits = [_hit(f"https://a.com/{i}") for i in range(4)]
    out = rerank_results("q", hits, _FlipScorer(), keep=4)
    assert out[0].url == "https://a.com/3"
    out2 = rerank_results("q", hits, _BrokenScorer(), keep=2)
    assert [h.url for h in out2] == [h.url for h in hits[:2]]
    assert rerank_results("q", hits, None, keep=2) == hits[:2]


# ----------------------------------------------------------------- facets

def test_facets_and_clustering():
    hits = [_hit("https://a.com/1", age_s=0),
            _hit("https://a.com/2", age_s=10 * 86400),
            _hit("https://b.com/1", lang="de")]
    f = compute_facets(hits)
    assert f["domains"]["a.com"] == 2
    assert f["languages"]["de"] == 1
    assert f["dates"]["today"] >= 1
    clustered = cluster_by_domain(hits, max_per_domain=1)
    assert clustered[1].domain == "b.com"


def test_near_dup_removal():
    hits = [_hit("https://a.com/1", snippet="the same exact snippet text here"),
            _hit("https://b.com/1", snippet="the same exact snippet text here"),
            _hit("https://c.com/1", snippet="completely different content")]
    out = remove_near_duplicates(hits, threshold=0.8)
    assert len(out) == 2
    assert jaccard("a b c", "a b c") == 1.0


# --------------------------------------------------------------- feedback

def test_feedback_boosts():
    fs = FeedbackStore()
    for _ in range(5):
        fs.record("https://good.com/x", "cite", "q")
    fs.record("https://bad.com/y", "skip", "q")
    assert fs.url_boost("https://good.com/x") > 0
    assert fs.url_boost("https://bad.com/y") < 0
    assert fs.top_urls()[0][0] == "https://good.com/x"
    assert fs.stats()["cite"] == 5
    fs.close()


# ---------------------------------------------------------------- explain

def test_explain(seeded_store):
    out = explain_search(seeded_store, "python tutorial")
    assert out and "components" in out[0]
    total = sum(c["value"] * c["weight"]
                for c in out[0]["components"].values())
    assert abs(total - out[0]["total"]) < 1e-3
    text = render_explanation(out)
    assert "bm25" in text


# -------------------------------------------------------------- formatter

def test_formatters(seeded_store):
    from infomesh_amd.search.query import search_local
    resp = search_local(seeded_store, "python tutorial")
    j = format_json(resp)
    assert '"results"' in j
    t = format_text(resp)
    assert "python" in t.lower()


# ---------------------------------------------------------------- quality

def test_quality_metrics():
    assert ndcg([3, 2, 1]) == 1.0
    assert ndcg([1, 2, 3]) < 1.0
    assert mrr([False, True]) == 0.5
    assert temporal_hint("latest rocm news") == "fresh"
    assert temporal_hint("gpu 2024 report") == "year:2024"
    assert temporal_hint("plain query") is None
    ab = ABTest()
    p = ab.assign("q1")
    assert p in ab.profiles
    assert ab.assign("q1") == p  # deterministic
    ab.record_outcome(p, 0.9)
    assert ab.report()[p] == 0.9
    out = diversify([1, 1, 1, 2], key_fn=lambda x: x, max_per_key=2)
    assert out == [1, 1, 2, 1]


# --------------------------------------------------------------- extended

def test_batch_and_summary_cache():
    calls = []
    out = batch_search(lambda q: calls.append(q) or q.upper(), ["a", "b"])
    assert out == ["A", "B"]
    sc = SummaryCache()
    sc.put_summary("https://a.com", "h1", "the summary")
    assert sc.get_summary("https://a.com", "h1") == "the summary"
    assert sc.get_summary("https://a.com", "h2") is None


def test_cross_validate_flags_outlier():
    lists = {
        "shard0": [{"url": "u1", "score": 0.5, "snippet": "gpu kernels fast",
                    "title": ""}],
        "shard1": [{"url": "u2", "score": 0.55, "snippet": "gpu kernels slow",
                    "title": ""},
                   {"url": "u3", "score": 99.0,
                    "snippet": "zzz qqq unrelated spam", "title": ""}],
    }
    flagged = cross_validate_results(lists)
    assert "shard1" in flagged
    assert flagged["shard1"][0]["url"] == "u3"


def test_keyword_translate():
    assert keyword_translate("fast DB", {"db": "database"}) == "fast database"


# ------------------------------------------------------------- summarizer

def test_extractive_summarizer():
    eng = SummarizationEngine(backend=ExtractiveBackend())
    text = ("The MI355X is an AMD accelerator. It has 288 GB of HBM3E. "
            "Bandwidth reaches eight terabytes per second. "
            "Unrelated filler sentence about something else entirely. " * 3)
    res = eng.summarize(text, title="MI355X")
    assert res.backend == "extractive"
    assert "MI355X" in res.summary or "AMD" in res.summary


def test_summarize_results():
    eng = SummarizationEngine(backend=ExtractiveBackend())
    res = eng.summarize_results(
        [{"title": "A", "snippet": "GPU kernels are fast on MI355X."}],
        query="gpu")
    assert res.summary


def test_verify_summary():
    source = ("The GPU has 288 GB of memory. The bandwidth is 8 TB/s. "
              "It supports bf16 compute.")
    good = "The GPU has 288 GB of memory and supports bf16 compute."
    rep = verify_summary(good, source)
    assert rep.support_score > 0.5
    assert rep.numbers_ok
    bad = "The GPU has 999 GB of memory from Mars with quantum lasers."
    rep2 = verify_summary(bad, source)
    assert not rep2.numbers_ok or rep2.support_score < 0.5


def test_verify_negation_contradiction():
    source = "The kernel does not support fp64 math."
    summary = "The kernel supports fp64 math for all operations today."
    rep = verify_summary(summary, source, support_threshold=0.4)
    assert rep.contradictions or rep.support_score < 1.0


def test_cross_validate_summaries():
    sims = cross_validate_summaries([
        "gpu kernels run fast on mi355x hardware",
        "gpu kernels run quickly on mi355x hardware",
        "bananas are yellow fruit entirely unrelated"])
    assert sims[0] > sims[2]


def test_extract_keywords():
    from infomesh_amd.search.nlp import extract_keywords
    text = ("The quantum computer uses quantum gates. Quantum "
            "error correction protects the computer from noise. "
            "the and of to in is 42")
    kws = extract_keywords(text, top_n=5)
    words = [w for w, _ in kws]
    assert words[0] == "quantum" and kws[0][1] == 3
    assert "computer" in words
    assert "the" not in words and "42" not in words
    assert len(kws) <= 5
