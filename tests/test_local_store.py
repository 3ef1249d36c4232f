"""LocalStore / FTS5 tests (reference parity: tests/test_index.py)."""
from __future__ import annotations

import time

import pytest

from infomesh_amd.errors import InfoMeshError
from infomesh_amd.index.local_store import (
    Document, LocalStore, sanitize_fts_query, extract_domain)


def test_add_and_count(store):
    rid = store.add_document(Document(url="https://a.com/1", title="T",
                                      text="hello world content"))
    assert rid is not None
    assert store.count() == 1


def test_dedup_by_url_unchanged(store):
    d = Document(url="https://a.com/1", title="T", text="same text")
    assert store.add_document(d) is not None
    assert store.add_document(d) is None
    assert store.count() == 1


def test_url_update_on_changed_content(store):
    store.add_document(Document(url="https://a.com/1", text="old text"))
    rid = store.add_document(Document(url="https://a.com/1", text="new text"))
    assert rid is not None
    assert store.count() == 1
    assert store.get_document_by_url("https://a.com/1").text == "new text"


def test_dedup_by_text_hash(store):
    store.add_document(Document(url="https://a.com/1", text="identical body"))
    assert store.add_document(
        Document(url="https://b.com/2", text="identical body")) is None


def test_search_basic(seeded_store):
    hits = seeded_store.search("python tutorial")
    assert hits
    assert hits[0].url == "https://docs.python.org/3/tutorial/"
    assert hits[0].bm25 > 0


def test_search_ranks_relevant_first(seeded_store):
    hits = seeded_store.search("BM25 ranking search")
    assert hits[0].url == "https://en.wikipedia.org/wiki/Okapi_BM25"


def test_search_language_filter(seeded_store):
    hits = seeded_store.search("Suchmaschinen", language="de")
    assert len(hits) == 1
    assert hits[0].language == "de"
    assert seeded_store.search("Suchmaschinen", language="en") == []


def test_search_domain_filter(seeded_store):
    hits = seeded_store.search("documentation python", domain="pytorch.org")
    assert all(h.domain == "pytorch.org" for h in hits)


def test_search_date_filters(store):
    store.add_document(Document(url="https://a.com/old", text="apple banana",
                                crawled_at=time.time() - 1e6))
    store.add_document(Document(url="https://a.com/new", text="apple cherry"))
    hits = store.search("apple", after=time.time() - 1000)
    assert [h.url for h in hits] == ["https://a.com/new"]


def test_snippet_contains_highlight(seeded_store):
    hits = seeded_store.search("asyncio coroutines")
    assert "<b>" in hits[0].snippet


def test_malicious_query_does_not_raise(seeded_store):
    for q in ['a"b', "x OR y", "(((", "col:val*", "NEAR/3 foo", '"unclosed']:
        seeded_store.search(q)  # must not raise


def test_sanitize_fts_query():
    assert sanitize_fts_query("hello world") == '"hello" "world"'
    assert sanitize_fts_query('injection" OR 1') == '"injection" "OR" "1"'
    assert sanitize_fts_query("   ") == ""


def test_suggest(seeded_store):
    assert any("Python" in s for s in seeded_store.suggest("Python"))


def test_delete(seeded_store):
    n = seeded_store.count()
    assert seeded_store.delete_by_url("https://pytorch.org/docs/")
    assert seeded_store.count() == n - 1
    assert seeded_store.search("pytorch neural") == []


def test_delete_by_domain(seeded_store):
    assert seeded_store.delete_by_domain("docs.python.org") == 2


def test_bad_tokenizer_rejected():
    with pytest.raises(InfoMeshError):
        LocalStore(":memory:", tokenizer="evil; DROP TABLE")


def test_extract_domain():
    assert extract_domain("https://WWW.Example.com:8443/a?b=c") == "www.example.com"
    assert extract_domain("not a url") == ""


def test_persistence(tmp_path):
    p = tmp_path / "idx.db"
    s = LocalStore(p)
    s.add_document(Document(url="https://a.com/x", title="T", text="persist me"))
    s.close()
    s2 = LocalStore(p)
    assert s2.count() == 1
    assert s2.search("persist")
    s2.close()


def test_recrawl_metadata(store):
    store.add_document(Document(url="https://a.com/1", text="v1"))
    store.update_recrawl("https://a.com/1", changed=False)
    doc = store.get_document_by_url("https://a.com/1")
    assert doc is not None
    store.update_recrawl("https://a.com/1", changed=True, etag="abc")
    doc2 = store.get_document_by_url("https://a.com/1")
    assert doc2.etag == "abc"


def test_optimize_runs(seeded_store):
    seeded_store.optimize()


def test_iter_for_shard(seeded_store):
    all_ids = set()
    for shard in range(3):
        for d in seeded_store.iter_for_shard(shard, 3):
            assert d.doc_id % 3 == shard
            all_ids.add(d.doc_id)
    assert len(all_ids) == seeded_store.count()
