"""Tests: commoncrawl WET import, starter, structured extraction,
content diff/WARC, shard manifest, version check."""
from __future__ import annotations

import gzip
import io

import numpy as np
import torch

from infomesh_amd.crawler.diff import (diff_content, significant_change,
                                       warc_export)
from infomesh_amd.crawler.structured import (extract_json_ld,
                                             extract_open_graph,
                                             extract_structured)
from infomesh_amd.index.commoncrawl import CommonCrawlImporter, parse_wet
from infomesh_amd.index.gpu_index import CpuShard
from infomesh_amd.index.local_store import LocalStore
from infomesh_amd.index.manifest import (load_shard, manifest_info,
                                         save_shard)
from infomesh_amd.index.snapshot import export_snapshot
from infomesh_amd.index.starter import (find_local_starters, load_starter,
                                        needs_starter)
from infomesh_amd.index.synth import synth_corpus_arrays
from infomesh_amd.utils.version_check import (check_for_update, is_newer,
                                              parse_version)


def _wet_bytes(records):
    out = []
    for url, text in records:
        payload = text.encode()
        out.append(
            f"WARC/1.0\r\nWARC-Type: conversion\r\n"
            f"WARC-Target-URI: {url}\r\n"
            f"Content-Length: {len(payload)}\r\n\r\n".encode() + payload +
            b"\r\n\r\n")
    return b"".join(out)


def test_parse_wet_and_import(tmp_path):
    body = ("This is a long enough plain text document about GPU "
            "kernels and search engines. " * 4)
    raw = _wet_bytes([("https://a.com/1", body),
                      ("https://b.com/2", "too short"),
                      ("https://c.com/3", body + "different")])
    recs = list(parse_wet(io.BytesIO(raw)))
    assert len(recs) == 2
    p = tmp_path / "test.wet.gz"
    with gzip.open(p, "wb") as f:
        f.write(raw)
    store = LocalStore(":memory:")
    res = CommonCrawlImporter(store).import_wet(p)
    assert res["imported"] == 2
    assert store.search("kernels")
    store.close()


def test_starter_flow(tmp_path, seeded_store):
    assert needs_starter(seeded_store)  # < 100 docs
    snap_dir = tmp_path / "starters"
    snap_dir.mkdir()
    export_snapshot(seeded_store, snap_dir / "community.infomesh-snapshot")
    assert len(find_local_starters([snap_dir])) == 1
    dst = LocalStore(":memory:")
    res = load_starter(dst, [snap_dir])
    assert res["imported"] == seeded_store.count()
    dst.close()


HTML = """<html><head>
<script type="application/ld+json">{"@type":"Article","name":"X"}</script>
<meta property="og:title" content="OG Title">
<meta property="og:type" content="article">
</head><body>
<pre>def kernel():\n    return mfma_tile()</pre>
<table><tr><th>a</th><th>b</th></tr><tr><td>1</td><td>2</td></tr></table>
</body></html>"""


def test_structured_extraction():
    sd = extract_structured(HTML)
    assert sd.json_ld[0]["@type"] == "Article"
    assert sd.open_graph["title"] == "OG Title"
    assert any("mfma_tile" in c for c in sd.code_blocks)
    assert sd.tables[0] == [["a", "b"], ["1", "2"]]
    assert extract_json_ld("<html>no data</html>") == []
    assert extract_open_graph("<html/>") == {}


def test_diff_and_warc(tmp_path):
    old = "line one\nline two\nline three"
    new = "line one\nline 2 changed\nline three\nline four"
    d = diff_content(old, new)
    assert d.added_lines == 2 and d.removed_lines == 1
    assert significant_change(old, new, threshold=0.05)
    assert not significant_change(old, old)
    p = tmp_path / "out.warc"
    n = warc_export(p, [{"url": "https://a.com", "content": "body"}])
    assert n == 1 and b"WARC-Target-URI: https://a.com" in p.read_bytes()


def test_shard_manifest_roundtrip(tmp_path):
    terms, docs, lens = synth_corpus_arrays(300, avg_len=20, seed=9)
    emb = torch.nn.functional.normalize(torch.randn(300, 16), dim=-1)\
        .bfloat16()
    shard = CpuShard()
    shard.build_from_arrays(terms, docs, lens,
                            np.arange(300, dtype=np.int64), emb)
    hits_before = shard.search([np.array([1, 2, 3])], None, k=5)
    p = tmp_path / "shard0.pt"
    meta = save_shard(shard, p, rank=0, world=1)
    assert meta["n_docs"] == 300
    assert manifest_info(p)["n_docs"] == 300
    loaded = load_shard(p, device="cpu")
    hits_after = loaded.search([np.array([1, 2, 3])], None, k=5)
    assert torch.equal(hits_before.bm25_ids, hits_after.bm25_ids)
    assert torch.allclose(hits_before.bm25_scores, hits_after.bm25_scores)


def test_shard_manifest_then_incremental_append(tmp_path):
    """Warm-start then ingest more docs: the round-1 ADVICE bug was
    that doc_lens was not persisted, so any build() after load_shard
    crashed. Now append-after-restore must match a from-scratch build."""
    rng = np.random.default_rng(11)
    docs = [rng.integers(0, 400, size=rng.integers(5, 30)).astype(np.int64)
            for _ in range(200)]
    shard = CpuShard()
    for i in range(150):
        shard.add_document(i, docs[i], None)
    shard.build()
    p = tmp_path / "warm.pt"
    save_shard(shard, p)
    loaded = load_shard(p, device="cpu")
    for i in range(150, 200):
        loaded.add_document(i, docs[i], None)
    loaded.build()   # crashed with AttributeError before the fix
    assert loaded.n_docs == 200

    oracle = CpuShard()
    for i in range(200):
        oracle.add_document(i, docs[i], None)
    oracle.build()
    q = [np.array([7, 42, 99]), np.array([300])]
    hl, ho = loaded.search(q, None, k=10), oracle.search(q, None, k=10)
    assert torch.equal(hl.bm25_ids, ho.bm25_ids)
    assert torch.allclose(hl.bm25_scores, ho.bm25_scores, atol=1e-5)


def test_version_check(tmp_path):
    assert parse_version("1.2.3") == (1, 2, 3)
    assert is_newer("9.9.9")
    assert not is_newer("0.0.1")
    out = check_for_update(state_path=tmp_path / "vc.json")
    assert out["current"] and not out["update_available"]
    # cached second call
    out2 = check_for_update(state_path=tmp_path / "vc.json")
    assert out2 == out


def test_dashboard_sparkline_and_history(tmp_path, monkeypatch):
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path))
    from infomesh_amd.dashboard.app import (
        DashboardData, History, render_dashboard, sparkline)
    assert sparkline([]) == ""
    sp = sparkline([0, 1, 2, 3, 4, 5, 6, 7])
    assert len(sp) == 8 and sp[0] != sp[-1]
    assert sparkline([5.0] * 4) == sparkline([5.0] * 4)  # flat ok
    h = History(maxlen=5)
    for i in range(10):
        h.push("x", i)
    assert h.get("x") == [5.0, 6.0, 7.0, 8.0, 9.0]
    data = DashboardData()
    layout = render_dashboard(data, h)
    assert layout is not None
    # two renders with history grow sparkline series without error
    render_dashboard(data, h)


def test_reputation_syncs_into_trust_score():
    """Summary-quality grades must reach the composite trust score
    (the 0.20-weighted component, reference scoring)."""
    from infomesh_amd.trust.reputation import SummaryReputation
    from infomesh_amd.trust.scoring import TrustStore

    rep = SummaryReputation()
    trust = TrustStore(":memory:")
    for _ in range(6):
        rep.record("good-node", 0.95)
        rep.record("bad-node", 0.05)
    assert rep.sync_to_trust(trust) == 2
    g = trust.score("good-node")
    b = trust.score("bad-node")
    assert g > b
    trust.close()


def test_detector_critical_verdict_isolates_immediately():
    """A CRITICAL detector verdict isolates the domain even before the
    consecutive-failure ladder would."""
    import asyncio

    from infomesh_amd.trust.audit import AuditScheduler
    from infomesh_amd.trust.detector import MaliciousNodeDetector
    from infomesh_amd.trust.scoring import TrustStore
    from infomesh_amd.index.local_store import Document, LocalStore

    store = LocalStore(":memory:")
    store.add_document(Document(url="http://evil.example/p", title="t",
                                text="page body content here"))
    trust = TrustStore(":memory:", isolation_failures=99)  # ladder off
    det = MaliciousNodeDetector()
    det.record("evil.example", "invalid_signature", 3)  # pre-escalated
    det.record("evil.example", "invalid_proof", 3)

    async def fetch(url):
        return "DIFFERENT content — audit mismatch"

    sched = AuditScheduler(store, trust, fetch, auditors=3,
                           detector=det)
    res = asyncio.run(sched.run_audit())
    assert res is not None and not res.passed
    assert trust.tier("evil.example") == "isolated"
    store.close(); trust.close()
