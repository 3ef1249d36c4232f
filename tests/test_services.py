"""Service-layer tests: AppContext wiring, index/crawl paths, engine
orchestration (CPU shard), runtime coordination, governor."""
from __future__ import annotations

import asyncio

import httpx
import pytest

from infomesh_amd.config import Config
from infomesh_amd.engine import HybridEngine
from infomesh_amd.errors import InfoMeshError
from infomesh_amd.index.local_store import Document
from infomesh_amd.runtime import (GracefulShutdown, PidFile, RuntimeStatus,
                                  StartupLock)
from infomesh_amd.services import AppContext
from infomesh_amd.utils.governor import (DegradeLevel, ResourceGovernor,
                                         run_preflight_checks)


@pytest.fixture
def ctx():
    c = AppContext.create(config=Config(), with_engine=False,
                          with_worker=True, in_memory=True)
    yield c
    c.close()


@pytest.fixture
def ctx_engine():
    c = AppContext.create(config=Config(), with_engine=True,
                          with_worker=False, in_memory=True)
    # CPU builds use the CpuShard-backed engine without encoder
    c.engine = HybridEngine(device="cpu", use_encoder=False)
    yield c
    c.close()


def _doc(i=1, text=None):
    return Document(url=f"https://a.com/{i}", title=f"Doc {i}",
                    text=text or f"content body number {i} about gpu kernels")


def test_index_document_flow(ctx):
    rid = ctx.index_document(_doc(1))
    assert rid is not None
    assert ctx.store.count() == 1
    assert len(ctx.attestations) == 1
    assert ctx.ledger.balance() > 0


def test_index_respects_gdpr_and_dmca(ctx):
    ctx.deletions.request_deletion("https://a.com/gone")
    with pytest.raises(InfoMeshError):
        ctx.index_document(Document(url="https://a.com/gone", text="x" * 60))
    ctx.takedowns.file_notice("domain:evil.com", "c", "claimant")
    with pytest.raises(InfoMeshError):
        ctx.index_document(Document(url="https://evil.com/p", text="y" * 60))


def test_search_modes_and_cache(ctx):
    for i in range(5):
        ctx.index_document(_doc(i, text=f"python tutorial part {i} "
                                        "teaches functions"))
    r1 = ctx.search("python tutorial")
    assert r1.results
    bal = ctx.ledger.balance()
    r2 = ctx.search("python tutorial")   # cached -> no extra deduction
    assert ctx.ledger.balance() == bal
    assert r2 is r1


def test_search_explicit_local_mode(ctx):
    ctx.index_document(_doc(7, text="asyncio event loop coroutines"))
    resp = ctx.search("asyncio coroutines", mode="local", use_cache=False)
    assert resp.mode == "local"


def test_crawl_and_index(ctx):
    html = ("<html><head><title>T</title></head><body><p>" +
            "Crawled content paragraph that is long enough to index. " * 3 +
            "</p></body></html>")

    def handler(request):
        return httpx.Response(200, text=html,
                              headers={"content-type": "text/html"})
    ctx.worker._client = httpx.AsyncClient(
        transport=httpx.MockTransport(handler))
    ctx.worker._own_client = True
    ctx.worker.resolve_dns = False
    ctx.worker.cfg = ctx.worker.cfg.__class__(respect_robots=False)

    async def run():
        out = await ctx.crawl_and_index("https://site.com/page")
        assert out["status"] == "ok"
        assert out["indexed"]
    asyncio.run(run())
    assert ctx.store.count() == 1


def test_engine_ingest_and_search(ctx_engine):
    for i in range(20):
        ctx_engine.index_document(Document(
            url=f"https://b.com/{i}", title=f"GPU doc {i}",
            text=f"document about rocm hip kernels iteration {i} "
                 f"with matrix cores and lds tiling"))
    assert ctx_engine.engine.pending_count == 20
    n = ctx_engine.flush_engine()
    assert n == 20
    assert ctx_engine.engine.shard.n_docs == 20
    resp = ctx_engine.search("rocm hip kernels", mode="hybrid",
                             use_cache=False)
    assert resp.mode in ("hybrid",)
    assert resp.results
    hits = ctx_engine.engine.search("matrix cores lds", limit=5)
    assert hits and all(h.doc_id >= 1 for h in hits)


def test_status_surface(ctx):
    st = ctx.status()
    assert st["node_id"]
    assert st["index"]["documents"] == 0
    assert "credits" in st and "cache" in st


# ------------------------------------------------------------- runtime

def test_pidfile(tmp_path):
    pf = PidFile(tmp_path, marker="python")
    pf.acquire()
    assert pf.read_running_pid() is not None
    pf.acquire()  # same-process re-acquire is fine
    pf.release()
    assert pf.read_running_pid() is None
    # stale pid (dead process) is cleaned up on read
    pf.path.write_text("999999")
    assert pf.read_running_pid() is None
    pf.acquire()
    pf.release()
    # a live UNRELATED process (pid 1) fails the cmdline marker check
    pf.path.write_text("1")
    assert pf.read_running_pid() is None


def test_startup_lock(tmp_path):
    with StartupLock(tmp_path):
        l2 = StartupLock(tmp_path)
        assert not l2.acquire()
    l3 = StartupLock(tmp_path)
    assert l3.acquire()
    l3.release()


def test_runtime_status_heartbeat(tmp_path):
    rs = RuntimeStatus(tmp_path)
    rs.write("running", docs=5)
    data = rs.read()
    assert data["state"] == "running" and data["docs"] == 5
    import json, time
    stale = json.loads(rs.path.read_text())
    stale["ts"] = time.time() - 100
    rs.path.write_text(json.dumps(stale))
    assert rs.read()["state"] == "stopped"


def test_graceful_shutdown_callbacks():
    gs = GracefulShutdown()
    hit = []
    gs.on_shutdown(lambda: hit.append(1))
    gs._handler(15, None)
    assert gs.requested and hit == [1]


# ------------------------------------------------------------- governor

def test_governor_levels():
    g = ResourceGovernor(max_rss_gb=10_000, min_mem_available_gb=0.001,
                         max_load_per_cpu=10_000)
    g.sample(force=True)
    assert g.level == DegradeLevel.NORMAL
    assert g.crawl_allowed() and g.writes_allowed()
    g2 = ResourceGovernor(max_load_per_cpu=-1.0, max_rss_gb=10_000,
                          min_mem_available_gb=0.001)
    g2.sample(force=True)
    assert g2.level == DegradeLevel.THROTTLE_CRAWL
    assert g2.throttle_factor() == 3.0


def test_preflight(tmp_path):
    problems = run_preflight_checks(tmp_path)
    assert problems == [] or all("disk" not in p for p in problems)


def test_engine_flush_is_incremental_on_encoding():
    """A second flush must encode ONLY the new docs (embedding reuse),
    and search must still see docs from both flushes."""
    import torch
    from infomesh_amd.engine import HybridEngine
    from infomesh_amd.index.local_store import Document

    class StubEncoder:
        def __init__(self):
            self.rows = 0

        def encode_texts(self, texts):
            self.rows += len(texts)
            g = torch.Generator().manual_seed(hash(tuple(texts)) & 0xFFFF)
            return torch.nn.functional.normalize(
                torch.randn(len(texts), 384, generator=g), dim=-1)

    eng = HybridEngine(device="cpu", use_encoder=False)
    eng.encoder = StubEncoder()
    for i in range(3):
        eng.add_document(Document(url=f"u{i}", title=f"alpha doc{i}",
                                  text="alpha beta", doc_id=i))
    eng.flush(embed_batch=4)
    first = eng.encoder.rows
    assert eng.shard.n_docs == 3
    for i in range(3, 5):
        eng.add_document(Document(url=f"u{i}", title=f"gamma doc{i}",
                                  text="gamma delta", doc_id=i))
    eng.flush(embed_batch=4)
    # second flush encoded one padded batch (4 rows), NOT all 5 docs
    assert eng.encoder.rows - first == 4
    assert eng.shard.n_docs == 5
    assert eng.shard.embeddings.shape[0] == 5
    hits_old = eng.search("alpha", limit=5)
    hits_new = eng.search("gamma", limit=5)
    assert hits_old and hits_new


def test_plugin_hooks_fire_in_real_paths(tmp_path, monkeypatch):
    """pre/post index + search hooks run inside AppContext flows."""
    from infomesh_amd.index.local_store import Document
    from infomesh_amd.services import AppContext
    from infomesh_amd.utils.plugins import GLOBAL_PLUGINS
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path))
    seen = {"pre_index": 0, "post_index": 0,
            "pre_search": 0, "post_search": 0}

    def pre_index(doc, **k):
        seen["pre_index"] += 1
        doc.title = doc.title + " [plug]"
        return doc

    def post_index(doc, **k):
        seen["post_index"] += 1

    def pre_search(q, **k):
        seen["pre_search"] += 1
        return q

    def post_search(resp, **k):
        seen["post_search"] += 1
        return resp

    GLOBAL_PLUGINS.register("pre_index", pre_index)
    GLOBAL_PLUGINS.register("post_index", post_index)
    GLOBAL_PLUGINS.register("pre_search", pre_search)
    GLOBAL_PLUGINS.register("post_search", post_search)
    try:
        ctx = AppContext.create(with_engine=False, with_worker=False)
        try:
            ctx.index_document(Document(url="https://pl.org/1",
                                        title="hooked",
                                        text="plugin hook body " * 10))
            doc = ctx.store.get_document_by_url("https://pl.org/1")
            assert doc.title.endswith("[plug]")   # pre_index transformed
            ctx.search("plugin")
            assert seen["pre_index"] == seen["post_index"] == 1
            assert seen["pre_search"] >= 1 and seen["post_search"] >= 1
        finally:
            ctx.close()
    finally:
        GLOBAL_PLUGINS._hooks.clear()


def test_engine_hbm_budget_guard():
    """gpu.hbm_budget_gb refuses a flush that would overflow HBM,
    leaving pendings intact (degrade-before-OOM, reference governor
    RSS-limit analogue)."""
    import numpy as np
    import pytest as _pt
    from infomesh_amd.engine import HybridEngine
    from infomesh_amd.errors import InfoMeshError
    from infomesh_amd.index.local_store import Document
    eng = HybridEngine(device="cpu", use_encoder=False,
                       hbm_budget_gb=1e-6)   # ~1 KB budget
    d = Document(url="https://x/1", title="T",
                 text="budget guard test body " * 50)
    d.doc_id = 1
    eng.add_document(d)
    with _pt.raises(InfoMeshError):
        eng.flush()
    assert eng.pending_count == 1          # nothing lost
    eng.hbm_budget_bytes = 10**9
    assert eng.flush() == 1                # retry succeeds


def test_config_knobs_reach_behavior(tmp_path):
    """Every documented config knob must change runtime behavior:
    credits.* → ledger math, trust.isolation_failures → TrustStore,
    index.max_text_chars → LocalStore truncation, trust.audits_per_hour
    → auditor rate (round-2 dead-knob audit)."""
    import dataclasses as dc
    cfg = Config()
    cfg = dc.replace(
        cfg,
        credits=dc.replace(cfg.credits, crawl_reward=2.5, search_cost=0.2,
                           grace_hours=1.0),
        trust=dc.replace(cfg.trust, isolation_failures=2,
                         audits_per_hour=7.0, auditors=5),
        index=dc.replace(cfg.index, max_text_chars=50))
    ctx = AppContext.create(config=cfg, with_worker=False, with_engine=False,
                            in_memory=True)
    try:
        from infomesh_amd.credits.ledger import Action
        e = ctx.ledger.record_action(Action.CRAWL, 4)
        assert e.credits == 2.5 * 4
        assert abs(ctx.ledger.search_cost() - 0.2) < 1e-9  # tier 1 override
        assert ctx.ledger._grace_hours == 1.0
        assert ctx.trust.isolation_failures == 2
        # 2 failed audits isolate under the tightened threshold
        ctx.trust.record_audit("nodeX", False)
        ctx.trust.record_audit("nodeX", False)
        assert ctx.trust.tier("nodeX") == "isolated"
        # oversized text is truncated at ingest
        from infomesh_amd.index.local_store import Document
        did = ctx.store.add_document(Document(url="http://x/1", title="t",
                                              text="a" * 500))
        assert len(ctx.store.get_document(did).text) == 50

        async def fake_fetch(url):
            return None
        aud = ctx.make_auditor(fake_fetch)
        assert aud.rate == 7.0 and aud.auditors == 5
        assert aud.due(now=aud.last_audit + 3600.0 / 7.0 + 1)
    finally:
        ctx.close()


def test_snapshot_compression_level_param(tmp_path):
    from infomesh_amd.index.local_store import Document, LocalStore
    from infomesh_amd.index.snapshot import export_snapshot, import_snapshot

    store = LocalStore(":memory:")
    store.add_document(Document(url="http://x/1", title="t", text="hello " * 200))
    p = tmp_path / "s.infomesh-snapshot"
    export_snapshot(store, p, level=1)
    dst = LocalStore(":memory:")
    info = import_snapshot(dst, p)
    assert info["imported"] == 1 or dst.count() == 1


def test_require_extension_fails_loudly(monkeypatch):
    """gpu.require_extension: a GPU engine must refuse to start when the
    HIP extension is missing — never a silent eager fallback (the
    round-end 'native code not loaded' check)."""
    from infomesh_amd.errors import GpuExtensionMissing
    from infomesh_amd.ops import _ext

    monkeypatch.setattr(_ext, "available", lambda: False)
    with pytest.raises(GpuExtensionMissing):
        HybridEngine(device="cuda", require_extension=True)


def test_create_fails_loudly_when_extension_required_and_missing(monkeypatch):
    """AppContext.create must NOT swallow GpuExtensionMissing into a
    CPU-only degrade — the knob exists to make that loud."""
    import torch

    from infomesh_amd.ops import _ext

    monkeypatch.setattr(torch.cuda, "is_available", lambda: True)
    monkeypatch.setattr(_ext, "available", lambda: False)
    from infomesh_amd.errors import GpuExtensionMissing
    with pytest.raises(GpuExtensionMissing):
        AppContext.create(config=Config(), with_worker=False,
                          with_engine=True, in_memory=True)


def test_ingest_defers_signatures_but_serves_them_signed():
    """index_document must not pay per-page Ed25519 on the hot path:
    attestations are stored unsigned and signed on first serve; crawl
    credits accrue via the batched ledger but are visible in balance()
    immediately."""
    from infomesh_amd.trust.attestation import verify_attestation

    ctx = AppContext.create(config=Config(), with_worker=False,
                            with_engine=False, in_memory=True)
    try:
        for i in range(5):
            ctx.index_document(Document(url=f"http://d/{i}", title="t",
                                        text=f"body {i} unique"))
        assert len(ctx.attestations) == 5
        assert all(not a.signature for a in ctx.attestations)  # deferred
        assert ctx.ledger.balance() > 0                        # pending-aware
        served = ctx.signed_attestations(3)
        assert len(served) == 3
        assert all(a.signature and verify_attestation(a) for a in served)
        n = ctx.ledger.flush_pending()
        assert n >= 1 and ctx.ledger.verify_chain()
    finally:
        ctx.close()


def test_feedback_boost_influences_local_ranking():
    """Recorded implicit feedback must actually move local ranking
    (reference applies FeedbackStore boosts in ranking)."""
    ctx = AppContext.create(config=Config(), with_worker=False,
                            with_engine=False, in_memory=True)
    try:
        for i in range(2):
            ctx.index_document(Document(
                url=f"http://b/{i}", title="same terms here",
                text=f"identical body words rocm hip {'pad' * i}"))
        base = ctx.search("identical body words", limit=2, mode="local",
                          use_cache=False, deduct=False)
        assert len(base.results) == 2
        loser = base.results[-1].url
        for _ in range(40):            # strong positive signal
            ctx.feedback.record(loser, "cite")
        ctx.feedback._boost_cache.clear()
        ctx.cache.invalidate()
        boosted = ctx.search("identical body words", limit=2,
                             mode="local", use_cache=False, deduct=False)
        assert boosted.results[0].url == loser
    finally:
        ctx.close()
