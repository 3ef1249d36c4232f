"""Full-lifecycle integration: crawl (mock HTTP) -> index -> engine
flush -> search via MCP -> snapshot export -> import on a second node ->
search parity + credits/trust side effects."""
from __future__ import annotations

import asyncio
import json

import httpx
import pytest

from infomesh_amd.services import AppContext

PAGES = {
    "https://site-a.org/gpu": """<html><head><title>GPU kernels guide</title>
      </head><body><p>Writing MFMA kernels for matrix cores requires
      tiling through LDS and careful scheduling of memory.</p>
      <a href="https://site-a.org/lds">LDS deep dive</a></body></html>""",
    "https://site-a.org/lds": """<html><head><title>LDS deep dive</title>
      </head><body><p>The local data share has banks; padding avoids
      conflicts in kernels. Stage tiles through shared memory.</p>
      </body></html>""",
    "https://site-b.net/other": """<html><head><title>Cooking pasta</title>
      </head><body><p>Boil water, add salt, cook the pasta until al
      dente, then drain and serve with sauce and cheese.</p></body></html>""",
}


def _mock_client():
    def handler(request: httpx.Request) -> httpx.Response:
        url = str(request.url)
        if url.endswith("robots.txt"):
            return httpx.Response(404, text="")
        page = PAGES.get(url)
        if page is None:
            return httpx.Response(404, text="nope")
        return httpx.Response(200, text=page,
                              headers={"content-type": "text/html",
                                       "etag": f'W/"{hash(url) & 0xffff}"'})
    return httpx.AsyncClient(transport=httpx.MockTransport(handler))


@pytest.fixture
def node(tmp_path, monkeypatch):
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path / "n1"))
    monkeypatch.setenv("INFOMESH_CRAWL_RESPECT_ROBOTS", "true")
    ctx = AppContext.create()
    # mock network + no DNS resolution in tests
    ctx.worker._client = _mock_client()
    ctx.worker._own_client = True
    ctx.worker.resolve_dns = False
    yield ctx
    ctx.close()


def test_full_lifecycle(node, tmp_path, monkeypatch):
    ctx = node
    # 1. crawl three pages (one link chain + one unrelated)
    for url in PAGES:
        out = asyncio.run(ctx.crawl_and_index(url))
        assert out["status"] == "ok" and out["indexed"], out
    assert ctx.store.count() == 3
    # credits accrued for crawling
    assert ctx.ledger.balance() > 0
    # attestations were created
    assert len(ctx.attestations) == 3

    # 2. engine flush -> searchable on the (CPU) shard
    if ctx.engine is not None:
        assert ctx.engine.pending_count == 3
        ctx.flush_engine()
        assert ctx.engine.shard.n_docs == 3

    # 3. search via MCP: topical query hits the right site
    from infomesh_amd.mcp.server import McpServer
    srv = McpServer(ctx)
    r = srv.handle_message({"jsonrpc": "2.0", "id": 1,
                            "method": "tools/call",
                            "params": {"name": "web_search",
                                       "arguments": {"query": "lds kernels",
                                                     "limit": 3}}})
    hits = json.loads(r["result"]["content"][0]["text"])["results"]
    assert hits and all("site-a.org" in h["url"] for h in hits[:2])

    # 4. snapshot export -> import into a SECOND node
    snap = tmp_path / "x.infomesh-snapshot"
    from infomesh_amd.index.snapshot import export_snapshot, import_snapshot
    export_snapshot(ctx.store, snap, node_name="n1")
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path / "n2"))
    ctx2 = AppContext.create(with_worker=False)
    try:
        res = import_snapshot(ctx2.store, snap)
        assert res["imported"] == 3 and res["skipped"] == 0
        # idempotent re-import
        res2 = import_snapshot(ctx2.store, snap)
        assert res2["imported"] == 0 and res2["skipped"] == 3
        # search parity on the second node (FTS plane)
        a = [h.url for h in ctx.store.search("pasta", limit=2)]
        b = [h.url for h in ctx2.store.search("pasta", limit=2)]
        assert a == b == ["https://site-b.net/other"]
    finally:
        ctx2.close()

    # 5. re-crawl is conditional: second crawl of same URL is deduped
    out = asyncio.run(ctx.crawl_and_index("https://site-a.org/gpu"))
    assert out["status"] in ("skipped", "ok")


def test_lifecycle_compliance_enforcement(node):
    """DMCA takedown + GDPR deletion remove content and block
    re-indexing through the real index path."""
    ctx = node
    for url in PAGES:
        asyncio.run(ctx.crawl_and_index(url))
    assert ctx.store.count() == 3

    # DMCA: takedown site-b -> applied -> gone and blocked
    ctx.takedowns.file_notice("domain:site-b.net", "copyright claim",
                              claimant="rights-holder@example.com")
    removed = ctx.takedowns.apply_pending()
    assert removed >= 1
    assert ctx.store.get_document_by_url("https://site-b.net/other") is None
    from infomesh_amd.errors import InfoMeshError
    from infomesh_amd.index.local_store import Document
    with pytest.raises(InfoMeshError):
        ctx.index_document(Document(url="https://site-b.net/back",
                                    title="t", text="returns " * 20))

    # GDPR: forget one site-a URL -> enforced -> blocked at re-crawl
    ctx.deletions.request_deletion("https://site-a.org/lds",
                                   reason="user request")
    ctx.deletions.enforce()
    assert ctx.store.get_document_by_url("https://site-a.org/lds") is None
    # re-crawling the forgotten URL is refused LOUDLY at index time
    # (the background crawl loop isolates this per-URL)
    with pytest.raises(InfoMeshError):
        asyncio.run(ctx.crawl_and_index("https://site-a.org/lds",
                                        force=True))
    assert ctx.store.get_document_by_url("https://site-a.org/lds") is None
    # the other site-a page is untouched
    assert ctx.store.get_document_by_url("https://site-a.org/gpu")
