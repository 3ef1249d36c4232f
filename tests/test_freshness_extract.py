"""Round-2 capability closures: priority recrawl queue, content
extraction, cross-validation, OTLP export, starter release flow."""
from __future__ import annotations

import numpy as np
import pytest
import torch

from infomesh_amd.crawler.content_extract import (extract_code_blocks,
                                                  extract_tables)
from infomesh_amd.crawler.freshness import (FreshnessTier,
                                            PriorityRecrawlQueue,
                                            RecrawlTrigger,
                                            classify_freshness)
from infomesh_amd.search.cross_validate import (SourceResult,
                                                cross_validate_results,
                                                snippet_similarity,
                                                validate_shard_hits)


# ------------------------------------------------------------- freshness

def test_classify_freshness_tiers():
    assert classify_freshness(3600, 0) is FreshnessTier.HOT
    assert classify_freshness(2 * 86400, 0) is FreshnessTier.WARM
    assert classify_freshness(14 * 86400, 0) is FreshnessTier.COLD
    assert classify_freshness(90 * 86400, 0) is FreshnessTier.STALE
    assert classify_freshness(None, 3600) is FreshnessTier.HOT
    assert classify_freshness(None, 60 * 86400) is FreshnessTier.STALE


def test_priority_queue_ordering_and_dedup():
    q = PriorityRecrawlQueue()
    assert q.enqueue("https://a/1", RecrawlTrigger.SCHEDULED)
    assert q.enqueue("https://a/2", RecrawlTrigger.RSS_UPDATE, source="f")
    assert q.enqueue("https://a/3", RecrawlTrigger.USER_REQUEST)
    assert not q.enqueue("https://a/1", RecrawlTrigger.USER_REQUEST)  # dup
    got = [q.dequeue().url for _ in range(3)]
    assert got == ["https://a/3", "https://a/2", "https://a/1"]
    assert q.dequeue() is None
    st = q.stats()
    assert st["enqueued"] == 3 and st["dequeued"] == 3


def test_priority_queue_eviction_at_cap():
    q = PriorityRecrawlQueue(max_size=2)
    q.enqueue("https://s/1", RecrawlTrigger.SCHEDULED)
    q.enqueue("https://s/2", RecrawlTrigger.SCHEDULED)
    # full of SCHEDULED: another SCHEDULED is refused...
    assert not q.enqueue("https://s/3", RecrawlTrigger.SCHEDULED)
    # ...but a USER_REQUEST evicts the worst
    assert q.enqueue("https://u/1", RecrawlTrigger.USER_REQUEST)
    urls = {q.dequeue().url, q.dequeue().url}
    assert "https://u/1" in urls and len(q) == 0


def test_priority_queue_discard():
    q = PriorityRecrawlQueue()
    q.enqueue("https://a", RecrawlTrigger.SCHEDULED)
    q.enqueue("https://b", RecrawlTrigger.SCHEDULED)
    q.discard("https://a")
    assert q.dequeue().url == "https://b" and q.dequeue() is None


# -------------------------------------------------------------- extract

HTML = """
<h1>Doc</h1>
<pre><code class="language-python">def f():
    return 1</code></pre>
<pre><code>plain block</code></pre>
<table>
  <tr><th>name</th><th>count</th></tr>
  <tr><td>alpha, one</td><td>1</td></tr>
  <tr><td>beta</td><td>2</td></tr>
</table>
Markdown too:
```rust
fn main() {}
```
"""


def test_extract_code_blocks():
    blocks = extract_code_blocks(HTML)
    assert len(blocks) == 3
    assert blocks[0].language == "python" and "def f" in blocks[0].code
    assert blocks[1].language == ""
    assert blocks[2].language == "rust" and "fn main" in blocks[2].code


def test_extract_tables_csv_and_dicts():
    tables = extract_tables(HTML)
    assert len(tables) == 1
    t = tables[0]
    assert t.headers == ("name", "count")
    assert t.rows == (("alpha, one", "1"), ("beta", "2"))
    csv = t.to_csv()
    assert csv.splitlines()[0] == "name,count"
    assert '"alpha, one"' in csv
    assert t.to_dict_list()[1] == {"name": "beta", "count": "2"}


# ------------------------------------------------------- cross-validate

def test_cross_validate_flags_score_deviation():
    src = {
        "engine": [SourceResult("https://x", "X", "alpha beta gamma", 1.0),
                   SourceResult("https://y", "Y", "delta words", 0.9)],
        "fts": [SourceResult("https://x", "X", "alpha beta gamma here", 1.1),
                SourceResult("https://y", "Y", "delta words too", 9.0)],
    }
    rep = cross_validate_results(src)
    verdicts = {r.url: r.verdict for r in rep.results}
    assert verdicts["https://x"] == "verified"
    assert verdicts["https://y"] == "suspicious"   # 0.9 vs 9.0
    assert rep.n_suspicious == 1


def test_cross_validate_single_source_unverified():
    rep = cross_validate_results(
        {"engine": [SourceResult("https://only", score=1.0)]})
    assert rep.results[0].verdict == "unverified"


def test_snippet_similarity():
    assert snippet_similarity("a b c", "a b c") == 1.0
    assert snippet_similarity("a b", "c d") == 0.0
    assert 0 < snippet_similarity("alpha beta", "alpha gamma") < 1


def test_validate_shard_hits_flags_outlier_shard():
    B, W, k = 4, 4, 8
    scores = torch.rand(B, W * k) + 1.0
    scores[:, 2 * k:3 * k] *= 50.0   # shard 2 wildly off
    out = validate_shard_hits(None, None, scores, world=W)
    assert out["suspicious_shards"] == [2]


# ------------------------------------------------------------------ OTLP

def test_otlp_export_shape_and_batching():
    import time
    from infomesh_amd.utils.observability import (OtlpExporter, QueryTrace)
    tr = QueryTrace("hip kernels")
    with tr.span("encode"):
        time.sleep(0.001)
    with tr.span("shard"):
        pass
    posted = []
    exp = OtlpExporter(endpoint="http://collector:4318", batch_size=2,
                       post_fn=lambda url, payload: posted.append(
                           (url, payload)))
    exp.export(tr)
    assert not posted            # below batch size
    exp.export(tr)
    assert len(posted) == 2      # flushed
    url, payload = posted[0]
    assert url.endswith("/v1/traces")
    spans = payload["resourceSpans"][0]["scopeSpans"][0]["spans"]
    names = [s["name"] for s in spans]
    assert names[0] == "search" and "encode" in names and "shard" in names
    root = spans[0]
    for s in spans[1:]:
        assert s["parentSpanId"] == root["spanId"]
        assert s["traceId"] == root["traceId"]
        assert int(s["endTimeUnixNano"]) >= int(s["startTimeUnixNano"])
    disabled = OtlpExporter(endpoint="")
    disabled.export(tr)   # no-op, no error
    assert disabled.exported == 0


# ------------------------------------------------- starter release flow

def test_fetch_release_starter_with_mock_transport(tmp_path):
    import httpx
    from infomesh_amd.index.local_store import Document, LocalStore
    from infomesh_amd.index.snapshot import export_snapshot
    from infomesh_amd.index.starter import fetch_release_starter

    # build a real snapshot to serve as the release asset
    src_store = LocalStore(":memory:")
    for i in range(5):
        src_store.add_document(Document(
            url=f"https://seed/{i}", title=f"Seed {i}",
            text=f"starter document {i} about hip kernels"))
    snap = tmp_path / "starter.infomesh-snapshot"
    export_snapshot(src_store, snap)
    blob = snap.read_bytes()

    def handler(request: httpx.Request) -> httpx.Response:
        if "releases/latest" in str(request.url):
            return httpx.Response(200, json={"assets": [
                {"name": "community.infomesh-snapshot",
                 "size": len(blob),
                 "browser_download_url": "https://dl.test/x.snap"}]})
        return httpx.Response(200, content=blob,
                              headers={"content-length": str(len(blob))})

    client = httpx.Client(transport=httpx.MockTransport(handler))
    dst_store = LocalStore(":memory:")
    seen = []
    res = fetch_release_starter(dst_store, tmp_path / "dl",
                                progress=lambda d, t: seen.append((d, t)),
                                client=client)
    assert res is not None and res["imported"] == 5
    assert res["release_asset"] == "community.infomesh-snapshot"
    assert dst_store.count() == 5
    assert seen and seen[-1][0] == len(blob)


# -------------------------------------------------------- dashboard tabs

def test_dashboard_tabs_render_headless(tmp_path, monkeypatch):
    """All six tabs render against a real data dir (daemon-less reads,
    reference TUI parity: 6 screens)."""
    from rich.console import Console
    from infomesh_amd.config import Config
    from infomesh_amd.dashboard.tabs import (RENDERERS, TABS, History,
                                             TabbedData, next_tab,
                                             render_tabbed)
    from infomesh_amd.index.local_store import Document
    from infomesh_amd.services import AppContext

    cfg = Config()
    object.__setattr__(cfg.node, "data_dir", str(tmp_path)) \
        if hasattr(cfg.node, "data_dir") else None
    monkeypatch.setattr(Config, "data_dir",
                        property(lambda self: tmp_path), raising=False)
    ctx = AppContext.create(config=cfg, with_engine=False,
                            with_worker=False)
    for i in range(3):
        ctx.index_document(Document(url=f"https://t/{i}", title=f"T{i}",
                                    text=f"doc {i} text body"),
                           attest=False, credit=True)
    ctx.ledger.flush_pending()
    ctx.close()

    data = TabbedData(cfg)
    hist = History()
    console = Console(width=100, record=True, file=__import__("io").StringIO())
    assert set(RENDERERS) == set(TABS)
    for tab in TABS:
        console.print(render_tabbed(data, hist, tab))
    out = console.export_text()
    assert "overview" in out and "settings" in out
    assert "https://t/" in out          # crawl tab recent docs
    assert "crawl" in out and "credits" in out


def test_dashboard_tab_switching():
    from infomesh_amd.dashboard.tabs import TABS, next_tab
    assert next_tab("overview", "3") == TABS[2]
    assert next_tab("overview", "l") == "crawl"
    assert next_tab("crawl", "h") == "overview"
    assert next_tab("overview", "h") == TABS[-1]   # wraps
    assert next_tab("overview", "q") is None
    assert next_tab("overview", "x") == "overview"


# --------------------------------------------------- sync-debug proxy

def test_sync_debug_lib_proxy(monkeypatch):
    """INFOMESH_SYNC_DEBUG=1 wraps every kernel entry with a
    synchronize-and-raise check (sanitizer-style launch validation)."""
    import importlib
    from infomesh_amd.ops import _ext as ext
    monkeypatch.setenv("INFOMESH_SYNC_DEBUG", "1")
    mod = importlib.reload(ext)
    try:
        if not mod.available():
            pytest.skip("extension not built in this environment")
        wrapped = mod.lib()
        assert isinstance(wrapped, mod._SyncDebugLib)
        fn = wrapped.infomesh_topk_workspace_u32
        assert callable(fn)
        assert fn(4) > 0   # passthrough result survives wrapping
    finally:
        monkeypatch.delenv("INFOMESH_SYNC_DEBUG")
        importlib.reload(mod)


def test_otlp_wired_into_search(monkeypatch):
    """With api.otlp_endpoint configured, every engine search exports a
    trace to the collector (batched in the exporter)."""
    import dataclasses
    from infomesh_amd.config import Config
    from infomesh_amd.index.local_store import Document
    from infomesh_amd.services import AppContext
    cfg = Config(api=dataclasses.replace(Config().api,
                                         otlp_endpoint="http://c:4318"))
    ctx = AppContext.create(config=cfg, with_engine=True,
                            with_worker=False, in_memory=True)
    assert ctx.otlp is not None and ctx.otlp.enabled
    posted = []
    ctx.otlp._post = lambda url, payload: posted.append(url)
    ctx.otlp.batch_size = 1
    ctx.index_document(Document(url="https://t/1", title="T",
                                text="otlp traced search body"),
                       attest=False, credit=False)
    ctx.flush_engine()
    ctx.search("traced search", use_cache=False, deduct=False)
    assert posted and posted[0].endswith("/v1/traces")
    ctx.close()
