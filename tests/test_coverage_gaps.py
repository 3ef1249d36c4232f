"""Direct tests for modules only covered indirectly elsewhere."""
from __future__ import annotations

import time


def test_link_graph_authority(tmp_path):
    from infomesh_amd.index.link_graph import LinkGraph
    lg = LinkGraph(tmp_path / "links.db")
    # hub.org links to a.io and b.io; everyone links to popular.com
    lg.add_links("https://hub.org/", ["https://a.io/x",
                                      "https://popular.com/"])
    lg.add_links("https://a.io/x", ["https://popular.com/"])
    lg.add_links("https://b.io/y", ["https://popular.com/"])
    pop = lg.url_authority("https://popular.com/anything")
    a = lg.url_authority("https://a.io/other")
    assert 0.0 <= a <= 1.0 and 0.0 <= pop <= 1.0
    assert pop > a  # 3 inbound domains beat 1
    assert lg.url_authority("https://unknown.zz/") <= a
    lg.close()


def test_session_store_ttl_and_gc():
    from infomesh_amd.mcp.session import SessionStore
    t = [1000.0]
    st = SessionStore(ttl_s=10.0, max_sessions=3)
    st._now = lambda: t[0]  # type: ignore[attr-defined]
    s1 = st.create()
    assert st.get(s1.session_id) is not None
    assert st.get("nope") is None
    for _ in range(5):
        st.create()
    assert st.count() <= 3  # capped


def test_analytics_tracker_report():
    from infomesh_amd.mcp.session import AnalyticsTracker
    a = AnalyticsTracker()
    for ms in (5.0, 10.0, 15.0):
        a.record("web_search", ms)
    a.record("crawl_url", 100.0, error=True)
    rep = a.report()
    ws = rep["web_search"]
    assert ws["calls"] == 3 and 5.0 <= ws["avg_ms"] <= 15.0
    assert rep["crawl_url"]["errors"] == 1


def test_webhook_registry_fire():
    from infomesh_amd.mcp.session import WebhookRegistry
    reg = WebhookRegistry()
    got = []
    reg.register("crawl", lambda payload: got.append(payload))
    n = reg.fire("crawl", {"url": "https://x"})
    assert n == 1 and got == [{"url": "https://x"}]
    # a raising hook is isolated
    reg.register("crawl", lambda payload: 1 / 0)
    n = reg.fire("crawl", {"url": "https://y"})
    assert len(got) == 2


def test_text_report_renders(tmp_path, monkeypatch):
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path))
    from infomesh_amd.services import AppContext
    from infomesh_amd.utils.text_report import render_report
    ctx = AppContext.create(with_engine=False, with_worker=False)
    try:
        out = render_report(ctx)
        assert "documents" in out.lower() or "docs" in out.lower()
        assert len(out) > 100
    finally:
        ctx.close()


def test_graphed_callable_cpu_eager():
    import torch
    from infomesh_amd.ops.graphs import GraphedCallable
    calls = []

    def fn(x):
        calls.append(1)
        return x * 2

    g = GraphedCallable(fn)
    # no CUDA here -> always eager, but transparent
    out = g(torch.tensor([1.0, 2.0]))
    assert torch.equal(out, torch.tensor([2.0, 4.0]))
    assert len(calls) == 1
