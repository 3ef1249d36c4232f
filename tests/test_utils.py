"""Cross-cutting utils tests: observability, slo, plugins, scalability,
data_quality, security_ops, diagnostics, benchmarks, crawl loop."""
from __future__ import annotations

import asyncio

import httpx
import pytest

from infomesh_amd.utils.benchmarks import run_micro_suite, time_fn
from infomesh_amd.utils.data_quality import (cross_reference,
                                             format_citation,
                                             grade_document)
from infomesh_amd.utils.diagnostics import run_doctor
from infomesh_amd.utils.observability import (MetricsRegistry, QueryTrace,
                                              grafana_dashboard_json)
from infomesh_amd.utils.plugins import GLOBAL_PLUGINS, PluginManager
from infomesh_amd.utils.scalability import (BloomFilter, RoundRobinPool,
                                            batch_ingest)
from infomesh_amd.utils.security_ops import ApiKeyManager, AuditLogger
from infomesh_amd.utils.slo import SLOTracker


def test_metrics_registry_render():
    m = MetricsRegistry()
    m.inc("requests_total", labels={"path": "/x"})
    m.inc("requests_total", labels={"path": "/x"})
    m.set_gauge("docs", 42)
    m.observe("latency_seconds", 0.02)
    text = m.render()
    assert 'requests_total{path="/x"} 2.0' in text
    assert "docs 42" in text
    assert "latency_seconds_count 1" in text
    assert 'le="0.05"' in text


def test_query_trace_spans():
    t = QueryTrace("q")
    with t.span("encode"):
        pass
    with t.span("score"):
        pass
    rep = t.report()
    assert set(rep["spans"]) == {"encode", "score"}
    assert grafana_dashboard_json()["panels"]


def test_slo_tracker():
    s = SLOTracker()
    s.define("search", target_p95_ms=100, target_success_rate=0.9)
    for _ in range(20):
        s.record("search", 10.0, ok=True)
    assert s.report()["search"]["met"]
    for _ in range(80):
        s.record("search", 500.0, ok=False)
    assert not s.report()["search"]["met"]


def test_plugins():
    pm = PluginManager()

    @pm.hook("pre_search")
    def lower(q):
        return q.lower()

    @pm.hook("pre_search")
    def strip_x(q):
        return q.replace("x", "")

    assert pm.run("pre_search", "XQueryX") == "query"
    with pytest.raises(ValueError):
        pm.register("bogus", lambda v: v)
    # failing plugin is isolated
    pm.register("post_search", lambda v: 1 / 0)
    assert pm.run("post_search", "ok") == "ok"
    assert GLOBAL_PLUGINS.count() >= 0


def test_bloom_filter():
    bf = BloomFilter(capacity=1000, error_rate=0.01)
    for i in range(500):
        bf.add(f"url-{i}")
    assert all(f"url-{i}" in bf for i in range(500))
    false_pos = sum(1 for i in range(1000, 3000) if f"url-{i}" in bf)
    assert false_pos < 60  # ~1% target
    assert 0 < bf.fill_ratio() < 1


def test_batch_ingest_and_pool():
    batches = []
    total = batch_ingest(range(25), lambda b: batches.append(len(b)) or len(b),
                         batch_size=10)
    assert total == 25 and batches == [10, 10, 5]
    pool = RoundRobinPool(lambda: object(), size=2)
    a, b, c = pool.get(), pool.get(), pool.get()
    assert a is c and a is not b


def test_data_quality():
    import time
    g = grade_document(time.time(), trust=0.9, text_len=3000, has_title=True)
    assert g.grade in ("A", "B")
    g2 = grade_document(time.time() - 90 * 86400, trust=0.1, text_len=50,
                        has_title=False)
    assert g2.grade in ("D", "F")
    assert "Retrieved" in format_citation("https://a.com", "T", time.time())
    ref = cross_reference({"a.com": 0.8, "b.com": 0.7, "c.com": 0.1})
    assert ref["verdict"] == "corroborated"


def test_api_key_manager():
    km = ApiKeyManager()
    key = km.create_key("ci")
    assert km.verify(key)
    assert not km.verify("imk_wrong")
    assert not km.verify("")
    km.revoke("ci")
    assert not km.verify(key)
    assert km.list_keys()[0]["revoked"] == 1
    km.close()


def test_audit_logger(tmp_path):
    log = AuditLogger(tmp_path / "audit.jsonl")
    log.log("search", q="x")
    log.log("delete", url="https://a.com")
    assert log.verify()
    # tamper
    lines = (tmp_path / "audit.jsonl").read_text().splitlines()
    lines[0] = lines[0].replace('"search"', '"SEARCH"')
    (tmp_path / "audit.jsonl").write_text("\n".join(lines) + "\n")
    log2 = AuditLogger(tmp_path / "audit.jsonl")
    assert not log2.verify()


def test_doctor_runs(tmp_data_dir):
    report = run_doctor()
    names = {c["name"] for c in report["checks"]}
    assert {"python", "torch", "sqlite fts5", "zstd"} <= names
    assert all(c["ok"] for c in report["checks"]
               if c["name"] in ("python", "sqlite fts5", "zstd"))


def test_micro_bench_suite():
    out = run_micro_suite(iterations=5)
    assert "query_expansion" in out and out["query_expansion"]["ops_per_sec"] > 0
    t = time_fn(lambda: None, iterations=10, warmup=1)
    assert t["avg_ms"] >= 0


def test_crawl_loop_bounded(tmp_data_dir):
    """Crawl loop with a mock transport: seeds -> crawl -> index."""
    from infomesh_amd.config import Config, CrawlConfig
    from infomesh_amd.crawler.crawl_loop import seed_and_crawl_loop
    from infomesh_amd.services import AppContext
    import dataclasses

    html = ("<html><head><title>Seed</title></head><body><p>" +
            "Seed page content that is long enough to index properly. " * 4 +
            "</p></body></html>")

    def handler(request):
        return httpx.Response(200, text=html,
                              headers={"content-type": "text/html"})

    cfg = dataclasses.replace(Config(), crawl=CrawlConfig(
        politeness_delay_s=0, respect_robots=False, max_urls_per_hour=1000))
    ctx = AppContext.create(config=cfg, with_engine=False, with_worker=True,
                            in_memory=True)
    ctx.worker._client = httpx.AsyncClient(
        transport=httpx.MockTransport(handler))
    ctx.worker.resolve_dns = False

    stats = asyncio.run(seed_and_crawl_loop(ctx, max_iterations=6))
    assert stats["crawled"] >= 1
    assert ctx.store.count() >= 1  # dedup collapses identical bodies
    ctx.close()


def test_dashboard_renders(tmp_data_dir):
    """Live dashboard renders against a populated data dir (WAL reads)."""
    from infomesh_amd.config import Config
    from infomesh_amd.dashboard.app import DashboardData, render_dashboard
    from infomesh_amd.runtime import RuntimeStatus
    from infomesh_amd.services import AppContext
    from infomesh_amd.index.local_store import Document
    from rich.console import Console
    import io

    ctx = AppContext.create(config=Config(), with_engine=False,
                            with_worker=False)
    ctx.index_document(Document(url="https://a.com/1",
                                text="dashboard test doc body " * 5))
    RuntimeStatus(ctx.config.data_dir).write("running", engine_docs=1)
    ctx.close()

    data = DashboardData()
    snap = data.snapshot()
    assert snap["docs"] == 1
    assert snap["runtime"]["state"] == "running"
    console = Console(file=io.StringIO(), width=100)
    console.print(render_dashboard(data))
    out = console.file.getvalue()
    assert "infomesh-amd" in out and "documents" in out


def test_adaptive_crawl_tuner():
    from infomesh_amd.crawler.intelligence import AdaptiveCrawlTuner
    t = AdaptiveCrawlTuner(base_delay_s=1.0)
    assert t.delay_for("a.com") == 1.0
    t.record("a.com", ok=False, status=429)
    assert t.delay_for("a.com") == 4.0
    for _ in range(10):
        t.record("a.com", ok=True, latency_ms=100)
    assert t.delay_for("a.com") < 4.0
    assert "a.com" in t.stats()


def test_js_render_surface():
    from infomesh_amd.crawler import js_render
    if not js_render.available():
        import pytest
        with pytest.raises(RuntimeError):
            js_render.render("https://example.com")


def test_cli_shard_roundtrip(tmp_data_dir):
    from click.testing import CliRunner
    from infomesh_amd.cli import cli
    import json as _json
    runner = CliRunner()
    # index a doc, save the (CPU) shard, reload and probe it
    from infomesh_amd.services import AppContext
    from infomesh_amd.index.local_store import Document
    ctx = AppContext.create(with_engine=False, with_worker=False)
    ctx.index_document(Document(url="https://a.com/1",
                                text="manifest roundtrip body " * 5),
                       attest=False, credit=False)
    ctx.close()
    # monkey: CLI builds its own ctx; engine falls back to CpuShard on CPU
    r = runner.invoke(cli, ["shard", "save", str(tmp_data_dir / "s.pt")])
    assert r.exit_code == 0, r.output
    assert _json.loads(r.output)["n_docs"] == 1
    r2 = runner.invoke(cli, ["shard", "info", str(tmp_data_dir / "s.pt")])
    assert _json.loads(r2.output)["n_docs"] == 1
    r3 = runner.invoke(cli, ["shard", "load", str(tmp_data_dir / "s.pt"),
                             "--query", "manifest roundtrip"])
    assert r3.exit_code == 0, r3.output
    assert _json.loads(r3.output)["probe"]


def test_cli_update_check(tmp_data_dir):
    from click.testing import CliRunner
    from infomesh_amd.cli import cli
    import json as _json
    r = CliRunner().invoke(cli, ["update-check"])
    assert r.exit_code == 0
    assert "current" in _json.loads(r.output)


# --------------------------------------------------------- security_ext


def test_security_ext_rbac_and_ipfilter():
    from infomesh_amd.utils.security_ext import IpFilter, role_allows
    assert role_allows("admin", "config")
    assert role_allows("reader", "search")
    assert not role_allows("reader", "config")
    assert not role_allows("ghost", "search")
    f = IpFilter(allow=["127.0.0.0/8", "10.0.0.0/8"], deny=["10.1.0.0/16"])
    assert f.permitted("127.0.0.1")
    assert f.permitted("10.2.3.4")
    assert not f.permitted("10.1.2.3")     # deny wins
    assert not f.permitted("8.8.8.8")      # not in allow
    assert not f.permitted("not-an-ip")
    assert IpFilter().permitted("8.8.8.8")  # default allow


def test_security_ext_tokens():
    from infomesh_amd.utils.security_ext import issue_token, verify_token
    sec = b"s3cret"
    t = [1000.0]
    tok = issue_token(sec, "alice", "operator", ttl_s=60,
                      now=lambda: t[0])
    data = verify_token(sec, tok, now=lambda: t[0])
    assert data and data["sub"] == "alice" and data["role"] == "operator"
    assert verify_token(b"wrong", tok, now=lambda: t[0]) is None
    assert verify_token(sec, tok + "x", now=lambda: t[0]) is None
    t[0] += 61
    assert verify_token(sec, tok, now=lambda: t[0]) is None


def test_security_ext_webhook_hmac():
    from infomesh_amd.utils.security_ext import (
        sign_webhook, verify_webhook)
    sec, body = b"whsec", b'{"event":"crawl"}'
    h = sign_webhook(sec, body)
    assert h.startswith("sha256=")
    assert verify_webhook(sec, body, h)
    assert not verify_webhook(sec, b"tampered", h)
    assert not verify_webhook(sec, body, "")


def test_plugin_module_loading(tmp_path, monkeypatch):
    import sys
    from infomesh_amd.utils.plugins import (
        PluginManager, load_plugin_module, load_plugins_from_config)
    plug_dir = tmp_path / "plugs"
    plug_dir.mkdir()
    (plug_dir / "myplug.py").write_text(
        "def setup(plugins):\n"
        "    plugins.register('pre_search', lambda q, **k: q.upper())\n")
    (plug_dir / "broken.py").write_text("raise RuntimeError('boom')\n")
    monkeypatch.syspath_prepend(str(plug_dir))
    mgr = PluginManager()
    n = load_plugin_module("myplug", mgr)
    assert n == 1
    assert mgr.run("pre_search", "hello") == "HELLO"
    # broken plugin isolated, good one still counted
    mgr2 = PluginManager()
    total = load_plugins_from_config(["myplug", "broken", "ghost"], mgr2)
    assert total == 1
    sys.modules.pop("myplug", None)
    sys.modules.pop("broken", None)


def test_tokenizer_and_scorer_plugin_slots():
    import numpy as np
    from infomesh_amd.index.gpu_index import bm25_term_ids
    from infomesh_amd.index.ranking import ScoreBreakdown
    from infomesh_amd.utils.plugins import GLOBAL_PLUGINS
    try:
        default = bm25_term_ids("hello world")
        GLOBAL_PLUGINS.register("tokenizer", lambda t: ["onlytoken"])
        custom = bm25_term_ids("hello world")
        assert len(custom) == 1 and not np.array_equal(default, custom)
        GLOBAL_PLUGINS.register("scorer", lambda sb: 42.0)
        sb = ScoreBreakdown(bm25=1.0, freshness=1.0, trust=1.0,
                            authority=1.0, title_match=1.0, url_path=1.0)
        assert sb.total == 42.0
    finally:
        GLOBAL_PLUGINS._hooks.clear()
    assert ScoreBreakdown(bm25=1.0, freshness=0, trust=0, authority=0,
                          title_match=0, url_path=0).total > 0


def test_crawl_loop_drains_priority_queue_first(tmp_data_dir):
    """URLs enqueued with triggers (RSS/user) are crawled ahead of the
    BFS scheduler (reference freshness.py:67-212 behavior)."""
    import dataclasses
    from infomesh_amd.config import Config, CrawlConfig
    from infomesh_amd.crawler.crawl_loop import seed_and_crawl_loop
    from infomesh_amd.crawler.freshness import (PriorityRecrawlQueue,
                                                RecrawlTrigger)
    from infomesh_amd.services import AppContext

    crawled: list[str] = []

    def handler(request):
        crawled.append(str(request.url))
        return httpx.Response(
            200, text="<html><head><title>P</title></head><body><p>" +
                      "priority queue crawl body text. " * 6 +
                      "</p></body></html>",
            headers={"content-type": "text/html"})

    cfg = dataclasses.replace(Config(), crawl=CrawlConfig(
        politeness_delay_s=0, respect_robots=False,
        max_urls_per_hour=1000))
    ctx = AppContext.create(config=cfg, with_engine=False,
                            with_worker=True, in_memory=True)
    ctx.worker._client = httpx.AsyncClient(
        transport=httpx.MockTransport(handler))
    ctx.worker.resolve_dns = False
    pq = ctx.recrawl_queue = PriorityRecrawlQueue()
    pq.enqueue("https://feed.example/fresh-item",
               RecrawlTrigger.RSS_UPDATE, source="https://feed.example/rss")
    pq.enqueue("https://user.example/forced", RecrawlTrigger.USER_REQUEST)

    stats = asyncio.run(seed_and_crawl_loop(ctx, max_iterations=3))
    assert stats["crawled"] >= 2
    # priority items beat the seed list, USER_REQUEST first
    assert crawled[0].startswith("https://user.example/")
    assert crawled[1].startswith("https://feed.example/")
    assert len(pq) == 0 and pq.total_dequeued == 2
    ctx.close()
