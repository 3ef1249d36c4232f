"""Entry-point tests: MCP server (direct JSON-RPC), admin API
(TestClient), CLI (CliRunner), SDK (MockTransport), integrations.
Reference parity: tests/test_mcp*.py, test_local_api.py,
test_cli_search.py.
"""
from __future__ import annotations

import json

import httpx
import pytest
from click.testing import CliRunner
from fastapi.testclient import TestClient

from infomesh_amd.api.local_api import create_app
from infomesh_amd.cli import cli
from infomesh_amd.config import Config
from infomesh_amd.index.local_store import Document
from infomesh_amd.integrations import (InfoMeshDocumentStore,
                                       InfoMeshReader, InfoMeshRetriever)
from infomesh_amd.mcp.server import McpServer
from infomesh_amd.mcp.tools import resolve_tool
from infomesh_amd.sdk.client import InfoMeshClient
from infomesh_amd.services import AppContext


@pytest.fixture
def ctx():
    c = AppContext.create(config=Config(), with_engine=False,
                          with_worker=False, in_memory=True)
    for i in range(6):
        c.index_document(Document(
            url=f"https://a.com/{i}", title=f"Python doc {i}",
            text=f"python tutorial part {i} covering functions and classes "
                 f"with plenty of example code to read"),
            attest=False, credit=False)
    yield c
    c.close()


# ------------------------------------------------------------------- MCP

def _rpc(server, method, params=None, mid=1):
    return server.handle_message({"jsonrpc": "2.0", "id": mid,
                                  "method": method, "params": params or {}})


def test_mcp_initialize_and_list(ctx):
    s = McpServer(ctx)
    init = _rpc(s, "initialize")
    assert init["result"]["serverInfo"]["name"] == "infomesh-amd"
    tools = _rpc(s, "tools/list")["result"]["tools"]
    assert {t["name"] for t in tools} == \
        {"web_search", "fetch_page", "crawl_url", "fact_check", "status"}


def test_mcp_web_search_call(ctx):
    s = McpServer(ctx)
    resp = _rpc(s, "tools/call", {"name": "web_search",
                                  "arguments": {"query": "python tutorial"}})
    payload = json.loads(resp["result"]["content"][0]["text"])
    assert payload["results"]
    assert payload["results"][0]["url"].startswith("https://a.com/")
    assert not resp["result"]["isError"]


def test_mcp_legacy_aliases(ctx):
    s = McpServer(ctx)
    assert resolve_tool("search") == "web_search"
    assert resolve_tool("nope") is None
    resp = _rpc(s, "tools/call", {"name": "explain",
                                  "arguments": {"query": "python"}})
    payload = json.loads(resp["result"]["content"][0]["text"])
    assert payload["mode"] == "explain"


def test_mcp_rag_modes(ctx):
    s = McpServer(ctx)
    resp = _rpc(s, "tools/call", {
        "name": "web_search",
        "arguments": {"query": "python functions", "chunk_size": 256,
                      "answer_mode": True}})
    payload = json.loads(resp["result"]["content"][0]["text"])
    assert "chunks" in payload and "answer" in payload


def test_mcp_fetch_and_status(ctx):
    s = McpServer(ctx)
    resp = _rpc(s, "tools/call", {"name": "fetch_page",
                                  "arguments": {"url": "https://a.com/1"}})
    payload = json.loads(resp["result"]["content"][0]["text"])
    assert payload["found"] and "python tutorial" in payload["text"]
    st = json.loads(_rpc(s, "tools/call", {"name": "status"})
                    ["result"]["content"][0]["text"])
    assert st["index"]["documents"] == 6


def test_mcp_fact_check(ctx):
    s = McpServer(ctx)
    resp = _rpc(s, "tools/call", {
        "name": "fact_check",
        "arguments": {"claim": "python tutorial covers functions"}})
    payload = json.loads(resp["result"]["content"][0]["text"])
    assert payload["supported"]


def test_mcp_unknown_method(ctx):
    s = McpServer(ctx)
    resp = _rpc(s, "bogus/method")
    assert "error" in resp


def test_mcp_stdio_roundtrip(ctx):
    import io
    s = McpServer(ctx)
    stdin = io.StringIO(json.dumps(
        {"jsonrpc": "2.0", "id": 5, "method": "ping"}) + "\n")
    stdout = io.StringIO()
    s.run_stdio(stdin, stdout)
    out = json.loads(stdout.getvalue())
    assert out["id"] == 5


def test_mcp_analytics(ctx):
    s = McpServer(ctx)
    _rpc(s, "tools/call", {"name": "status"})
    rep = s.analytics.report()
    assert rep["status"]["calls"] == 1


# ------------------------------------------------------------- admin API

@pytest.fixture
def api_client(ctx):
    app = create_app(ctx)
    return TestClient(app)


def test_api_health_and_search(api_client):
    assert api_client.get("/health").json()["ok"]
    r = api_client.get("/search", params={"q": "python tutorial"})
    assert r.status_code == 200
    assert r.json()["results"]
    assert r.headers["X-Frame-Options"] == "DENY"


def test_api_status_config_stats(api_client):
    assert api_client.get("/status").json()["index"]["documents"] == 6
    assert "crawl" in api_client.get("/config").json()
    assert api_client.get("/index/stats").json()["documents"] == 6
    assert "balance" in api_client.get("/credits/balance").json()
    assert api_client.get("/network/peers").json()["world_size"] >= 1


def test_api_metrics_prometheus(api_client):
    api_client.get("/search", params={"q": "python"})
    text = api_client.get("/metrics").text
    assert "api_requests_total" in text


def test_api_dashboard_text(api_client):
    text = api_client.get("/dashboard").text
    assert "infomesh-amd node report" in text


def test_api_key_enforced(ctx):
    app = create_app(ctx, api_key="sekrit")
    c = TestClient(app)
    assert c.get("/status").status_code == 401
    assert c.get("/status", headers={"x-api-key": "sekrit"}).status_code == 200
    assert c.get("/health").status_code == 200  # health exempt


def test_api_empty_query_400(api_client):
    assert api_client.get("/search", params={"q": "  "}).status_code == 400


# ------------------------------------------------------------------- SDK

def test_sdk_client_roundtrip(ctx):
    app = create_app(ctx)
    tc = TestClient(app)

    def forward(request: httpx.Request) -> httpx.Response:
        r = tc.request(request.method, request.url.raw_path.decode(),
                       headers=dict(request.headers))
        return httpx.Response(r.status_code, content=r.content,
                              headers=r.headers)

    client = InfoMeshClient(base_url="http://testserver",
                            transport=httpx.MockTransport(forward))
    results = client.search("python tutorial")
    assert results and results[0]["url"].startswith("https://a.com/")
    assert client.status()["index"]["documents"] == 6
    assert client.health()
    client.close()


def test_sdk_async_client(ctx):
    import asyncio
    from infomesh_amd.sdk.client import AsyncInfoMeshClient
    app = create_app(ctx)

    async def run():
        client = AsyncInfoMeshClient(
            base_url="http://testserver",
            transport=httpx.ASGITransport(app=app))
        results = await client.search("python tutorial")
        assert results
        st = await client.status()
        assert st["index"]["documents"] == 6
        await client.close()
    asyncio.run(run())


# ------------------------------------------------------------------- CLI

def test_cli_help_lists_commands():
    r = CliRunner().invoke(cli, ["--help"])
    assert r.exit_code == 0
    for cmd in ("start", "stop", "status", "search", "crawl", "mcp",
                "index", "config", "keys", "doctor", "bench"):
        assert cmd in r.output


def test_cli_search_and_index(tmp_data_dir):
    runner = CliRunner()
    # build an index via snapshot import
    from infomesh_amd.index.local_store import LocalStore
    from infomesh_amd.index.snapshot import export_snapshot
    src = LocalStore(":memory:")
    src.add_document(Document(url="https://x.com/1", title="CLI Doc",
                              text="command line interface search test "
                                   "body with enough words"))
    snap = tmp_data_dir / "x.infomesh-snapshot"
    export_snapshot(src, snap)
    src.close()
    r = runner.invoke(cli, ["index", "import", str(snap)])
    assert r.exit_code == 0, r.output
    assert json.loads(r.output)["imported"] == 1
    r2 = runner.invoke(cli, ["search", "command", "line", "--json"])
    assert r2.exit_code == 0, r2.output
    assert "x.com" in r2.output
    r3 = runner.invoke(cli, ["index", "stats"])
    assert json.loads(r3.output)["documents"] == 1


def test_cli_config_set(tmp_data_dir):
    runner = CliRunner()
    r = runner.invoke(cli, ["config", "set", "crawl.max_concurrent", "9"])
    assert r.exit_code == 0, r.output
    r2 = runner.invoke(cli, ["config", "show"])
    assert json.loads(r2.output)["crawl"]["max_concurrent"] == 9


def test_cli_keys_and_doctor(tmp_data_dir):
    runner = CliRunner()
    r = runner.invoke(cli, ["keys", "show"])
    assert r.exit_code == 0
    assert json.loads(r.output)["node_id"]
    r2 = runner.invoke(cli, ["doctor"])
    assert "python" in r2.output


def test_cli_dashboard(tmp_data_dir):
    r = CliRunner().invoke(cli, ["dashboard", "--text"])
    assert r.exit_code == 0
    assert "node report" in r.output


# ----------------------------------------------------------- integrations

def test_retriever_and_reader_inprocess(ctx):
    retr = InfoMeshRetriever(ctx=ctx, k=3)
    docs = retr.get_relevant_documents("python tutorial")
    assert docs and docs[0].metadata["url"].startswith("https://a.com/")
    assert retr.invoke("python")  # langchain-style alias
    reader = InfoMeshReader(ctx=ctx)
    assert reader.load_data("python", limit=2)


def test_document_store(ctx):
    ds = InfoMeshDocumentStore(ctx)
    n = ds.write_documents([
        {"id": "https://h.com/1", "content": "haystack style document body "
                                             "with sufficient length here",
         "meta": {"title": "H1"}}])
    assert n == 1
    assert ds.count_documents() == 7
    assert ds.filter_documents("haystack")


def test_mcp_resources_and_prompts(tmp_path, monkeypatch):
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path))
    from infomesh_amd.mcp.server import McpServer
    ctx = AppContext.create(with_engine=False, with_worker=False)
    try:
        srv = McpServer(ctx)
        init = srv.handle_message({"jsonrpc": "2.0", "id": 1,
                                   "method": "initialize", "params": {}})
        caps = init["result"]["capabilities"]
        assert "resources" in caps and "prompts" in caps
        rl = srv.handle_message({"jsonrpc": "2.0", "id": 2,
                                 "method": "resources/list"})
        uris = [r["uri"] for r in rl["result"]["resources"]]
        assert "infomesh://index/stats" in uris
        rd = srv.handle_message({"jsonrpc": "2.0", "id": 3,
                                 "method": "resources/read",
                                 "params": {"uri": "infomesh://index/stats"}})
        import json as _json
        body = _json.loads(rd["result"]["contents"][0]["text"])
        assert "documents" in body or body  # stats dict present
        rt = srv.handle_message({"jsonrpc": "2.0", "id": 4,
                                 "method": "resources/templates/list"})
        tmpl = rt["result"]["resourceTemplates"]
        assert any(t["uriTemplate"] == "infomesh://doc/{url}"
                   for t in tmpl)
        pl = srv.handle_message({"jsonrpc": "2.0", "id": 4,
                                 "method": "prompts/list"})
        names = [p["name"] for p in pl["result"]["prompts"]]
        assert "research" in names
        assert all("template" not in p for p in pl["result"]["prompts"])
        pg = srv.handle_message({"jsonrpc": "2.0", "id": 5,
                                 "method": "prompts/get",
                                 "params": {"name": "research",
                                            "arguments": {"topic": "gpus"}}})
        txt = pg["result"]["messages"][0]["content"]["text"]
        assert "gpus" in txt
        err = srv.handle_message({"jsonrpc": "2.0", "id": 6,
                                  "method": "resources/read",
                                  "params": {"uri": "infomesh://nope"}})
        assert "error" in err
    finally:
        ctx.close()


def test_driver_contract_script():
    import subprocess, sys, pathlib
    root = pathlib.Path(__file__).resolve().parents[1]
    out = subprocess.run([sys.executable, "scripts/verify_contract.py"],
                         cwd=root, capture_output=True, text=True,
                         timeout=900)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "ALL OK" in out.stdout


def test_web_search_reference_style_args(tmp_path, monkeypatch):
    """Reference agents call web_search with top_k / domain lists /
    recency_days / answer_mode enum (infomesh mcp/tools.py:53-136) —
    all must be honored."""
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path))
    from infomesh_amd.index.local_store import Document
    ctx = AppContext.create(with_engine=False, with_worker=False)
    try:
        for i, dom in enumerate(["good.org", "good.org", "spam.net"]):
            ctx.index_document(Document(
                url=f"https://{dom}/p{i}", title=f"quantum doc {i}",
                text="quantum computing research " * 10))
        h = ctx.handlers if hasattr(ctx, "handlers") else None
        from infomesh_amd.mcp.handlers import Handlers
        h = Handlers(ctx)
        out = h.web_search(query="quantum", top_k=5,
                           domain_allowlist=["good.org"])
        assert out["results"]
        assert all("good.org" in r["url"] for r in out["results"])
        out = h.web_search(query="quantum", top_k=5,
                           domain_blocklist=["good.org"])
        assert all("spam.net" in r["url"] for r in out["results"])
        out = h.web_search(query="quantum", recency_days=30, top_k=5)
        assert out["results"]  # crawled just now -> within 30 days
        out = h.web_search(query="quantum", answer_mode="answer", top_k=3)
        assert "answer" in out and "chunks" in out
        out = h.web_search(query="quantum", local_only=True, top_k=2)
        assert len(out["results"]) <= 2
        out = h.web_search(query="quantum", fetch_full_content=True,
                           top_k=2)
        assert all("text" in r for r in out["results"])
    finally:
        ctx.close()


def test_mcp_legacy_utility_tools(tmp_path, monkeypatch):
    """The reference dispatches ping/credit_balance/index_stats/
    network_stats/batch_search/search_history/analytics/webhooks/
    remove_url (mcp/server.py:205-457) — all must resolve and work."""
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path))
    from infomesh_amd.index.local_store import Document
    from infomesh_amd.mcp.server import McpServer
    ctx = AppContext.create(with_engine=False, with_worker=False)
    try:
        ctx.index_document(Document(url="https://lg.org/1", title="t",
                                    text="legacy tools body " * 10))
        srv = McpServer(ctx)

        def call(name, args=None):
            r = srv.handle_message({"jsonrpc": "2.0", "id": 1,
                                    "method": "tools/call",
                                    "params": {"name": name,
                                               "arguments": args or {}}})
            import json as _json
            assert "result" in r, r
            return _json.loads(r["result"]["content"][0]["text"])

        assert call("ping")["pong"] is True
        assert "documents" in call("index_stats") or call("index_stats")
        assert isinstance(call("credit_balance"), dict)
        assert "world_size" in call("network_stats")
        out = call("batch_search", {"queries": ["legacy", "tools"],
                                    "limit": 3})
        assert len(out["batches"]) == 2
        call("search", {"query": "legacy"})      # populates history
        hist = call("search_history")
        assert any("legacy" in h for h in hist["history"])
        assert isinstance(call("analytics"), dict)
        assert call("register_webhook",
                    {"event": "crawl"})["registered"] == "crawl"
        assert call("unregister_webhook",
                    {"event": "crawl"})["unregistered"] == "crawl"
        rm = call("remove_url", {"url": "https://lg.org/1"})
        assert rm["removed"] and rm["deletion_recorded"]
        assert ctx.store.get_document_by_url("https://lg.org/1") is None
    finally:
        ctx.close()


def test_api_reference_routes(tmp_path, monkeypatch):
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path))
    from starlette.testclient import TestClient
    from infomesh_amd.api.local_api import create_app
    ctx = AppContext.create(with_engine=False, with_worker=False)
    try:
        app = create_app(ctx)
        with TestClient(app) as client:
            r = client.get("/analytics/tools")
            assert r.status_code == 200 and "tool_usage" in r.json()
            r = client.get("/index/compression")
            body = r.json()
            assert r.status_code == 200
            assert {"documents", "db_size_mb", "avg_doc_kb"} <= set(body)
            r = client.get("/openapi-spec")
            assert r.status_code == 200 and "openapi" in r.json()
    finally:
        ctx.close()


def test_feeds_cli_persistence(tmp_path, monkeypatch):
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path))
    from click.testing import CliRunner
    from infomesh_amd.cli import cli as root_cli
    runner = CliRunner()
    r = runner.invoke(root_cli, ["feeds", "add",
                                 "https://ex.org/feed.xml", "--tier", "1"])
    assert r.exit_code == 0, r.output
    r = runner.invoke(root_cli, ["feeds", "list"])
    assert "ex.org/feed.xml" in r.output
    # persists to a new context (fresh process equivalent)
    assert (tmp_path / "feeds.json").exists()
    opml = tmp_path / "feeds.opml"
    opml.write_text('<opml><body>'
                    '<outline type="rss" xmlUrl="https://a.io/rss"/>'
                    '<outline type="rss" xmlUrl="https://b.io/rss"/>'
                    '</body></opml>')
    r = runner.invoke(root_cli, ["feeds", "import", str(opml)])
    assert "imported 2" in r.output
    r = runner.invoke(root_cli, ["feeds", "remove",
                                 "https://ex.org/feed.xml"])
    assert r.exit_code == 0
    r = runner.invoke(root_cli, ["feeds", "list"])
    assert "ex.org" not in r.output and "a.io" in r.output


def test_manifest_tokenizer_version_guard(tmp_path):
    import numpy as np
    import pytest as _pytest
    from infomesh_amd.index.gpu_index import CpuShard, bm25_term_ids
    from infomesh_amd.index import manifest as M
    shard = CpuShard()
    shard.add_document(1, bm25_term_ids("hello world"), None)
    shard.build()
    path = tmp_path / "s.shard"
    meta = M.save_shard(shard, path)
    assert meta["tokenizer_version"] == M.TOKENIZER_VERSION
    s2 = M.load_shard(path, device="cpu")
    assert s2.n_docs == 1
    # a stale tokenizer version refuses to load (term-id mismatch);
    # the authoritative meta lives in the torch blob, not the sidecar
    import torch
    blob = torch.load(path, map_location="cpu", weights_only=False)
    blob["meta"]["tokenizer_version"] = 1
    torch.save(blob, path)
    with _pytest.raises(ValueError, match="tokenizer"):
        M.load_shard(path, device="cpu")


def test_mcp_http_transport(tmp_path, monkeypatch):
    """Streamable-HTTP transport: JSON-RPC over POST, api-key auth,
    health, and error paths."""
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path))
    from starlette.testclient import TestClient
    from infomesh_amd.mcp.server import McpServer
    ctx = AppContext.create(with_engine=False, with_worker=False)
    try:
        srv = McpServer(ctx, api_key="sekrit")
        client = TestClient(srv.asgi_app())
        # no key -> 401
        r = client.post("/mcp", json={"jsonrpc": "2.0", "id": 1,
                                      "method": "ping"})
        assert r.status_code == 401
        h = {"x-api-key": "sekrit"}
        r = client.post("/mcp", json={"jsonrpc": "2.0", "id": 1,
                                      "method": "ping"}, headers=h)
        assert r.status_code == 200 and r.json()["result"] == {}
        r = client.get("/health", headers=h)
        assert r.status_code == 200 and r.json()["ok"]
        r = client.post("/mcp", content=b"not json", headers=h)
        assert r.status_code == 400
        r = client.get("/nope", headers=h)
        assert r.status_code == 404
        r = client.post("/mcp", json={"jsonrpc": "2.0", "id": 2,
                                      "method": "tools/list"}, headers=h)
        assert len(r.json()["result"]["tools"]) == 5
    finally:
        ctx.close()


def test_mcp_suggest_tool(tmp_path, monkeypatch):
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path))
    from infomesh_amd.index.local_store import Document
    from infomesh_amd.mcp.server import McpServer
    ctx = AppContext.create(with_engine=False, with_worker=False)
    try:
        ctx.index_document(Document(url="https://s.org/1",
                                    title="Quantum computing primer",
                                    text="intro " * 20))
        srv = McpServer(ctx)
        r = srv.handle_message({"jsonrpc": "2.0", "id": 1,
                                "method": "tools/call",
                                "params": {"name": "suggest",
                                           "arguments": {"prefix": "Quan"}}})
        payload = json.loads(r["result"]["content"][0]["text"])
        assert payload["suggestions"] == ["Quantum computing primer"]
    finally:
        ctx.close()


def test_tool_argument_validation():
    """Tool calls are validated against their declared inputSchema
    (types, bounds, string/list caps) before dispatch."""
    from infomesh_amd.mcp.tools import validate_args
    assert validate_args("web_search", {"query": "ok", "limit": 5}) == []
    assert validate_args("web_search", {"query": ""})      # missing/empty
    assert validate_args("web_search", {"query": 42})      # wrong type
    assert validate_args("web_search", {"query": "x" * 20_000})
    assert validate_args("crawl_url", {"url": "https://a", "depth": 99})
    assert validate_args("crawl_url", {"url": "https://a", "depth": 1}) == []
    assert validate_args("batch_search",
                         {"queries": ["a"] * 2000})        # list cap


def test_handlers_reject_invalid_args(mcp_ctx=None):
    import pytest as _pt
    from infomesh_amd.config import Config
    from infomesh_amd.mcp.handlers import Handlers
    from infomesh_amd.services import AppContext
    ctx = AppContext.create(config=Config(), with_engine=False,
                            with_worker=False, in_memory=True)
    h = Handlers(ctx)
    out = None
    try:
        out = h.call("web_search", {"query": 42})
    except Exception as e:
        out = {"error": str(e)}
    assert out and "error" in out or "invalid" in str(out).lower()
    ok = h.call("status", {})
    assert "error" not in ok
    ctx.close()


def test_bench_multi_rank_cpu_contract(tmp_path):
    """The driver's N>1 launch shape (torch.distributed.run, gloo on
    CPU) must produce exactly one valid JSON contract line from rank 0
    — de-risks the round-end 8-GPU SCALE run."""
    import json as _json
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29515", str(root / "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--docs-per-gpu", "5000"],
        capture_output=True, text=True, timeout=240, cwd=root)
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [ln for ln in proc.stdout.splitlines()
             if ln.startswith("{") and '"metric"' in ln]
    assert len(lines) == 1, proc.stdout[-2000:]
    rec = _json.loads(lines[0])
    assert rec["n_gpus"] == 2 and rec["scaling"] == "weak"
    assert rec["config"]["parallelism"] == "shard2"
    assert rec["value"] > 0 and rec["ms_per_step"] > 0


def test_session_and_analytics_thread_safety():
    """SessionStore/AnalyticsTracker are hammered by the 256-thread
    handler pool: concurrent create/touch/gc/record must never corrupt
    state or raise (dict-changed-during-iteration class of bug)."""
    import concurrent.futures as cf
    import random

    from infomesh_amd.mcp.session import AnalyticsTracker, SessionStore

    store = SessionStore(ttl_s=0.05, max_sessions=50)
    tracker = AnalyticsTracker(max_samples=64)
    rng = random.Random(3)
    errs = []

    def worker(i):
        try:
            for j in range(200):
                r = rng.random()
                if r < 0.4:
                    store.create()
                elif r < 0.8:
                    store.touch(f"sess-{j % 20}")
                else:
                    store.get(f"sess-{j % 20}")
                tracker.record(f"tool{j % 5}", r * 10.0,
                               error=(j % 17 == 0))
        except Exception as e:   # pragma: no cover - the assertion
            errs.append(repr(e))

    with cf.ThreadPoolExecutor(max_workers=32) as pool:
        list(pool.map(worker, range(32)))
    assert not errs, errs[:3]
    assert store.count() <= 50 + 32   # bounded (gc races are benign)
    rep = tracker.report()
    assert rep and all(v["calls"] > 0 for v in rep.values())


def test_cli_new_commands_smoke(tmp_path):
    """Round-2 CLI parity additions: keys export, config github,
    feedback record/stats/top-urls, index import-wet all run clean."""
    from click.testing import CliRunner

    from infomesh_amd.cli import cli

    r = CliRunner()
    env = {"INFOMESH_NODE_DATA_DIR": str(tmp_path)}
    out = r.invoke(cli, ["keys", "export"], env=env)
    assert out.exit_code == 0 and "public_key" in out.output
    out = r.invoke(cli, ["config", "github", "--email", "o@example.com"],
                   env=env)
    assert out.exit_code == 0 and "o@example.com" in out.output
    assert r.invoke(cli, ["feedback", "record", "http://x/9"],
                    env=env).exit_code == 0
    out = r.invoke(cli, ["feedback", "stats"], env=env)
    assert out.exit_code == 0 and "fetch" in out.output
    out = r.invoke(cli, ["feedback", "top-urls"], env=env)
    assert out.exit_code == 0 and "http://x/9" in out.output
    wet = tmp_path / "c.wet"
    wet.write_text("not a wet file")
    out = r.invoke(cli, ["index", "import-wet", str(wet)], env=env)
    assert out.exit_code == 0   # total: imports 0, never crashes


def test_examples_quickstart_runs():
    """The shipped quickstart example must keep working end to end."""
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    out = subprocess.run([sys.executable, str(root / "examples" /
                                              "quickstart.py")],
                         capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr[-800:]
    assert "restored docs: 3" in out.stdout


def test_examples_serve_and_query_runs():
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    out = subprocess.run([sys.executable, str(root / "examples" /
                                              "serve_and_query.py")],
                         capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr[-800:]
    assert "health: True" in out.stdout


def test_examples_mcp_stdio_runs():
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    out = subprocess.run([sys.executable, str(root / "examples" /
                                              "mcp_stdio_client.py")],
                         capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr[-800:]
    assert "web_search" in out.stdout
    assert "mentions MFMA: True" in out.stdout


def test_examples_rag_with_adapters_runs():
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    out = subprocess.run([sys.executable, str(root / "examples" /
                                              "rag_with_adapters.py")],
                         capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr[-800:]
    assert "retrieved:" in out.stdout and "chunks:" in out.stdout


def test_admin_api_endpoint_surface():
    """Every documented admin endpoint answers (reference parity:
    local_api route table)."""
    import httpx

    from infomesh_amd.api.local_api import create_app

    ctx = AppContext.create(config=Config(), with_worker=False,
                            with_engine=False, in_memory=True)
    try:
        ctx.index_document(Document(url="http://a/1", title="t",
                                    text="endpoint surface body"))
        from fastapi.testclient import TestClient
        app = create_app(ctx)
        client = TestClient(app)
        for path in ("/health", "/readiness", "/status", "/config",
                     "/index/stats", "/credits/balance",
                     "/network/peers", "/analytics", "/analytics/tools",
                     "/index/compression", "/metrics", "/openapi-spec",
                     "/dashboard"):
            r = client.get(path)
            assert r.status_code == 200, (path, r.status_code, r.text[:100])
        r = client.get("/search", params={"q": "endpoint surface"})
        assert r.status_code == 200 and r.json()
    finally:
        ctx.close()
