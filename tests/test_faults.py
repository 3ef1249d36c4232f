"""Fault-injection tests: malformed inputs, partial
failures and concurrent access must degrade loudly or safely, never
corrupt state."""
from __future__ import annotations

import json
import struct
import threading

import pytest

from infomesh_amd.index.local_store import Document, LocalStore


# ----------------------------------------------------------- MCP faults

def _mcp(tmp_path, monkeypatch):
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path))
    from infomesh_amd.mcp.server import McpServer
    from infomesh_amd.services import AppContext
    ctx = AppContext.create(with_engine=False, with_worker=False)
    return ctx, McpServer(ctx)


def test_mcp_malformed_messages(tmp_path, monkeypatch):
    ctx, srv = _mcp(tmp_path, monkeypatch)
    try:
        # unknown method -> JSON-RPC error, not an exception
        r = srv.handle_message({"jsonrpc": "2.0", "id": 1,
                                "method": "nope/nope"})
        assert r["error"]["code"] == -32601
        # tools/call with missing tool
        r = srv.handle_message({"jsonrpc": "2.0", "id": 2,
                                "method": "tools/call",
                                "params": {"name": "ghost_tool"}})
        assert "error" in r
        # tools/call with garbage arguments
        r = srv.handle_message({"jsonrpc": "2.0", "id": 3,
                                "method": "tools/call",
                                "params": {"name": "web_search",
                                           "arguments": {"query": None}}})
        assert r is not None  # error or empty result, never a crash
        # notification (no id) never produces a response
        assert srv.handle_message({"jsonrpc": "2.0",
                                   "method": "ping"}) is None
    finally:
        ctx.close()


def test_mcp_stdio_garbage_lines(tmp_path, monkeypatch):
    import io
    ctx, srv = _mcp(tmp_path, monkeypatch)
    try:
        stdin = io.StringIO('this is not json\n'
                            '{"jsonrpc":"2.0","id":9,"method":"ping"}\n')
        stdout = io.StringIO()
        srv.run_stdio(stdin=stdin, stdout=stdout)
        lines = [json.loads(l) for l in stdout.getvalue().splitlines() if l]
        # the garbage line yields a parse error, the ping still works
        assert any(l.get("id") == 9 and "result" in l for l in lines)
    finally:
        ctx.close()


# ------------------------------------------------------ snapshot faults

def test_snapshot_truncated_payload(tmp_path):
    from infomesh_amd.index.snapshot import export_snapshot, import_snapshot
    store = LocalStore(tmp_path / "a.db")
    for i in range(5):
        store.add_document(Document(url=f"https://x/{i}", title=f"t{i}",
                                    text=f"body {i} " * 30))
    path = tmp_path / "s.infomesh-snapshot"
    export_snapshot(store, path, node_name="n")
    data = path.read_bytes()
    # cut the msgpack docs section in half
    (tmp_path / "trunc.infomesh-snapshot").write_bytes(
        data[: len(data) - len(data) // 3])
    dst = LocalStore(tmp_path / "b.db")
    with pytest.raises(Exception):
        import_snapshot(dst, tmp_path / "trunc.infomesh-snapshot")
    assert dst.count() == 0  # nothing half-imported silently
    store.close()
    dst.close()


def test_snapshot_header_len_lies(tmp_path):
    from infomesh_amd.index.snapshot import import_snapshot
    p = tmp_path / "lie.infomesh-snapshot"
    p.write_bytes(struct.pack(">I", 2 ** 31) + b"x" * 64)
    dst = LocalStore(tmp_path / "c.db")
    with pytest.raises(Exception):
        import_snapshot(dst, p)
    dst.close()


# ------------------------------------------------- engine crash safety

def test_engine_flush_failure_keeps_old_epoch():
    """If embedding fails mid-flush, the PREVIOUS shard keeps serving
    (epoch flip happens only after a complete build)."""
    import numpy as np
    from infomesh_amd.engine import HybridEngine

    class GoodThenBadEncoder:
        def __init__(self):
            self.calls = 0

        def encode_texts(self, texts):
            import torch
            self.calls += 1
            if self.calls > 1:
                raise RuntimeError("simulated encoder OOM")
            return torch.nn.functional.normalize(
                torch.randn(len(texts), 384), dim=-1)

    eng = HybridEngine(device="cpu", use_encoder=False)
    eng.encoder = GoodThenBadEncoder()
    eng.add_document(Document(url="u0", title="alpha", text="alpha body",
                              doc_id=0))
    eng.flush(embed_batch=4)
    assert eng.shard.n_docs == 1
    old_shard = eng.shard
    eng.add_document(Document(url="u1", title="beta", text="beta body",
                              doc_id=1))
    with pytest.raises(RuntimeError):
        eng.flush(embed_batch=4)
    # old epoch still intact and serving; the failed doc stays pending
    assert eng.shard is old_shard and eng.shard.n_docs == 1
    assert eng.pending_count == 1
    # encoder recovers -> retry flush succeeds and both docs searchable
    eng.encoder.calls = -10
    eng.flush(embed_batch=4)
    assert eng.shard.n_docs == 2
    assert eng.shard.embeddings.shape[0] == 2
    assert eng.search("alpha", limit=3) and eng.search("beta", limit=3)


# --------------------------------------------- concurrent SQLite access

def test_localstore_concurrent_readers_during_writes(tmp_path):
    """WAL + busy_timeout: a reader in another thread never errors while
    the writer inserts."""
    path = tmp_path / "conc.db"
    store = LocalStore(path)
    errs = []
    stop = threading.Event()

    def reader():
        r = LocalStore(path)
        try:
            while not stop.is_set():
                r.count()
                r.search("body", limit=3)
        except Exception as e:  # noqa: BLE001
            errs.append(e)
        finally:
            r.close()

    t = threading.Thread(target=reader)
    t.start()
    try:
        for i in range(200):
            store.add_document(Document(url=f"https://c/{i}", title="t",
                                        text=f"body {i} " * 10))
    finally:
        stop.set()
        t.join(timeout=30)
    assert not errs
    assert store.count() == 200
    store.close()


def test_read_only_degrade_blocks_writes(tmp_path, monkeypatch):
    """At DegradeLevel.READ_ONLY, index_document refuses loudly and
    search still works (reference governor ladder)."""
    monkeypatch.setenv("INFOMESH_NODE_DATA_DIR", str(tmp_path))
    from infomesh_amd.errors import InfoMeshError
    from infomesh_amd.services import AppContext
    ctx = AppContext.create(with_engine=False, with_worker=False)
    try:
        ctx.index_document(Document(url="https://g.org/1", title="t",
                                    text="governor body " * 10))

        class FrozenGov:
            def writes_allowed(self):
                return False

            def crawl_allowed(self):
                return False

        ctx.governor = FrozenGov()
        with pytest.raises(InfoMeshError):
            ctx.index_document(Document(url="https://g.org/2", title="t",
                                        text="more body " * 10))
        assert ctx.search("governor").results  # reads unaffected
    finally:
        ctx.governor = None
        ctx.close()
