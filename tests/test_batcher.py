"""Dynamic query batcher + single-fusion serving path.

Round-1 VERDICT #1: the benched batch-128 QPS must be reachable from
the real entry points, and the engine serving path must fuse exactly
once with hydrated url/title. These tests run on CPU (CpuShard engine).
"""
from __future__ import annotations

import threading
import time

import numpy as np
import pytest
import torch

from infomesh_amd.config import Config
from infomesh_amd.index.local_store import Document
from infomesh_amd.search.batcher import QueryBatcher
from infomesh_amd.services import AppContext


# --------------------------------------------------------------- batcher

def test_batcher_groups_concurrent_requests():
    """Requests submitted while the executor is busy coalesce into one
    batch; every caller gets its own query's result."""
    gate = threading.Event()
    calls: list[list[str]] = []

    def execute(queries, limit):
        calls.append(list(queries))
        gate.wait(5)  # hold the FIRST batch until all others queue up
        return [[f"hit:{q}"] for q in queries]

    b = QueryBatcher(execute=execute, max_batch=64, max_wait_ms=40)
    results: dict[str, list] = {}

    def client(q):
        results[q] = b.submit(q, limit=1)

    threads = [threading.Thread(target=client, args=(f"q{i}",))
               for i in range(9)]
    threads[0].start()
    while not calls:          # first batch is in execute() now
        time.sleep(0.002)
    for t in threads[1:]:
        t.start()
    time.sleep(0.08)          # let the other 8 enqueue
    gate.set()
    for t in threads:
        t.join(10)
    assert results == {f"q{i}": [f"hit:q{i}"] for i in range(9)}
    # batch 1 = the first request alone; the rest grouped
    assert len(calls) <= 3
    assert max(len(c) for c in calls) >= 8
    assert b.stats()["queries"] == 9
    b.close()


def test_batcher_respects_max_batch_and_propagates_errors():
    seen = []

    def execute(queries, limit):
        seen.append(len(queries))
        if "boom" in queries:
            raise ValueError("executor exploded")
        return [[q] for q in queries]

    b = QueryBatcher(execute=execute, max_batch=4, max_wait_ms=0)
    assert b.submit("a") == ["a"]
    with pytest.raises(ValueError, match="exploded"):
        b.submit("boom")
    # batcher still alive after an error
    assert b.submit("c") == ["c"]
    assert all(n <= 4 for n in seen)
    b.close()
    with pytest.raises(RuntimeError):
        b.submit("after-close")


def test_batcher_per_request_limit():
    def execute(queries, limit):
        # limit passed to the executor is the max over the batch
        return [[(q, i) for i in range(limit)] for q in queries]

    b = QueryBatcher(execute=execute, max_batch=8, max_wait_ms=0)
    out = b.submit("x", limit=3)
    assert len(out) == 3
    b.close()


# ------------------------------------------------- single-fusion serving

CORPUS = [
    ("https://rocm.docs/hip", "HIP programming guide",
     "HIP kernels compile with hipcc for CDNA4 gfx950 wavefront64."),
    ("https://rocm.docs/mfma", "Matrix cores",
     "MFMA instructions drive bf16 matrix multiplication throughput."),
    ("https://kernel.org/sched", "CFS scheduler",
     "The completely fair scheduler balances runnable tasks on cores."),
    ("https://pytorch.org/dist", "Distributed training",
     "torch distributed all reduce gradients over process groups rccl."),
    ("https://xgmi.amd/links", "xGMI fabric",
     "Seven xGMI links connect eight GPUs point to point at high speed."),
]


@pytest.fixture()
def ctx():
    c = AppContext.create(config=Config(), with_engine=True,
                          with_worker=False, in_memory=True)
    assert c.engine is not None and c.batcher is not None
    for url, title, text in CORPUS:
        c.index_document(Document(url=url, title=title, text=text),
                         attest=False, credit=False)
    c.flush_engine()
    yield c
    c.close()


def test_engine_path_single_fusion_and_hydration(ctx):
    """services.search on the engine path must (a) return exactly the
    engine's fused ranking (no second RRF against FTS5) and (b) hydrate
    url/title/snippet from the LocalStore."""
    resp = ctx.search("mfma matrix multiplication", use_cache=False,
                      deduct=False)
    assert resp.mode == "hybrid" and resp.results
    # oracle: the engine's own fused output for the same effective query
    from infomesh_amd.search.query import preprocess_query
    eff = preprocess_query("mfma matrix multiplication")
    oracle = ctx.engine.search(eff, limit=ctx.config.search.max_results)
    oracle_ids = [h.doc_id for h in oracle]
    got_ids = [h.doc_id for h in resp.results]
    assert got_ids == oracle_ids[: len(got_ids)]
    for h in resp.results:
        assert h.url and h.title, "hydration must fill url/title"
    top = resp.results[0]
    assert top.url == "https://rocm.docs/mfma"
    assert "<b>" in top.snippet.lower() or top.snippet


def test_engine_path_served_through_batcher(ctx):
    before = ctx.batcher.stats()["queries"]
    ctx.search("xgmi links", use_cache=False, deduct=False)
    assert ctx.batcher.stats()["queries"] == before + 1


def test_filtered_queries_take_fts_path(ctx):
    """site:/language filters live in SQLite — they bypass the engine."""
    before = ctx.batcher.stats()["queries"]
    resp = ctx.search("scheduler site:kernel.org", use_cache=False,
                      deduct=False)
    assert ctx.batcher.stats()["queries"] == before  # engine untouched
    assert all("kernel.org" in h.url for h in resp.results)


def test_concurrent_clients_through_services(ctx):
    """Many threads through AppContext.search: everyone gets the right
    answer and the batcher actually grouped some of them."""
    queries = ["hip kernels gfx950", "mfma bf16", "scheduler tasks",
               "rccl all reduce", "xgmi links"] * 4
    out: dict[int, object] = {}

    def client(i, q):
        out[i] = ctx.search(q, use_cache=False, deduct=False)

    threads = [threading.Thread(target=client, args=(i, q))
               for i, q in enumerate(queries)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(15)
    assert len(out) == len(queries)
    expect_top = {
        "hip kernels gfx950": "https://rocm.docs/hip",
        "mfma bf16": "https://rocm.docs/mfma",
        "scheduler tasks": "https://kernel.org/sched",
        "rccl all reduce": "https://pytorch.org/dist",
        "xgmi links": "https://xgmi.amd/links",
    }
    for i, q in enumerate(queries):
        resp = out[i]
        assert resp.results, f"no results for {q!r}"
        assert resp.results[0].url == expect_top[q], q
    st = ctx.batcher.stats()
    assert st["queries"] >= len(queries)
