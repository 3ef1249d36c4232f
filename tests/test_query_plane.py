"""Query-plane + shard tests on CPU (gloo multi-process where needed).

Mirrors the reference's fake-collective strategy (dict-
backed MockInfoMeshDHT) with CpuShard + a world-1 Fabric, plus a real
2-process gloo run of the full collective path.
"""
from __future__ import annotations

import os
import time

import numpy as np
import torch

from infomesh_amd.index.gpu_index import CpuShard, BM25_VOCAB, bm25_term_ids
from infomesh_amd.index.synth import synth_corpus_arrays
from infomesh_amd.ops import reference as R
from infomesh_amd.parallel.fabric import Fabric
from infomesh_amd.parallel.query_plane import (DistributedQueryPlane,
                                               rrf_fuse)


def _small_shard(n_docs=500, rank=0, world=1, with_dense=True, seed=3):
    terms, docs, lens = synth_corpus_arrays(n_docs, avg_len=30, seed=seed)
    emb = torch.nn.functional.normalize(
        torch.randn(n_docs, 16, generator=torch.Generator().manual_seed(seed)),
        dim=-1).bfloat16() if with_dense else None
    shard = CpuShard()
    gids = np.arange(n_docs, dtype=np.int64) * world + rank
    shard.build_from_arrays(terms, docs, lens, gids, emb)
    return shard


# ---------------------------------------------------------------- CpuShard

def test_cpu_shard_bm25_matches_reference():
    """CSR BM25 scoring == the scalar dict-based reference formula."""
    n_docs = 200
    terms, docs, lens = synth_corpus_arrays(n_docs, avg_len=20, seed=1)
    shard = CpuShard()
    shard.build_from_arrays(terms, docs, lens,
                            np.arange(n_docs, dtype=np.int64), None)
    postings: dict[int, list[tuple[int, int]]] = {}
    from collections import Counter
    per_doc = [Counter() for _ in range(n_docs)]
    for t, d in zip(terms, docs):
        per_doc[d][int(t)] += 1
    for d, cnt in enumerate(per_doc):
        for t, tf in cnt.items():
            postings.setdefault(t, []).append((d, tf))
    for t in postings:
        postings[t].sort()
    queries = [np.array([int(terms[0]), int(terms[5])]),
               np.array([int(terms[100])])]
    ref = R.bm25_scores(postings, torch.from_numpy(lens),
                        [list(map(int, q)) for q in queries], n_docs)
    hits = shard.search(queries, None, k=n_docs)
    dense_scores = torch.zeros(2, n_docs)
    for qi in range(2):
        for j in range(n_docs):
            gid = int(hits.bm25_ids[qi, j])
            if gid >= 0:
                dense_scores[qi, gid] = hits.bm25_scores[qi, j]
    assert torch.allclose(dense_scores, ref, atol=1e-4)


def test_shard_search_returns_global_ids():
    shard = _small_shard(rank=2, world=4)
    q = [np.array([1, 2, 3])]
    hits = shard.search(q, torch.randn(1, 16), k=10)
    valid = hits.bm25_ids[hits.bm25_ids >= 0]
    assert ((valid % 4) == 2).all()


def test_bm25_term_ids():
    ids = bm25_term_ids("Hello hello WORLD")
    assert len(ids) == 3
    assert ids[0] == ids[1]
    assert (ids < BM25_VOCAB).all()


# ---------------------------------------------------------------- rrf_fuse

def test_rrf_fuse_single_source_order():
    ids = torch.tensor([[10, 20, 30, -1]])
    scores = torch.tensor([[0.5, 0.9, 0.1, -1.0]])
    out_ids, out_scores = rrf_fuse([ids], [scores], [1.0], n=3)
    assert out_ids[0].tolist() == [20, 10, 30]
    assert (out_scores[0][:-1] >= out_scores[0][1:]).all()


def test_rrf_fuse_overlap_wins():
    a_ids = torch.tensor([[1, 2, 3]])
    a_sc = torch.tensor([[0.9, 0.8, 0.7]])
    b_ids = torch.tensor([[9, 2, 8]])
    b_sc = torch.tensor([[0.9, 0.8, 0.7]])
    out_ids, _ = rrf_fuse([a_ids, b_ids], [a_sc, b_sc], [1.0, 1.0], n=5)
    assert out_ids[0, 0] == 2  # appears in both lists


def test_rrf_fuse_matches_python_reference():
    """Vectorized segment-sum RRF == straightforward dict RRF."""
    g = torch.Generator().manual_seed(0)
    B, M = 4, 50
    lists = []
    for _ in range(2):
        ids = torch.randint(0, 100, (B, M), generator=g, dtype=torch.int64)
        sc = torch.rand(B, M, generator=g)
        lists.append((ids, sc))
    out_ids, out_sc = rrf_fuse([l[0] for l in lists],
                               [l[1] for l in lists], [1.0, 0.5], n=10)
    for b in range(B):
        ref: dict[int, float] = {}
        for (ids, sc), w in zip(lists, [1.0, 0.5]):
            order = torch.argsort(sc[b], descending=True)
            for rank, j in enumerate(order.tolist(), start=1):
                gid = int(ids[b, j])
                ref[gid] = ref.get(gid, 0.0) + w / (60 + rank)
        expect = sorted(ref.items(), key=lambda p: -p[1])[:10]
        got = list(zip(out_ids[b].tolist(), out_sc[b].tolist()))
        for (eid, esc), (gid, gsc) in zip(expect, got):
            assert abs(esc - gsc) < 1e-6
            # ids may swap on exactly-tied fused scores
            if abs(esc - gsc) < 1e-9 and eid != gid:
                assert abs(ref[gid] - esc) < 1e-9
            else:
                assert eid == gid


# -------------------------------------------------------- world-1 plane

def test_plane_world1_end_to_end():
    shard = _small_shard()
    plane = DistributedQueryPlane(shard, Fabric(), k_per_shard=20)
    terms = [np.array([1, 2, 3, 4]), np.array([7])]
    emb = torch.nn.functional.normalize(torch.randn(2, 16), dim=-1)
    fused = plane.search_batch(terms, emb, B=2, dim=16, n_results=5)
    assert fused is not None
    assert fused.ids.shape == (2, 5)
    assert (fused.ids >= 0).any()
    assert (fused.scores[:, :-1] >= fused.scores[:, 1:]).all()


def test_plane_bm25_only_mode():
    shard = _small_shard(with_dense=False)
    plane = DistributedQueryPlane(shard, Fabric(), k_per_shard=10)
    fused = plane.search_batch([np.array([1, 2])], None, B=1, dim=16,
                               n_results=5, use_dense=False)
    assert fused is not None and (fused.ids >= 0).any()


# ------------------------------------------------- 2-process gloo run

def _gloo_worker(rank: int, world: int, port: int, ok_file: str):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    fabric = Fabric(backend="gloo")
    shard = _small_shard(n_docs=300, rank=rank, world=world, seed=5 + rank)
    plane = DistributedQueryPlane(shard, fabric, k_per_shard=15)
    terms = [np.array([1, 2, 3]), np.array([4, 5])] if rank == 0 else None
    emb = (torch.nn.functional.normalize(torch.randn(2, 16), dim=-1)
           if rank == 0 else None)
    fused = plane.search_batch(terms, emb, B=2, dim=16, n_results=8)
    if rank == 0:
        assert fused is not None
        valid = fused.ids[fused.ids >= 0]
        # results must come from BOTH shards (parity of global ids)
        assert (valid % 2 == 0).any() and (valid % 2 == 1).any()
        with open(ok_file, "w") as f:
            f.write("ok")
    else:
        assert fused is None
    fabric.destroy()


def test_plane_gloo_world2(tmp_path):
    import torch.multiprocessing as mp
    ok_file = str(tmp_path / "ok")
    port = 29511
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_gloo_worker, args=(r, 2, port, ok_file))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert os.path.exists(ok_file)


# -------------------------------- 4-process gloo vs single-shard oracle

def _corpus_arrays(n_docs, seed):
    """Deterministic synthetic corpus: list of per-doc term arrays +
    unit embeddings (so sharded and unsharded builds agree)."""
    rng = np.random.default_rng(seed)
    docs = [rng.integers(0, 800, size=rng.integers(5, 30)).astype(np.int64)
            for _ in range(n_docs)]
    g = torch.Generator().manual_seed(seed)
    emb = torch.nn.functional.normalize(
        torch.randn(n_docs, 16, generator=g), dim=-1).bfloat16()
    return docs, emb


def _build_from_docs(docs, emb, gids):
    lens = np.array([len(t) for t in docs], dtype=np.int64)
    flat_terms = np.concatenate(docs)
    flat_docs = np.repeat(np.arange(len(docs), dtype=np.int64),
                          [len(t) for t in docs])
    shard = CpuShard()
    shard.build_from_arrays(flat_terms, flat_docs, lens,
                            np.asarray(gids, dtype=np.int64), emb)
    return shard


_W4 = 4
_N_DOCS4 = 400


def _gloo4_worker(rank: int, port: int, out_file: str):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(_W4),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    fabric = Fabric(backend="gloo")
    docs, emb = _corpus_arrays(_N_DOCS4, seed=11)
    mine = list(range(rank, _N_DOCS4, _W4))   # hash-partition analogue
    shard = _build_from_docs([docs[i] for i in mine], emb[mine], mine)
    plane = DistributedQueryPlane(shard, fabric, k_per_shard=20)
    qterms = [np.array([7, 13, 40]), np.array([99, 100])]
    g = torch.Generator().manual_seed(21)
    qemb = torch.nn.functional.normalize(torch.randn(2, 16, generator=g),
                                         dim=-1)
    fused = plane.search_batch(qterms if rank == 0 else None,
                               qemb if rank == 0 else None,
                               B=2, dim=16, n_results=10)
    if rank == 0:
        bm = torch.sort(fused.bm25_scores, dim=1,
                        descending=True).values[:, :20]
        dn = torch.sort(fused.dense_scores, dim=1,
                        descending=True).values[:, :20]
        torch.save({"bm": bm, "dn": dn, "ids": fused.ids}, out_file)
    fabric.destroy()


def test_plane_gloo_world4_matches_single_shard(tmp_path):
    """4-way sharded fan-out must fuse to the same top results as ONE
    shard holding the whole corpus (exact search, exhaustive fan-out)."""
    import torch.multiprocessing as mp
    out_file = str(tmp_path / "ids")
    port = 29523
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_gloo4_worker, args=(r, port, out_file))
             for r in range(_W4)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    sharded = torch.load(out_file, weights_only=True)

    # Single-shard oracle (world-1 plane, same corpus, same queries).
    # DENSE cosine is shard-invariant, and the union of per-shard
    # top-20 always contains the global top-20 — the sorted top-20
    # dense SCORES must match exactly (tie-robust, unlike fused ids).
    # BM25 is NOT compared against the oracle: idf/avgdl are per-shard
    # statistics by design (the reference's per-node FTS5 behaves the
    # same way), so per-shard scores legitimately differ from a
    # whole-corpus build.
    docs, emb = _corpus_arrays(_N_DOCS4, seed=11)
    shard = _build_from_docs(docs, emb, list(range(_N_DOCS4)))
    plane = DistributedQueryPlane(shard, Fabric(), k_per_shard=20)
    qterms = [np.array([7, 13, 40]), np.array([99, 100])]
    g = torch.Generator().manual_seed(21)
    qemb = torch.nn.functional.normalize(torch.randn(2, 16, generator=g),
                                         dim=-1)
    fused = plane.search_batch(qterms, qemb, B=2, dim=16, n_results=10)
    dn_o = torch.sort(fused.dense_scores, dim=1,
                      descending=True).values[:, :20]
    assert torch.allclose(sharded["dn"], dn_o, atol=1e-3), \
        (sharded["dn"][0, :5], dn_o[0, :5])
    # BM25 sanity on the sharded side: 4 shards contributed and the
    # gathered blocks carry real scores
    assert (sharded["bm"][:, 0] > 0).all()
    ids = sharded["ids"]
    valid = ids[ids >= 0]
    assert len({int(i) % _W4 for i in valid}) >= 3, \
        "fused results should draw from (nearly) all shards"


# ----------------------------------- rank-fault degraded serving (gloo)

_W3 = 3


def _fault_worker(rank: int, port: int, out_file: str):
    """3 ranks; rank 2 crashes after batch 1. Survivors detect via
    heartbeat staleness after the collective timeout, shrink to the
    pre-created exclusion subgroup, and batch 2 serves degraded from
    shards 0+1 (reference query.py:471-490 behavior)."""
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(_W3),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    fabric = Fabric(backend="gloo", timeout_s=6, heartbeat_s=0.3)
    docs, emb = _corpus_arrays(_N_DOCS4, seed=13)
    mine = list(range(rank, _N_DOCS4, _W3))
    shard = _build_from_docs([docs[i] for i in mine], emb[mine], mine)
    plane = DistributedQueryPlane(shard, fabric, k_per_shard=20)
    plane.fault_stale_s = 2.0
    qterms = [np.array([7, 13, 40]), np.array([99, 100])]
    g = torch.Generator().manual_seed(23)
    qemb = torch.nn.functional.normalize(torch.randn(2, 16, generator=g),
                                         dim=-1)

    def run_batch():
        return plane.search_batch(qterms if rank == 0 else None,
                                  qemb if rank == 0 else None,
                                  B=2, dim=16, n_results=12)

    f1 = run_batch()
    if rank == 0:
        assert f1 is not None and not f1.degraded
    if rank == 2:
        os._exit(0)   # crash-stop: no destroy, heartbeats cease
    time.sleep(3.0)   # let rank 2's heartbeat go stale
    f2 = run_batch()  # collective times out -> degrade -> retry
    assert fabric.degraded and fabric.active_ranks == [0, 1]
    f3 = run_batch()  # subsequent batches stay on the shrunk group
    if rank == 0:
        assert f2 is not None and f2.degraded and f3.degraded
        valid = f2.ids[f2.ids >= 0]
        # results now come only from shards 0 and 1 (gid mod 3)
        assert ((valid % _W3) != 2).all()
        assert (valid % _W3 == 0).any() and (valid % _W3 == 1).any()
        torch.save({"ok": True}, out_file)
    fabric.destroy()


def test_plane_rank_drop_degraded(tmp_path):
    import torch.multiprocessing as mp
    out_file = str(tmp_path / "ok")
    port = 29531
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_fault_worker, args=(r, port, out_file))
             for r in range(_W3)]
    for p in procs:
        p.start()
    deadline = time.time() + 150
    for i, p in enumerate(procs):
        p.join(timeout=max(1.0, deadline - time.time()))
        if i != 2:
            assert p.exitcode == 0, f"rank {i} exit {p.exitcode}"
    assert os.path.exists(out_file)


# ------------------------------------- sharded encode (gloo, world 2)

def _fake_encode(ids: torch.Tensor, lens: torch.Tensor) -> torch.Tensor:
    """Deterministic stand-in encoder: every rank computes the same
    embedding for the same token rows (like the seed-identical BERT)."""
    freqs = torch.arange(1, 9, dtype=torch.float32) * 0.13
    x = torch.sin(ids.float().unsqueeze(2) * freqs).sum(dim=1)
    return torch.nn.functional.normalize(x, dim=-1)


def _enc_shard_worker(rank: int, port: int, out_file: str):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": "2",
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    fabric = Fabric(backend="gloo")
    docs, emb = _corpus_arrays(200, seed=31)
    # embeddings must live in the fake encoder's 8-dim space: re-embed
    # docs with the same fake encoder over their first terms
    doc_ids = torch.zeros(200, 6, dtype=torch.int32)
    for i, d in enumerate(docs):
        t = torch.from_numpy(d[:6].astype(np.int64))
        doc_ids[i, :len(t)] = t.to(torch.int32)
    demb = _fake_encode(doc_ids, None).bfloat16()
    mine = list(range(rank, 200, 2))
    shard = _build_from_docs([docs[i] for i in mine], demb[mine], mine)
    plane = DistributedQueryPlane(shard, fabric, k_per_shard=15)
    B, S = 3, 6
    g = torch.Generator().manual_seed(41)
    qids = torch.randint(0, 800, (B, S), generator=g, dtype=torch.int32)
    qlens = torch.full((B,), S, dtype=torch.int32)
    qterms = [qids[i].numpy().astype(np.int64) for i in range(B)]
    shard_arg = {"S": S, "fn": _fake_encode,
                 "qids": qids if rank == 0 else None,
                 "qlens": qlens if rank == 0 else None}
    fused = plane.search_batch(qterms if rank == 0 else None, None,
                               B=B, dim=8, n_results=10,
                               encode_shard=shard_arg)
    # oracle: same plane, embeddings computed on rank 0 + broadcast
    qemb = _fake_encode(qids, qlens)
    fused2 = plane.search_batch(qterms if rank == 0 else None,
                                qemb if rank == 0 else None,
                                B=B, dim=8, n_results=10)
    if rank == 0:
        assert torch.equal(fused.ids, fused2.ids), \
            "sharded encode diverges from rank-0 encode + broadcast"
        assert torch.allclose(fused.scores, fused2.scores, atol=1e-5)
        with open(out_file, "w") as f:
            f.write("ok")
    fabric.destroy()


def test_plane_sharded_encode_matches_broadcast(tmp_path):
    import torch.multiprocessing as mp
    out_file = str(tmp_path / "ok")
    port = 29541
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_enc_shard_worker,
                         args=(r, port, out_file)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert os.path.exists(out_file)
