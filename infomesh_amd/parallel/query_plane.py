"""Distributed query plane: broadcast queries to every GPU shard, score
locally, all-gather fixed-size top-k, fuse on rank 0.

This is the MI355X-native replacement for the reference's
QueryRouter.route_query scatter-gather over libp2p
(infomesh/p2p/routing.py:133-267) and DHT-pointer aggregation
(infomesh/index/distributed.py:235-276): exhaustive fan-out to all
shards instead of probabilistic top-5 peers, all-gather of
k-per-shard candidate records over xGMI instead of msgpack streams.

The RRF fusion is fully vectorized (sort + segment-sum) so the host
merge never bottlenecks the GPU planes.
"""
from __future__ import annotations

from dataclasses import dataclass

import numpy as np
import torch

from ..index.gpu_index import GpuShard, ShardHits
from .fabric import Fabric

RRF_K = 60
MAX_QUERY_TERMS = 32


@dataclass
class FusedHits:
    """Final fused results: global doc ids + fused scores [B, n]."""
    ids: torch.Tensor     # [B, n] i64 (-1 pad)
    scores: torch.Tensor  # [B, n] f32
    bm25_ids: torch.Tensor
    bm25_scores: torch.Tensor
    dense_ids: torch.Tensor
    dense_scores: torch.Tensor
    degraded: bool = False   # served by a shrunk rank group


def rrf_fuse(ids_lists: list[torch.Tensor], score_lists: list[torch.Tensor],
             weights: list[float], n: int, k: int = RRF_K) -> tuple[torch.Tensor, torch.Tensor]:
    """Vectorized multi-source RRF.

    Each source: ids [B, M_s] i64 (-1 = pad), scores [B, M_s] f32 (higher
    better). Per source, global rank = position after a descending sort;
    contribution = w / (k + rank). Contributions are summed per id via a
    sort + segment-sum, then the top-n fused ids are returned."""
    B = ids_lists[0].shape[0]
    contribs, all_ids = [], []
    for ids, scores, w in zip(ids_lists, score_lists, weights):
        order = torch.argsort(scores, dim=1, descending=True)
        sorted_ids = torch.gather(ids, 1, order)
        ranks = torch.arange(1, ids.shape[1] + 1, device=ids.device,
                             dtype=torch.float32).expand(B, -1)
        c = w / (k + ranks)
        valid = sorted_ids >= 0
        all_ids.append(torch.where(valid, sorted_ids,
                                   torch.full_like(sorted_ids, 2**62)))
        contribs.append(torch.where(valid, c, torch.zeros_like(c)))
    ids_cat = torch.cat(all_ids, dim=1)          # [B, M]
    c_cat = torch.cat(contribs, dim=1)
    # Segment-sum per row over equal ids.
    ids_sorted, order = torch.sort(ids_cat, dim=1)
    c_sorted = torch.gather(c_cat, 1, order)
    csum = torch.cumsum(c_sorted, dim=1)
    M = ids_cat.shape[1]
    is_last = torch.ones_like(ids_sorted, dtype=torch.bool)
    is_last[:, :-1] = ids_sorted[:, :-1] != ids_sorted[:, 1:]
    # fused score at each last-position = csum[last] - csum[prev_last]
    prev = torch.zeros_like(csum)
    prev[:, 1:] = csum[:, :-1]
    seg_start = torch.ones_like(is_last)
    seg_start[:, 1:] = ids_sorted[:, 1:] != ids_sorted[:, :-1]
    # carry the csum value at the position before each segment start
    base = torch.where(seg_start, prev, torch.zeros_like(prev))
    base_ff = torch.cummax(
        torch.where(seg_start, prev,
                    torch.full_like(prev, -1.0)), dim=1).values
    fused = csum - base_ff
    fused = torch.where(is_last & (ids_sorted < 2**62), fused,
                        torch.full_like(fused, -1.0))
    top = torch.topk(fused, min(n, M), dim=1)
    out_ids = torch.gather(ids_sorted, 1, top.indices)
    out_ids = torch.where(top.values > 0, out_ids,
                          torch.full_like(out_ids, -1))
    out_scores = torch.clamp(top.values, min=0.0)
    return out_ids, out_scores


def _as_i64(sl: torch.Tensor) -> torch.Tensor:
    """Bit-cast a float32 id slice back to int64 [..., 2k] -> [..., k],
    robust to storage offset/stride parity (a contiguous slice keeps
    its parent's offset, and float->int64 view demands even offset,
    stride-1 last dim and even length)."""
    shp = sl.shape[:-1] + (sl.shape[-1] // 2,)
    flat = sl.contiguous().reshape(-1)
    if flat.storage_offset() % 2:
        flat = flat.clone()
    return flat.view(torch.int64).reshape(shp)


class DistributedQueryPlane:
    """SPMD query plane: every rank calls search_batch collectively."""

    def __init__(self, shard: GpuShard, fabric: Fabric | None = None,
                 k_per_shard: int = 100):
        self.shard = shard
        self.fabric = fabric or Fabric()
        self.k = k_per_shard
        self.fault_stale_s = 5.0
        self._scores_buf: torch.Tensor | None = None
        self._bm25_stream = None
        # RRF fusion is ~30 tiny fixed-shape kernels (~0.5 ms of launch
        # overhead per batch) -> hipGraph-captured per n_results.
        self._fuse_graphs: dict = {}

    @property
    def world_size(self) -> int:
        return self.fabric.world

    def search_batch(self, queries_terms: list[np.ndarray] | None,
                     query_emb: torch.Tensor | None,
                     B: int, dim: int = 384, n_results: int = 10,
                     use_dense: bool = True,
                     phase_t: dict | None = None,
                     encode_fn=None,
                     encode_shard=None) -> FusedHits | None:
        """Collective search with rank-fault tolerance: on a collective
        failure (dead rank → timeout) the fabric shrinks to the
        pre-created exclusion subgroup and the batch retries once,
        flagged degraded (reference parity: local-only serving when
        peers are absent, infomesh/search/query.py:471-490)."""
        try:
            return self._search_batch(queries_terms, query_emb, B, dim,
                                      n_results, use_dense, phase_t,
                                      encode_fn, encode_shard)
        except RuntimeError as e:
            if not self.fabric.initialized or not self.fabric.collective_ok:
                raise
            import logging
            import time as _t
            log = logging.getLogger("infomesh.plane")
            # heartbeat staleness disambiguates a dead rank from a
            # plain kernel error; survivors may hit their collective
            # failures at different times, so give the heartbeat a
            # staleness window before deciding
            dead = self.fabric.dead_ranks(stale_s=self.fault_stale_s)
            if not dead:
                _t.sleep(self.fault_stale_s)
                dead = self.fabric.dead_ranks(stale_s=self.fault_stale_s)
            if not dead:
                raise   # not a rank failure (e.g. a kernel error)
            log.warning("collective failed (%s); dead ranks %s — "
                        "shrinking group", type(e).__name__, dead)
            self.fabric.degrade(dead)
            # resync through the store (collective-free: a timed-out op
            # closes its gloo pairs / poisons its NCCL comm, so the
            # fresh subgroup must stay untouched until every survivor
            # has drained its own failure)
            if self.fabric.collective_ok and not self.fabric.resync():
                more = self.fabric.dead_ranks(stale_s=self.fault_stale_s)
                log.warning("resync failed (dead %s) — serving "
                            "local-only", more)
                self.fabric.degrade(more or list(
                    r for r in self.fabric.active_ranks
                    if r != self.fabric.rank))
            return self._search_batch(queries_terms, query_emb, B, dim,
                                      n_results, use_dense, phase_t,
                                      encode_fn, encode_shard)

    def _sharded_encode(self, encode_shard: dict, B: int,
                        dim: int) -> torch.Tensor:
        """Per-rank query-encode sharding (BACKLOG 8-GPU win): rank 0
        broadcasts the token-id batch; every ACTIVE rank encodes an
        equal slice with its own (identically-seeded) encoder; slices
        all-gather back into the full [B, dim] embedding matrix on
        every rank — which also replaces the embedding broadcast.
        Encoder wall time scales 1/W at the cost of one [B,S] i32
        broadcast + one [B/W, dim] all-gather (tens of KB over xGMI)."""
        fa = self.fabric
        dev = fa.device
        S = int(encode_shard["S"])
        fn = encode_shard["fn"]
        W = fa.effective_world
        Bs = (B + W - 1) // W
        Bp = Bs * W
        ids_t = torch.zeros(Bp, S, dtype=torch.int32, device=dev)
        lens_t = torch.ones(Bp, dtype=torch.int32, device=dev)
        if fa.rank == 0:
            qids = encode_shard["qids"]
            qlens = encode_shard["qlens"]
            ids_t[:B] = qids.to(dev)
            lens_t[:B] = qlens.to(dev)
        fa.broadcast(ids_t)
        fa.broadcast(lens_t)
        my = fa.active_ranks.index(fa.rank)
        sl = slice(my * Bs, (my + 1) * Bs)
        emb_slice = fn(ids_t[sl].contiguous(),
                       lens_t[sl].contiguous())
        gath = fa.all_gather(emb_slice.contiguous())
        return gath.reshape(Bp, dim)[:B]

    def _search_batch(self, queries_terms: list[np.ndarray] | None,
                      query_emb: torch.Tensor | None,
                      B: int, dim: int = 384, n_results: int = 10,
                      use_dense: bool = True,
                      phase_t: dict | None = None,
                      encode_fn=None, encode_shard=None
                      ) -> FusedHits | None:
        """One collective search attempt. Rank 0 passes real queries and
        gets the FusedHits; other ranks pass None and get None.

        With encode_fn (rank 0), the BM25 plane is launched on a side
        stream BEFORE the query encoding runs, overlapping the encoder's
        MFMA work with the BM25 scatter/top-k (they are independent
        until fusion). Collective order stays identical on all ranks:
        bcast(terms) -> [local overlap] -> bcast(emb) -> all-gathers."""
        import time as _time

        def mark(name, t0):
            if phase_t is None:
                return t0
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            t1 = _time.perf_counter()
            phase_t[name] = phase_t.get(name, 0.0) + (t1 - t0)
            return t1

        tp = _time.perf_counter()
        if torch.cuda.is_available() and self.shard.device.type == "cuda":
            hits = self._search_overlapped(queries_terms, query_emb,
                                           encode_fn, B, dim, use_dense,
                                           phase_t=phase_t,
                                           encode_shard=encode_shard)
            tp = mark("plane.shard", tp)
        else:
            terms = self._broadcast_terms(queries_terms, B)
            if encode_shard is not None:
                emb = self._sharded_encode(encode_shard, B, dim).float()
            else:
                if encode_fn is not None and self.fabric.rank == 0:
                    query_emb = encode_fn()
                dev = self.fabric.device
                emb = (query_emb.to(dev).float()
                       if query_emb is not None
                       else torch.zeros(B, dim, device=dev))
                self.fabric.broadcast(emb)
            tp = mark("plane.pack", tp)
            hits = self.shard.search(
                terms, emb if use_dense else None, k=self.k,
                scores_buf=self._get_scores_buf(B), phase_t=phase_t)
            tp = mark("plane.shard", tp)
        # ONE packed all-gather instead of four: xGMI collectives at
        # this payload size (a few hundred KB) are latency-dominated,
        # so the [B,k] score blocks and the bit-reinterpreted i64 id
        # blocks ride together -> [W, B, 6k], unpacked after
        B_, k_ = hits.bm25_scores.shape
        packed = torch.cat([
            hits.bm25_scores, hits.dense_scores,
            hits.bm25_ids.view(torch.float32).reshape(B_, 2 * k_),
            hits.dense_ids.view(torch.float32).reshape(B_, 2 * k_),
        ], dim=1)
        g = self.fabric.all_gather(packed)        # [W, B, 6k]
        bm_s = g[:, :, :k_].contiguous()
        dn_s = g[:, :, k_:2 * k_].contiguous()
        # .contiguous() is a no-op for an already-contiguous slice, which
        # at world=1 leaves an ODD float storage offset when k is odd —
        # view(int64) then raises. _as_i64 guarantees offset-0 storage.
        bm_i = _as_i64(g[:, :, 2 * k_:4 * k_])
        dn_i = _as_i64(g[:, :, 4 * k_:6 * k_])
        for tk in (self.shard._get_topk(),
                   getattr(self.shard, "_topk_dense", None)):
            if tk is not None and getattr(tk, "defer_check", False):
                tk.check_pending()
        tp = mark("plane.gather", tp)
        if self.fabric.rank != 0:
            return None
        W, _, k = bm_s.shape
        bm_s = bm_s.permute(1, 0, 2).reshape(B, W * k)
        bm_i = bm_i.permute(1, 0, 2).reshape(B, W * k)
        dn_s = dn_s.permute(1, 0, 2).reshape(B, W * k)
        dn_i = dn_i.permute(1, 0, 2).reshape(B, W * k)
        if use_dense:
            if bm_i.is_cuda:
                gf = self._fuse_graphs.get(n_results)
                if gf is None:
                    from ..ops.graphs import GraphedCallable

                    def _fuse(a, b, c, d, _n=n_results):
                        return rrf_fuse([a, b], [c, d], [1.0, 1.0], _n)
                    gf = GraphedCallable(_fuse)
                    self._fuse_graphs[n_results] = gf
                ids, scores = gf(bm_i, dn_i, bm_s, dn_s)
                # detach results from the graph's static output buffers
                # (the next replay overwrites them)
                ids, scores = ids.clone(), scores.clone()
            else:
                ids, scores = rrf_fuse([bm_i, dn_i], [bm_s, dn_s],
                                       [1.0, 1.0], n_results)
        else:
            order = torch.argsort(bm_s, dim=1, descending=True)
            ids = torch.gather(bm_i, 1, order)[:, :n_results]
            scores = torch.gather(bm_s, 1, order)[:, :n_results]
        mark("plane.fuse", tp)
        return FusedHits(ids=ids, scores=scores, bm25_ids=bm_i,
                         bm25_scores=bm_s, dense_ids=dn_i,
                         dense_scores=dn_s,
                         degraded=self.fabric.degraded)

    def _search_overlapped(self, queries_terms, query_emb, encode_fn,
                           B, dim, use_dense, phase_t=None,
                           encode_shard=None) -> ShardHits:
        """BM25 on a side stream || query encoding on the main stream.

        Broadcast order (terms, then embeddings) is identical on all
        ranks; the BM25 launch between the two is rank-local.

        phase_t (diagnostic only): sub-phase wall times WITH syncs —
        the syncs break the overlap, so only enable to locate time."""
        import time as _time

        def mark(name, t0):
            if phase_t is None:
                return t0
            torch.cuda.synchronize()
            t1 = _time.perf_counter()
            phase_t[name] = phase_t.get(name, 0.0) + (t1 - t0)
            return t1
        dev = self.shard.device
        self.shard._get_topk().defer_check = True
        self.shard._get_topk_dense().defer_check = True
        if self._bm25_stream is None:
            self._bm25_stream = torch.cuda.Stream(dev)
        # terms must be broadcast before any rank's shard work
        tp = _time.perf_counter()
        terms = self._broadcast_terms(queries_terms, B)
        tp = mark("ov.terms", tp)
        main = torch.cuda.current_stream(dev)
        self._bm25_stream.wait_stream(main)
        with torch.cuda.stream(self._bm25_stream):
            bm_vals, bm_idx = self.shard.search_bm25(
                terms, self.k, scores_buf=self._get_scores_buf(B))
            bm_ids = self.shard.to_global(bm_idx)
        tp = mark("ov.bm25", tp)
        if encode_shard is not None:
            emb_t = self._sharded_encode(encode_shard, B, dim).float()
        else:
            if self.fabric.rank == 0:
                emb = encode_fn() if encode_fn is not None else query_emb
            else:
                emb = None
            emb_t = (emb.to(dev).float() if emb is not None
                     else torch.zeros(B, dim, device=dev))
            self.fabric.broadcast(emb_t)
        tp = mark("ov.encode", tp)
        if use_dense and self.shard.embeddings is not None:
            dn_vals, dn_idx = self.shard.search_dense(emb_t, self.k)
            dn_ids = self.shard.to_global(dn_idx)
            tp = mark("ov.dense", tp)
        else:
            k = min(self.k, self.shard.n_docs)
            dn_vals = torch.full((B, k), -float("inf"), device=dev)
            dn_ids = torch.full((B, k), -1, device=dev, dtype=torch.int64)
        main.wait_stream(self._bm25_stream)
        return ShardHits(bm25_scores=bm_vals, bm25_ids=bm_ids,
                         dense_scores=dn_vals, dense_ids=dn_ids)

    def _broadcast_terms(self, queries_terms, B):
        if self.fabric.world == 1 and queries_terms is not None:
            # no collective needed: skip the pack + GPU round-trip +
            # device sync entirely
            return queries_terms
        dev = self.fabric.device
        terms_t = torch.full((B, MAX_QUERY_TERMS), -1, dtype=torch.int64)
        if queries_terms is not None:
            for i, t in enumerate(queries_terms):
                t = t[:MAX_QUERY_TERMS]
                terms_t[i, :len(t)] = torch.from_numpy(t.astype(np.int64))
        terms_t = terms_t.to(dev)
        self.fabric.broadcast(terms_t)
        tt = terms_t.cpu().numpy()
        return [tt[i][tt[i] >= 0] for i in range(B)]

    def _get_scores_buf(self, B: int) -> torch.Tensor | None:
        N = self.shard.n_docs
        if N == 0:
            return None
        if (self._scores_buf is None or
                self._scores_buf.shape != (B, N)):
            self._scores_buf = torch.zeros(
                B, N, device=self.shard.device, dtype=torch.float32)
        return self._scores_buf
