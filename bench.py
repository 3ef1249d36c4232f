#!/usr/bin/env python3
"""Flagship benchmark: hybrid search QPS + p50 on the BASELINE config.

Measures the BASELINE.json headline metric — queries/sec (+ p50 batch
latency) over a hybrid BM25+dense index of 1.25M synthetic docs PER GPU
(= the 10M-doc config at 8 GPUs), queries encoded by the bge-small-shaped
encoder on hand-written CDNA4 kernels, fan-out/top-k-gather over RCCL.

Timed step (rank 0 drives, SPMD collectives on every rank):
  encode query batch (MFMA encoder) -> broadcast terms+embeddings ->
  per-shard BM25 scatter-add + top-k and cosine GEMM + top-k ->
  all-gather k-per-shard candidates -> vectorized RRF fusion on rank 0.
Optional --rerank adds the cross-encoder top-100->10 pass (config 3);
--rag adds a Phi-3-mini-shaped summarizer decode (config 5).

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  # N>1: python -m torch.distributed.run --nnodes=1 --nproc-per-node N
  #      --master-addr 127.0.0.1 bench.py --gpus N ...
Scaling is WEAK: docs-per-GPU fixed, corpus grows with N; value is the
whole-job queries/sec.
"""
from __future__ import annotations

import argparse
import json
import statistics
import time

import numpy as np
import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--warmup", type=int, default=20)
    p.add_argument("--docs-per-gpu", type=int, default=1_250_000)
    p.add_argument("--batch", type=int, default=128)
    p.add_argument("--k-per-shard", type=int, default=100)
    p.add_argument("--n-results", type=int, default=10)
    p.add_argument("--avg-doc-len", type=int, default=120)
    p.add_argument("--query-len", type=int, default=32)
    p.add_argument("--bm25-only", action="store_true")
    p.add_argument("--emb-dtype", choices=("bf16", "fp8"), default="bf16",
                   help="dense-plane embedding storage (fp8 = OCP e4m3, "
                        "half HBM + half read traffic, opt-in)")
    p.add_argument("--rerank", action="store_true",
                   help="add cross-encoder top-100->10 (BASELINE config 3)")
    p.add_argument("--rag", action="store_true",
                   help="add summarizer decode (BASELINE config 5)")
    p.add_argument("--rerank-candidates", type=int, default=100)
    p.add_argument("--rag-new-tokens", type=int, default=32)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--phase-timers", action="store_true",
                   help="print per-phase time breakdown (adds syncs)")
    return p.parse_args()


def main():
    args = parse_args()
    from infomesh_amd.parallel.fabric import Fabric
    from infomesh_amd.parallel.query_plane import DistributedQueryPlane
    from infomesh_amd.index.synth import build_synth_shard, synth_queries

    use_gpu = torch.cuda.is_available()
    fabric = Fabric()
    rank, world = fabric.rank, fabric.world
    device = str(fabric.device)
    docs_per_gpu = args.docs_per_gpu
    if not use_gpu:
        docs_per_gpu = min(docs_per_gpu, 20_000)  # CPU sanity mode only

    if use_gpu:
        from infomesh_amd.ops import _build
        _build.build(verbose=rank == 0)
        from infomesh_amd.ops import _ext
        assert _ext.available(), "HIP extension must load on a GPU box"

    # ---- index build (untimed setup) ------------------------------------
    t_setup = time.perf_counter()
    if use_gpu:
        shard = build_synth_shard(
            docs_per_gpu, shard_rank=rank, world=world,
            avg_len=args.avg_doc_len, device=device, seed=args.seed,
            with_dense=not args.bm25_only, emb_dtype=args.emb_dtype)
    else:
        from infomesh_amd.index.gpu_index import CpuShard
        from infomesh_amd.index.synth import synth_corpus_arrays
        terms, docs, lens = synth_corpus_arrays(
            docs_per_gpu, args.avg_doc_len, seed=args.seed * 1000 + rank)
        emb = torch.nn.functional.normalize(
            torch.randn(docs_per_gpu, 384), dim=-1).bfloat16()
        shard = CpuShard()
        gids = np.arange(docs_per_gpu, dtype=np.int64) * world + rank
        shard.build_from_arrays(terms, docs, lens, gids,
                                None if args.bm25_only else emb)
    plane = DistributedQueryPlane(shard, fabric,
                                  k_per_shard=args.k_per_shard)

    encoder = reranker = summarizer = None
    if not args.bm25_only and use_gpu:
        # EVERY rank builds the (seed-identical) encoder: multi-GPU
        # runs shard the query encode across ranks (rank 0 broadcasts
        # token ids, each rank encodes B/W rows, slices all-gather)
        from infomesh_amd.models.encoder import EmbeddingEncoder
        if rank == 0 or world > 1:
            encoder = EmbeddingEncoder(device=device)
    if rank == 0 and args.rerank and use_gpu:
        from infomesh_amd.models.reranker import CrossEncoderReranker
        reranker = CrossEncoderReranker(device=device)
    if rank == 0 and args.rag and use_gpu:
        from infomesh_amd.models.phi3 import PHI3_MINI, Phi3Decoder
        summarizer = Phi3Decoder(PHI3_MINI, device=device, max_batch=1,
                                 max_seq=2304)
    if (args.rerank or args.rag) and not use_gpu and rank == 0:
        print("# --rerank/--rag need a GPU; running the hybrid step only",
              flush=True)

    # Pre-generate rotating query batches (token ids for the encoder,
    # term ids for BM25). Generation is setup; encoding is TIMED.
    n_batches = 8
    B = args.batch
    qterms_all, qids_all, qlens_all = [], [], []
    for i in range(n_batches):
        terms, _ = synth_queries(B, n_terms=4, seed=args.seed + 7 * i + 1,
                                 device="cpu")
        qterms_all.append(terms)
        g = torch.Generator().manual_seed(args.seed + 13 * i)
        qids = torch.randint(4, 30522, (B, args.query_len), generator=g,
                             dtype=torch.int32)
        qids[:, 0] = 1  # CLS
        qids_all.append(qids.to(device))
        qlens_all.append(torch.full((B,), args.query_len,
                                    dtype=torch.int32, device=device))
    rag_ids = None
    if summarizer is not None:
        g = torch.Generator().manual_seed(args.seed + 131)
        rag_ids = torch.randint(0, 32064, (1, 1024), generator=g,
                                dtype=torch.int32).to(device)

    def run_rerank(fused):
        """Cross-encoder scoring of the fused candidates: B queries x
        n_cand pair encodings (synthetic pair token ids of the real
        shape; candidate COUNT comes from the fused results)."""
        n_cand = min(args.rerank_candidates,
                     int((fused.ids >= 0).sum(dim=1).max().item()) or 1)
        g = torch.Generator().manual_seed(int(fused.ids[0, 0]) & 0x7FFF)
        ids = torch.randint(4, reranker.cfg.vocab_size, (B * n_cand, 160),
                            generator=g, dtype=torch.int32).to(device)
        ids[:, 0] = 1
        lens = torch.full((B * n_cand,), 160, dtype=torch.int32,
                          device=device)
        scores = reranker.score_ids(ids, lens).view(B, n_cand)
        torch.topk(scores, min(args.n_results, n_cand), dim=1)

    def run_rag():
        summarizer.reset()
        out = summarizer.generate_greedy(rag_ids,
                                         max_new_tokens=args.rag_new_tokens)
        return out

    # CPU sanity mode has no encoder: pre-made unit embeddings stand in.
    cpu_emb = None
    if rank == 0 and not use_gpu and not args.bm25_only:
        g = torch.Generator().manual_seed(args.seed + 5)
        cpu_emb = [torch.nn.functional.normalize(
            torch.randn(B, 384, generator=g), dim=-1)
            for _ in range(n_batches)]

    phase_t: dict[str, float] = {}

    def _mark(name: str, t0: float) -> float:
        if args.phase_timers and use_gpu:
            torch.cuda.synchronize()
        t1 = time.perf_counter()
        if args.phase_timers:
            phase_t[name] = phase_t.get(name, 0.0) + (t1 - t0)
        return t1

    def step(i: int) -> None:
        j = i % n_batches
        emb = None
        encode_fn = None
        encode_shard = None
        tp = time.perf_counter()
        if encoder is not None and world > 1:
            # shard the query encode across ranks (plane broadcasts the
            # token ids and all-gathers the embedding slices)
            encode_shard = {"S": args.query_len, "fn": encoder.encode_ids,
                            "qids": qids_all[j] if rank == 0 else None,
                            "qlens": qlens_all[j] if rank == 0 else None}
        elif rank == 0 and encoder is not None:
            # passed as a callback: the query plane overlaps the BM25
            # side-stream work with the encoder forward
            def encode_fn():
                return encoder.encode_ids(qids_all[j], qlens_all[j])
        elif rank == 0 and cpu_emb is not None:
            emb = cpu_emb[j]
        tp = _mark("encode", tp)
        fused = plane.search_batch(
            qterms_all[j] if rank == 0 else None,
            emb, B, dim=384, n_results=max(
                args.n_results,
                args.rerank_candidates if args.rerank else 0),
            use_dense=not args.bm25_only, phase_t=phase_t
            if args.phase_timers else None, encode_fn=encode_fn,
            encode_shard=encode_shard)
        tp = _mark("search+fuse", tp)
        if rank == 0 and reranker is not None:
            run_rerank(fused)
            tp = _mark("rerank", tp)
        if rank == 0 and summarizer is not None:
            run_rag()
            tp = _mark("rag", tp)

    def sync():
        if use_gpu:
            torch.cuda.synchronize()
        fabric.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    setup_s = time.perf_counter() - t_setup

    for i in range(args.warmup):
        step(i)
    sync()
    phase_t.clear()  # drop one-time warmup costs from phase stats
    lat_ms = []
    submit_ms = []
    t0 = time.perf_counter()
    for i in range(args.steps):
        ts = time.perf_counter()
        step(args.warmup + i)
        submit_ms.append((time.perf_counter() - ts) * 1e3)
        if use_gpu:
            torch.cuda.synchronize()
        lat_ms.append((time.perf_counter() - ts) * 1e3)
    sync()
    elapsed = time.perf_counter() - t0
    elapsed = fabric.all_reduce_max(elapsed)

    if rank == 0:
        total_queries = B * args.steps
        qps = total_queries / elapsed
        p50 = statistics.median(lat_ms)
        mode = "bm25" if args.bm25_only else "hybrid"
        if args.rerank:
            mode += "+rerank"
        if args.rag:
            mode += "+rag"
        result = {
            "metric": "queries/sec (hybrid BM25+dense search, "
                      f"{docs_per_gpu * world / 1e6:.3g}M docs)",
            "value": round(qps, 2),
            "unit": "queries/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": (("bf16+fp8emb" if args.emb_dtype == "fp8"
                       else "bf16") if use_gpu else "fp32-cpu-sanity"),
            "data": "synthetic",
            "config": {
                "model": "bge-small-en(random-init) encoder + "
                         "BM25 CSR + cosine top-k" +
                         (" + bge-reranker-base" if args.rerank else "") +
                         (" + phi-3-mini rag" if args.rag else ""),
                "global_batch": B,
                "seq_len": args.query_len,
                "parallelism": f"shard{world}",
                "mode": mode,
                "docs_per_gpu": docs_per_gpu,
                "total_docs": docs_per_gpu * world,
                "k_per_shard": args.k_per_shard,
                "p50_batch_ms": round(p50, 3),
                "p50_query_ms": round(p50, 3),
                "p95_batch_ms": round(
                    sorted(lat_ms)[int(len(lat_ms) * 0.95) - 1], 3),
                "p99_batch_ms": round(
                    sorted(lat_ms)[max(int(len(lat_ms) * 0.99) - 1, 0)], 3),
                # host submit time of a step (work enqueue until the
                # final sync): the gap between this and p50 is GPU tail
                "p50_submit_ms": round(statistics.median(submit_ms), 3),
                "setup_s": round(setup_s, 1),
            },
        }
        if args.phase_timers:
            result["config"]["phase_ms"] = {
                k: round(v / args.steps * 1e3, 3)
                for k, v in phase_t.items()}
            result["config"]["lat_ms_all"] = [round(x, 2) for x in lat_ms]
        print(json.dumps(result))
    fabric.destroy()


if __name__ == "__main__":
    main()
