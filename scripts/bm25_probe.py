#!/usr/bin/env python3
"""BM25 plane microbenchmark: isolate the doc-block kernel + top-k at
BASELINE shapes. Round-1 baseline to beat: 982 us/batch kernel time at
1.25M docs / B=128 (v1 global-atomic scatter, profiles/r01_*), with an
extra ~0.6 ms zero-fill it needed on the scores buffer.

Usage (GPU box): python scripts/bm25_probe.py [--docs N] [--batch B]
"""
from __future__ import annotations

import argparse
import sys
import time
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--docs", type=int, default=1_250_000)
    ap.add_argument("--batch", type=int, default=128)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--k", type=int, default=100)
    ap.add_argument("--segments", type=int, default=1,
                    help="split the corpus into this many segments")
    ap.add_argument("--bd", type=int, default=0,
                    help="override the doc-block size")
    args = ap.parse_args()

    from infomesh_amd.ops import _build
    _build.build()
    from infomesh_amd.index.gpu_index import GpuShard
    from infomesh_amd.index.synth import synth_corpus_arrays, synth_queries

    dev = "cuda"
    t0 = time.perf_counter()
    terms, docs, lens = synth_corpus_arrays(args.docs, 120, seed=0)
    shard = GpuShard(dev)
    if args.segments == 1:
        gids = np.arange(args.docs, dtype=np.int64)
        shard.build_from_arrays(terms, docs, lens, gids, None)
    else:
        per = args.docs // args.segments
        bounds = [0] + [per * i for i in range(1, args.segments)] + [args.docs]
        for s in range(args.segments):
            lo, hi = bounds[s], bounds[s + 1]
            sel = (docs >= lo) & (docs < hi)
            shard.build_from_arrays(terms[sel], docs[sel] - lo,
                                    lens[lo:hi],
                                    np.arange(lo, hi, dtype=np.int64), None)
    torch.cuda.synchronize()
    print(f"build: {time.perf_counter() - t0:.1f}s  "
          f"segments={len(shard.segments)}  "
          f"postings={sum(s.doc_ids.numel() for s in shard.segments)}")

    qterms, _ = synth_queries(args.batch, n_terms=4, seed=1, device=dev)
    B, N = args.batch, args.docs
    scores = torch.empty(B, N, device=dev, dtype=torch.float32)

    # warmup
    for _ in range(5):
        shard.search_bm25(qterms, args.k, scores_buf=scores)
    torch.cuda.synchronize()

    # full path (host prep + kernel + topk)
    t0 = time.perf_counter()
    for _ in range(args.iters):
        shard.search_bm25(qterms, args.k, scores_buf=scores)
    torch.cuda.synchronize()
    full_ms = (time.perf_counter() - t0) / args.iters * 1e3

    # kernel-only (no topk): time score writes via events
    from infomesh_amd.ops import kernels as K
    qrows, tset = shard.dedupe_terms(qterms)
    uterms, qt_ut = np.unique(tset, return_inverse=True)
    idf = shard._idf_table()[tset]
    qt_off = np.zeros(B + 1, dtype=np.int64)
    np.cumsum(np.bincount(qrows, minlength=B), out=qt_off[1:])
    qt_off_d = torch.from_numpy(qt_off).to(torch.int32).to(dev)
    qt_ut_d = torch.from_numpy(qt_ut.astype(np.int32)).to(dev)
    qt_idf_d = torch.from_numpy(idf).to(dev)
    bd = args.bd or shard._pick_bd(B)
    segs = []
    for seg in shard.segments:
        qb = torch.from_numpy(seg.h_offs[uterms]).to(dev)
        qe = torch.from_numpy(seg.h_offs[uterms + 1]).to(dev)
        nblocks = (seg.n_docs + bd - 1) // bd
        bw = torch.empty(len(uterms) * nblocks * 2, dtype=torch.int32,
                         device=dev)
        segs.append((seg, qb, qe, bw))
    ev0, ev1 = torch.cuda.Event(True), torch.cuda.Event(True)
    ev0.record()
    for _ in range(args.iters):
        for seg, qb, qe, bw in segs:
            K.bm25_block(seg.doc_ids, seg.tfdl, qt_off_d, qt_ut_d,
                         qt_idf_d, qb, qe, bw, scores,
                         seg.doc_base, seg.n_docs, bd, shard.avgdl)
    ev1.record()
    torch.cuda.synchronize()
    kern_ms = ev0.elapsed_time(ev1) / args.iters

    npost = int(qt_idf_d.numel() and sum(
        int((s.h_offs[tset + 1] - s.h_offs[tset]).sum())
        for s, _, _, _ in segs))
    traffic_gb = (npost * 8 + B * N * 4) / 1e9
    print(f"B={B} N={N} bd={bd} postings/batch={npost}")
    print(f"bm25 kernel: {kern_ms * 1e3:.0f} us/batch  "
          f"({traffic_gb / (kern_ms / 1e3):.2f} GB/s effective)")
    print(f"search_bm25 full (prep+kernel+topk): {full_ms * 1e3:.0f} us")


if __name__ == "__main__":
    main()
