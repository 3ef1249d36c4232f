"""Synthetic corpus/query generator for benchmarks and tests.

BASELINE configs 2/4 run over synthetic docs with random-init encoder
weights (no network for datasets): BM25 postings follow a Zipf term
distribution; dense embeddings are random unit vectors; queries are
random term sets + unit embeddings. All deterministic per (seed, shard).
"""
from __future__ import annotations

import numpy as np
import torch

from .gpu_index import BM25_VOCAB, GpuShard


def synth_corpus_arrays(n_docs: int, avg_len: int = 120,
                        vocab: int = BM25_VOCAB, seed: int = 0,
                        zipf_a: float = 1.3):
    """Returns (flat_terms, flat_docs, doc_lens) numpy arrays.

    Zipf term sampling dominates generation at 10M docs (~110 s
    single-threaded for 567M draws); large corpora fan the draw out
    over worker processes (deterministic per (seed, chunk), same
    distribution — numpy's exact rejection sampler throughout)."""
    rng = np.random.default_rng(seed)
    doc_lens = np.clip(
        rng.lognormal(mean=np.log(avg_len), sigma=0.4, size=n_docs),
        8, avg_len * 6).astype(np.int64)
    total = int(doc_lens.sum())
    if total >= 50_000_000:
        import multiprocessing as mp
        import os
        # under torchrun every rank generates concurrently — share the
        # host's cores instead of spawning world*16 workers (the 8-GPU
        # SCALE run would oversubscribe and slow ALL ranks' setup)
        world = int(os.environ.get("WORLD_SIZE", "1") or 1)
        workers = max(2, min(16, (os.cpu_count() or 4) // max(world, 1)))
        per = (total + workers - 1) // workers
        sizes = [min(per, total - i * per) for i in range(workers)]
        sizes = [s for s in sizes if s > 0]
        jobs = [(seed * 1_000_003 + 17 * i + 1, s, zipf_a, vocab)
                for i, s in enumerate(sizes)]
        # spawn, not fork: the caller may hold a live HIP/RCCL context
        # (multi-rank bench generates after device binding) and forked
        # children of a CUDA process are a known hazard even when they
        # never touch the GPU themselves
        from ._synthworker import zipf_chunk
        ctx = mp.get_context("spawn")
        with ctx.Pool(len(jobs)) as pool:
            parts = pool.map(zipf_chunk, jobs)
        flat_terms = np.concatenate(parts)
    else:
        flat_terms = ((rng.zipf(zipf_a, size=total) - 1)
                      % vocab).astype(np.int64)
    flat_docs = np.repeat(np.arange(n_docs, dtype=np.int64), doc_lens)
    return flat_terms, flat_docs, doc_lens


def synth_embeddings(n_docs: int, dim: int = 384, seed: int = 0,
                     device: str = "cuda",
                     batch: int = 1_000_000) -> torch.Tensor:
    """Random unit-norm bf16 embeddings generated on-device."""
    g = torch.Generator(device=device).manual_seed(seed)
    out = torch.empty(n_docs, dim, device=device, dtype=torch.bfloat16)
    for i in range(0, n_docs, batch):
        j = min(i + batch, n_docs)
        e = torch.randn(j - i, dim, generator=g, device=device)
        out[i:j] = torch.nn.functional.normalize(e, dim=-1).bfloat16()
    return out


def build_synth_shard(n_docs: int, shard_rank: int = 0, world: int = 1,
                      avg_len: int = 120, dim: int = 384,
                      device: str = "cuda", seed: int = 0,
                      with_dense: bool = True,
                      emb_dtype: str = "bf16") -> GpuShard:
    """Build one GPU shard of a world-sharded synthetic corpus.

    Global ids interleave round-robin (gid = local * world + rank) so
    shard contents are disjoint and the union covers the corpus."""
    terms, docs, lens = synth_corpus_arrays(
        n_docs, avg_len, seed=seed * 1000 + shard_rank)
    emb = synth_embeddings(n_docs, dim, seed=seed * 1000 + shard_rank,
                           device=device) if with_dense else None
    gids = np.arange(n_docs, dtype=np.int64) * world + shard_rank
    shard = GpuShard(device=device, emb_dtype=emb_dtype)
    shard.build_from_arrays(terms, docs, lens, gids, emb)
    return shard


def synth_queries(n_queries: int, n_terms: int = 4, dim: int = 384,
                  vocab: int = BM25_VOCAB, seed: int = 1,
                  device: str = "cuda", zipf_a: float = 1.15,
                  skip_head: int = 50):
    """Random queries: term-id arrays (Zipf, slightly flatter than docs,
    skipping the `skip_head` most common terms — the stop-word-removal
    analogue of search/nlp.remove_stop_words) + unit embeddings.
    Returns (list[np.ndarray], tensor [B, dim] f32)."""
    rng = np.random.default_rng(seed)
    terms = [(skip_head + (rng.zipf(zipf_a, size=n_terms) - 1)
              % (vocab - skip_head)).astype(np.int64)
             for _ in range(n_queries)]
    g = torch.Generator(device=device).manual_seed(seed)
    emb = torch.nn.functional.normalize(
        torch.randn(n_queries, dim, generator=g, device=device), dim=-1)
    return terms, emb
