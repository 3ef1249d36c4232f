"""HybridEngine: the GPU-side search engine for REAL documents.

Bridges the durable LocalStore ground truth to GPU shards: documents are
tokenized (BM25 term ids) + encoded (bge-small MFMA encoder), staged in
pending buffers, and flipped into the scored index on flush() — the GPU
analogue of FTS5's WAL+optimize cycle (SURVEY.md §7 "hard parts").

On CPU-only machines the same engine runs with CpuShard + no encoder
(BM25-only or externally-supplied embeddings), so the whole orchestration
is testable without a GPU and the GPU path is identical code.
"""
from __future__ import annotations

import logging

import numpy as np
import torch

from .index.gpu_index import CpuShard, GpuShard, bm25_term_ids
from .index.local_store import Document, LocalStore, SearchHit
from .parallel.fabric import Fabric
from .parallel.query_plane import DistributedQueryPlane

log = logging.getLogger("infomesh.engine")


class HybridEngine:
    def __init__(self, device: str | None = None, k_per_shard: int = 100,
                 use_encoder: bool = True, encoder_max_len: int = 128,
                 fabric: Fabric | None = None, emb_dtype: str = "bf16",
                 hbm_budget_gb: float = 260.0,
                 embed_max_chars: int = 2000,
                 require_extension: bool = True):
        self.gpu = torch.cuda.is_available() if device is None \
            else device.startswith("cuda")
        self.device = device or ("cuda" if self.gpu else "cpu")
        self.fabric = fabric or Fabric()
        self.emb_dtype = emb_dtype
        self.hbm_budget_bytes = int(hbm_budget_gb * 1e9)
        self.embed_max_chars = int(embed_max_chars)
        if self.gpu and require_extension:
            # fail at startup, not at the first query (gpu.require_extension)
            from .ops import _ext
            if not _ext.available():
                from .errors import GpuExtensionMissing
                raise GpuExtensionMissing(
                    "GPU present but the HIP extension failed to load "
                    "(gpu.require_extension=true)")
        self.shard: GpuShard = (
            GpuShard(self.device, emb_dtype=emb_dtype) if self.gpu
            else CpuShard(emb_dtype=emb_dtype))
        self.plane = DistributedQueryPlane(self.shard, self.fabric,
                                           k_per_shard=k_per_shard)
        self.encoder = None
        if self.gpu and use_encoder:
            from .models.encoder import EmbeddingEncoder
            self.encoder = EmbeddingEncoder(device=self.device,
                                            max_len=encoder_max_len)
        # pending (not yet searchable) docs
        self._pending_tokens: list[np.ndarray] = []
        self._pending_texts: list[str] = []
        self._pending_ids: list[int] = []
        # (embeddings and postings of already-built docs live in the
        # shard itself: flush merges rather than re-accumulating, so
        # engine memory does not grow with corpus size)

    # ------------------------------------------------------------ ingest
    def add_document(self, doc: Document) -> None:
        assert doc.doc_id is not None
        text = f"{doc.title}\n{doc.text}"[:4000]
        self._pending_tokens.append(bm25_term_ids(text))
        # embed truncation (config index.embed_max_chars; reference
        # vector_store.py:144-157 truncates at 2000 chars)
        self._pending_texts.append(text[:self.embed_max_chars])
        self._pending_ids.append(doc.doc_id)

    @property
    def pending_count(self) -> int:
        return len(self._pending_ids)

    @property
    def doc_count(self) -> int:
        return self.shard.n_docs + self.pending_count

    def flush(self, embed_batch: int = 256) -> int:
        """Make pending docs searchable (epoch flip). Embeds only the
        NEW docs (incremental), then rebuilds the CSR postings from all
        accumulated docs — the segment-merge analogue."""
        if not self._pending_ids:
            return 0
        n_new = len(self._pending_ids)
        # HBM budget guard (config gpu.hbm_budget_gb — the 288 GB/GPU
        # sizing knob): estimate the new segment before committing and
        # refuse the flush instead of tripping the allocator mid-build.
        # Pendings stay pending; the crawl loop logs and retries later
        # (the reference governor's degrade-before-OOM behavior).
        esize = 1 if self.emb_dtype == "fp8" else 2
        est_new = sum(len(t) for t in self._pending_tokens) * 8 \
            + (n_new * 384 * esize if self.encoder is not None else 0) \
            + n_new * 8
        if self.shard.hbm_bytes() + est_new > self.hbm_budget_bytes:
            from .errors import InfoMeshError
            raise InfoMeshError(
                "GPU002",
                f"HBM budget exceeded: shard {self.shard.hbm_bytes()/1e9:.1f}"
                f" GB + ~{est_new/1e9:.2f} GB new > "
                f"{self.hbm_budget_bytes/1e9:.0f} GB budget")
        # Encode BEFORE committing any state: a failed embed (OOM, ...)
        # leaves the engine exactly as it was — the old epoch keeps
        # serving and the pending docs stay pending for a retry.
        new_emb = None
        if self.encoder is not None:
            chunks = []
            new_texts = self._pending_texts
            for i in range(0, len(new_texts), embed_batch):
                batch = new_texts[i:i + embed_batch]
                pad = embed_batch - len(batch)
                if pad:
                    # fixed batch shape keeps the encoder's hipGraph
                    # cache bounded (one graph per (B, S-bucket))
                    batch = batch + [""] * pad
                enc = self.encoder.encode_texts(batch).bfloat16()
                chunks.append(enc[:embed_batch - pad] if pad else enc)
            new_emb = torch.cat(chunks, 0)

        tokens = self._pending_tokens
        ids = self._pending_ids
        self._pending_tokens, self._pending_texts, self._pending_ids = \
            [], [], []
        if self.shard.n_docs == 0:
            lens = np.array([max(len(t), 1) for t in tokens],
                            dtype=np.int64)
            flat_terms = (np.concatenate(tokens)
                          if any(len(t) for t in tokens)
                          else np.zeros(0, np.int64))
            flat_docs = np.repeat(np.arange(len(tokens), dtype=np.int64),
                                  [len(t) for t in tokens])
            shard = (GpuShard(self.device, emb_dtype=self.emb_dtype)
                     if self.gpu else CpuShard(emb_dtype=self.emb_dtype))
            shard.build_from_arrays(
                flat_terms, flat_docs, lens,
                np.asarray(ids, dtype=np.int64), new_emb)
        else:
            # epoch flip via segment merge: only the NEW docs are
            # tokenized/encoded; readers keep the old shard until the
            # merged one is fully installed
            shard = self.shard.merged_with(tokens, ids, new_emb)
        self.shard = shard
        self.plane.shard = shard
        log.info("engine flush: %d new docs, %d total, %.1f MB HBM",
                 n_new, shard.n_docs, shard.hbm_bytes() / 1e6)
        return n_new

    # ------------------------------------------------------------ search
    def search(self, query: str, limit: int = 10,
               use_dense: bool | None = None) -> list[SearchHit]:
        """Single-query search against the engine (shard-fused)."""
        return self.search_many([query], limit=limit,
                                use_dense=use_dense)[0]

    def search_many(self, queries: list[str], limit: int = 10,
                    use_dense: bool | None = None) -> list[list[SearchHit]]:
        """Batched search: ONE collective plane.search_batch for the
        whole list (the batcher's execute path). Fusion happens exactly
        once, on the plane (RRF of the per-shard BM25 + dense top-k);
        callers hydrate url/title from the LocalStore."""
        if self.shard.n_docs == 0 or not queries:
            return [[] for _ in queries]
        B = len(queries)
        # pad the batch to a power-of-two bucket: the plane's hipGraph
        # cache, scores buffers and the encoder graphs are per-(B, k)
        # shape — unbounded serving batch sizes would thrash them
        Bp = 1
        while Bp < B:
            Bp *= 2
        terms = [bm25_term_ids(q) for q in queries]
        terms += [np.zeros(0, dtype=np.int64)] * (Bp - B)
        emb = None
        if use_dense is None:
            use_dense = self.encoder is not None
        if use_dense and self.encoder is not None:
            texts = queries + [""] * (Bp - B)
            emb = self.encoder.encode_texts(texts)
        fused = self.plane.search_batch(
            terms, emb, B=Bp, dim=emb.shape[1] if emb is not None else 384,
            n_results=limit, use_dense=use_dense and emb is not None)
        if fused is None:
            return [[] for _ in queries]
        ids = fused.ids.tolist()
        scores = fused.scores.tolist()
        out: list[list[SearchHit]] = []
        for qi in range(B):
            hits: list[SearchHit] = []
            for gid, score in zip(ids[qi], scores[qi]):
                if gid < 0:
                    continue
                hits.append(SearchHit(doc_id=int(gid), url="", title="",
                                      snippet="", bm25=0.0,
                                      score=float(score),
                                      source="gpu-hybrid"))
            out.append(hits)
        return out

    def stats(self) -> dict:
        return {
            "device": self.device,
            "docs_indexed": self.shard.n_docs,
            "docs_pending": self.pending_count,
            "hbm_bytes": self.shard.hbm_bytes(),
            "world_size": self.fabric.world,
            "encoder": self.encoder is not None,
        }
