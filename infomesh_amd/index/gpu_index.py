"""GPU hybrid index shard: CSR inverted index (BM25) + dense embeddings
(cosine) resident in HBM3E, scored by hand-written CDNA4 kernels.

Replaces intra-node: SQLite FTS5 MATCH+bm25() (reference
infomesh/index/local_store.py:316-332) and ChromaDB HNSW
(infomesh/index/vector_store.py:92-254). One shard per GPU; documents
hash-partitioned across shards; the query plane (parallel/query_plane.py)
fans out and all-gathers top-k.

Shard sizing: 1.25M docs × (≈120 postings × 6 B + 384 × 2 B embedding)
≈ 1.9 GB — far under the 288 GB HBM budget, so shards scale to 100M+
docs per GPU; the bench uses the BASELINE 10M-doc/8-GPU config.
"""
from __future__ import annotations

import math
import re
from dataclasses import dataclass

import numpy as np
import torch

from ..hashing import hash64

BM25_K1 = 1.2
BM25_B = 0.75
BM25_VOCAB = 1 << 17

# \w+ covers all unicode letters/digits (FTS5 unicode61 analogue);
# diacritics are NFKD-folded below like unicode61 remove_diacritics
_WORD_RE = re.compile(r"\w+")


def bm25_term_ids(text: str, vocab: int = BM25_VOCAB) -> np.ndarray:
    """Tokenize + hash into the BM25 term space. A registered
    'tokenizer' plugin (utils/plugins.py, reference dx.py:153-188)
    replaces the default word tokenizer for BOTH indexing and queries
    — the hash space keeps them consistent automatically.

    CJK runs are expanded into character bigrams (the GPU analogue of
    the reference's FTS5 trigram/CJK handling, search/cjk.py): the
    default latin word regex alone would drop CJK text entirely from
    the GPU BM25 plane."""
    from ..utils.plugins import GLOBAL_PLUGINS
    tok = GLOBAL_PLUGINS.get_single("tokenizer")
    if tok is not None:
        toks = tok(text)
    else:
        if any(ord(c) > 127 for c in text):
            # diacritic folding (cafe == café), matching the FTS5
            # unicode61 tokenizer's remove_diacritics so both score
            # planes see the same terms
            import unicodedata
            text = "".join(c for c in unicodedata.normalize("NFKD", text)
                           if not unicodedata.combining(c))
        toks = _WORD_RE.findall(text.lower())
        # fast ordinal check before paying the regex: any char >= U+2E80
        if any(ord(c) >= 0x2E80 for c in text):
            from ..search.cjk import _CJK_RUN_RE, ngram_expand
            for m in _CJK_RUN_RE.finditer(text):
                toks.extend(ngram_expand(m.group(0), 2))
    if not toks:
        return np.zeros(0, dtype=np.int64)
    return np.fromiter((hash64(t) % vocab for t in toks), dtype=np.int64,
                       count=len(toks))


@dataclass
class ShardHits:
    """Fixed-size per-shard top-k results (scores + GLOBAL doc ids)."""
    bm25_scores: torch.Tensor   # [B, k] f32
    bm25_ids: torch.Tensor      # [B, k] i64 (-1 pad)
    dense_scores: torch.Tensor  # [B, k] f32
    dense_ids: torch.Tensor     # [B, k] i64


class GpuShard:
    """One GPU's slice of the hybrid index."""

    def __init__(self, device: str = "cuda", vocab: int = BM25_VOCAB):
        self.device = torch.device(device)
        self.vocab = vocab
        self.n_docs = 0
        # CSR postings
        self.offsets: torch.Tensor | None = None   # [V+1] i64
        self.doc_ids: torch.Tensor | None = None   # [P] i32 (local ids)
        self.tfs: torch.Tensor | None = None       # [P] i16
        self.doc_norm: torch.Tensor | None = None  # [N] f32
        self.df: np.ndarray | None = None          # [V] i64 (host)
        self.avgdl = 1.0
        # dense
        self.embeddings: torch.Tensor | None = None  # [N, D] bf16 (unit)
        # local idx -> global doc id
        self.global_ids: torch.Tensor | None = None  # [N] i64
        self._topk = None
        # pending (un-built) batch buffers — the ingest side-buffer;
        # GPU visibility flips at build() (epoch-style).
        self._pend_tokens: list[np.ndarray] = []
        self._pend_emb: list[torch.Tensor] = []
        self._pend_gids: list[int] = []

    # ------------------------------------------------------------ build
    def add_document(self, global_id: int, term_ids: np.ndarray,
                     embedding: torch.Tensor | None) -> None:
        self._pend_tokens.append(term_ids.astype(np.int64))
        self._pend_gids.append(global_id)
        if embedding is not None:
            self._pend_emb.append(embedding.reshape(1, -1))

    def build(self) -> None:
        """(Re)build the CSR postings + embedding matrix from pending
        docs plus any existing index — the FTS5-optimize / segment-merge
        analogue. With existing docs this MERGES: the old postings are
        pulled back once, concatenated with the new aggregated postings
        (new docs get fresh local ids, so (term, doc) pairs never
        collide), and BM25 stats (avgdl, norms) are recomputed over the
        whole corpus."""
        if not self._pend_tokens:
            return
        token_lists = self._pend_tokens
        gids = list(self._pend_gids)
        embs = self._pend_emb
        self._pend_tokens, self._pend_gids, self._pend_emb = [], [], []
        lens2 = np.array([max(len(t), 1) for t in token_lists],
                         dtype=np.int64)
        n_old = self.n_docs
        n_new = len(token_lists)
        flat_terms = (np.concatenate(token_lists)
                      if any(len(t) for t in token_lists)
                      else np.zeros(0, np.int64))
        flat_docs = np.repeat(np.arange(n_new, dtype=np.int64),
                              [len(t) for t in token_lists])
        new_emb = torch.cat(embs, 0) if embs else None
        if n_old == 0:
            self.build_from_arrays(flat_terms, flat_docs, lens2,
                                   np.asarray(gids, dtype=np.int64),
                                   new_emb)
            return
        self._merge_install(self, flat_terms, flat_docs, lens2, gids,
                            new_emb)

    def merged_with(self, token_lists: list[np.ndarray],
                    gids: list[int],
                    new_emb: torch.Tensor | None) -> "GpuShard":
        """Return a NEW shard = this shard + the given docs (epoch-flip
        variant of the in-place incremental build(): readers of the old
        shard are never exposed to a partially-built index)."""
        lens2 = np.array([max(len(t), 1) for t in token_lists],
                         dtype=np.int64)
        flat_terms = (np.concatenate(
            [t.astype(np.int64) for t in token_lists])
            if any(len(t) for t in token_lists)
            else np.zeros(0, np.int64))
        flat_docs = np.repeat(np.arange(len(token_lists), dtype=np.int64),
                              [len(t) for t in token_lists])
        out = type(self)() if type(self).__init__ is not GpuShard.__init__ \
            else GpuShard(str(self.device), vocab=self.vocab)
        out.vocab = self.vocab
        self._merge_install(out, flat_terms, flat_docs, lens2,
                            list(gids), new_emb)
        return out

    def _merge_install(self, target: "GpuShard", flat_terms, flat_docs,
                       lens2, gids, new_emb) -> None:
        """Merge THIS shard's postings with new aggregated postings and
        install into `target` (which may be self)."""
        n_old = self.n_docs
        n_new = len(lens2)
        t2, d2, tf2 = self._aggregate(flat_terms, flat_docs, n_new)
        d2 = (d2.astype(np.int64) + n_old).astype(np.int32)
        # pull the OLD postings back (CSR -> flat aggregated form)
        t1 = np.repeat(np.arange(self.vocab, dtype=np.int64), self.df)
        d1 = self.doc_ids.cpu().numpy()
        tf1 = self.tfs.cpu().numpy().view(np.uint16)
        terms = np.concatenate([t1, t2])
        order = np.argsort(terms, kind="stable")
        terms = terms[order]
        docs = np.concatenate([d1, d2])[order]
        tfs = np.concatenate([tf1, tf2])[order]
        lens = np.concatenate([self._doc_lens, lens2])
        old_gids = self.global_ids.cpu().numpy()
        all_gids = np.concatenate(
            [old_gids, np.asarray(gids, dtype=np.int64)])
        emb = None
        if self.embeddings is not None:
            assert new_emb is not None and new_emb.shape[0] == n_new, \
                "dense shard requires embeddings for every pending doc"
            e2 = new_emb
            if e2.dtype != torch.bfloat16:
                e2 = torch.nn.functional.normalize(e2.float(),
                                                   dim=-1).bfloat16()
            emb = torch.cat([self.embeddings, e2.to(self.device)], 0)
        target.n_docs = n_old + n_new
        target._install_postings(terms, docs, tfs, lens, all_gids, emb)

    @staticmethod
    def _aggregate(flat_terms: np.ndarray, flat_docs: np.ndarray,
                   n: int) -> tuple[np.ndarray, np.ndarray, np.ndarray]:
        """(term, doc) pairs -> unique pairs + tf counts, term-sorted."""
        key = flat_terms * np.int64(max(n, 1)) + flat_docs
        key.sort(kind="stable")
        uniq, counts = np.unique(key, return_counts=True)
        terms_u = (uniq // max(n, 1)).astype(np.int64)
        docs_u = (uniq % max(n, 1)).astype(np.int32)
        tf_u = np.minimum(counts, 65535).astype(np.uint16)
        return terms_u, docs_u, tf_u

    def _upload(self, arr: np.ndarray, stream) -> torch.Tensor:
        """Host -> HBM via a pinned staging buffer on the ingest side
        stream (hipMemcpyAsync under the hood) so uploads overlap any
        query work on the compute stream — SURVEY.md §5.8 ingest path."""
        t = torch.from_numpy(np.ascontiguousarray(arr))
        if stream is None:
            return t.to(self.device)
        pinned = t.pin_memory()
        with torch.cuda.stream(stream):
            return pinned.to(self.device, non_blocking=True)

    def build_from_arrays(self, flat_terms: np.ndarray,
                          flat_docs: np.ndarray, doc_lens: np.ndarray,
                          global_ids: np.ndarray,
                          embeddings: torch.Tensor | None) -> None:
        """Bulk build from flat (term, doc) pairs. Dedups (term, doc)
        into term frequencies, sorts into CSR by term. Device uploads
        go through pinned buffers on a dedicated ingest stream."""
        n = len(doc_lens)
        self.n_docs = n
        # Aggregate tf per (term, doc) via a combined key sort.
        terms_u, docs_u, tf_u = self._aggregate(flat_terms, flat_docs, n)
        self._install_postings(terms_u, docs_u, tf_u, doc_lens,
                               global_ids, embeddings)

    def _install_postings(self, terms_u: np.ndarray, docs_u: np.ndarray,
                          tf_u: np.ndarray, doc_lens: np.ndarray,
                          global_ids: np.ndarray,
                          embeddings: torch.Tensor | None) -> None:
        """Install term-sorted aggregated postings as the CSR index and
        recompute BM25 stats; uploads via pinned staging."""
        n = self.n_docs
        self._doc_lens = np.asarray(doc_lens, dtype=np.int64)
        # CSR offsets per term (terms_u already sorted).
        df = np.bincount(terms_u, minlength=self.vocab).astype(np.int64)
        offsets = np.zeros(self.vocab + 1, dtype=np.int64)
        np.cumsum(df, out=offsets[1:])
        self.df = df
        on_gpu = self.device.type == "cuda"
        stream = torch.cuda.Stream(self.device) if on_gpu else None
        self.offsets = self._upload(offsets, stream)
        self.doc_ids = self._upload(docs_u, stream)
        self.tfs = self._upload(tf_u.astype(np.int16), stream)
        self.avgdl = float(doc_lens.mean()) if n else 1.0
        norm = BM25_K1 * (1 - BM25_B + BM25_B *
                          doc_lens.astype(np.float32) / self.avgdl)
        self.doc_norm = self._upload(norm, stream)
        self.global_ids = self._upload(global_ids.astype(np.int64), stream)
        if embeddings is not None:
            assert embeddings.shape[0] == n
            if embeddings.device.type == "cpu":
                e = embeddings
                if e.dtype != torch.bfloat16:
                    e = torch.nn.functional.normalize(
                        e.float(), dim=-1).bfloat16()
                if stream is not None:
                    pinned = e.contiguous().pin_memory()
                    with torch.cuda.stream(stream):
                        e = pinned.to(self.device, non_blocking=True)
                else:
                    e = e.to(self.device)
            else:
                e = embeddings.to(self.device)
                if e.dtype != torch.bfloat16:
                    e = torch.nn.functional.normalize(
                        e.float(), dim=-1).bfloat16()
            self.embeddings = e.contiguous()
        if stream is not None:
            # epoch flip: the shard becomes visible only after the side
            # stream's uploads complete on the compute stream.
            torch.cuda.current_stream(self.device).wait_stream(stream)

    def hbm_bytes(self) -> int:
        total = 0
        for t in (self.offsets, self.doc_ids, self.tfs, self.doc_norm,
                  self.embeddings, self.global_ids):
            if t is not None:
                total += t.numel() * t.element_size()
        return total

    # ----------------------------------------------------------- search
    def _get_topk(self):
        from ..ops.kernels import TopK
        if self._topk is None:
            self._topk = TopK(self.device)
        return self._topk

    def _get_topk_dense(self):
        """Separate selector (own workspace) so the dense plane can run
        on a different stream than the BM25 plane without racing."""
        from ..ops.kernels import TopK
        if getattr(self, "_topk_dense", None) is None:
            self._topk_dense = TopK(self.device)
        return self._topk_dense

    def _idf(self, term: int) -> float:
        df = float(self.df[term]) if self.df is not None else 0.0
        if df <= 0:
            return 0.0
        return math.log(1.0 + (self.n_docs - df + 0.5) / (df + 0.5))

    def _host_tables(self):
        """Cached host copies of offsets/df + idf table for chunking."""
        if not hasattr(self, "_h_offs") or self._h_offs is None:
            self._h_offs = self.offsets.cpu().numpy()
            df = self.df.astype(np.float64)
            with np.errstate(divide="ignore"):
                idf = np.log(1.0 + (self.n_docs - df + 0.5) / (df + 0.5))
            self._h_idf = np.where(df > 0, idf, 0.0).astype(np.float32)
        return self._h_offs, self._h_idf

    def bm25_chunks(self, queries_terms: list[np.ndarray],
                    chunk_size: int = 2048):
        """Vectorized host-side work chunking into (qrow, term, offset,
        idf) arrays — one work chunk per <=chunk_size posting slice."""
        offs, idf_t = self._host_tables()
        # Vectorized per-query term dedupe (the obvious per-query
        # np.unique loop costs ~1 ms of host time at B=128, serializing
        # the side-stream launch ahead of the encoder).
        B = len(queries_terms)
        if B:
            lens = np.fromiter((len(t) for t in queries_terms),
                               np.int64, B)
            T = int(lens.max()) if B else 0
        if B == 0 or T == 0:
            qrows = np.zeros(0, np.int64)
            terms = np.zeros(0, np.int64)
        else:
            if (lens == lens[0]).all():
                mat = np.stack(queries_terms).astype(np.int64, copy=False)
            else:
                mat = np.full((B, T), -1, dtype=np.int64)
                for qi, t in enumerate(queries_terms):
                    mat[qi, :len(t)] = t
            srt = np.sort(mat, axis=1)
            valid = srt >= 0
            valid[:, 1:] &= srt[:, 1:] != srt[:, :-1]
            qrows, cols = np.nonzero(valid)
            terms = srt[qrows, cols]
        begins = offs[terms]
        ends = offs[terms + 1]
        nchunks = np.maximum((ends - begins + chunk_size - 1) // chunk_size, 0)
        keep = nchunks > 0
        qrows, terms, begins, nchunks = (qrows[keep], terms[keep],
                                         begins[keep], nchunks[keep])
        if len(terms) == 0:
            return (np.zeros(0, np.int32), np.zeros(0, np.int32),
                    np.zeros(0, np.int64), np.zeros(0, np.float32))
        reps = nchunks.astype(np.int64)
        cq = np.repeat(qrows, reps).astype(np.int32)
        ct = np.repeat(terms, reps).astype(np.int32)
        base = np.repeat(begins, reps)
        # intra-term chunk index: arange within each repeated group
        total = int(reps.sum())
        grp_end = np.cumsum(reps)
        grp_start = grp_end - reps
        intra = np.arange(total, dtype=np.int64) - np.repeat(grp_start, reps)
        co = base + intra * chunk_size
        ci = idf_t[ct]
        return cq, ct, co, ci

    def _h2d(self, name: str, arr: np.ndarray,
             dtype: torch.dtype) -> torch.Tensor:
        """Stage a small host array through a persistent pinned buffer —
        pageable-memory copies block the host and defeat the side-stream
        overlap; pinned + non_blocking stays truly async."""
        t = torch.from_numpy(np.ascontiguousarray(arr)).to(dtype)
        if self.device.type != "cuda":
            return t
        pins = getattr(self, "_h2d_pins", None)
        if pins is None:
            pins = self._h2d_pins = {}
        n = t.numel()
        buf = pins.get(name)
        if buf is None or buf.numel() < n:
            buf = torch.empty(max(2 * n, 4096), dtype=dtype,
                              pin_memory=True)
            pins[name] = buf
        buf[:n].copy_(t)
        return buf[:n].to(self.device, non_blocking=True)

    def search_bm25(self, queries_terms: list[np.ndarray], k: int,
                    scores_buf: torch.Tensor | None = None,
                    mark=None) -> tuple[torch.Tensor, torch.Tensor]:
        """BM25 plane only (needs terms, not embeddings) — callable on a
        side stream to overlap with query encoding."""
        import time as _time
        from ..ops import kernels as K
        if mark is None:
            def mark(name, t0):
                return t0
        B = len(queries_terms)
        dev = self.device
        N = self.n_docs
        k = min(k, N)
        topk = self._get_topk()
        tp = _time.perf_counter()
        if scores_buf is not None and scores_buf.shape == (B, N):
            scores = scores_buf
            scores.zero_()
        else:
            scores = torch.zeros(B, N, device=dev, dtype=torch.float32)
        cq, ct, co, ci = self.bm25_chunks(queries_terms)
        tp = mark("shard.chunks", tp)
        if len(cq):
            # don't overwrite the pinned staging buffers while a prior
            # step's async H2D copy could still be in flight
            evt = getattr(self, "_h2d_evt", None)
            if evt is not None:
                evt.synchronize()
            K.bm25_score(
                self.offsets, self.doc_ids, self.tfs, self.doc_norm,
                self._h2d("cq", cq, torch.int32),
                self._h2d("ct", ct, torch.int32),
                self._h2d("co", co, torch.int64),
                self._h2d("ci", ci, torch.float32),
                scores, k1=BM25_K1)
            if dev.type == "cuda":
                if evt is None:
                    evt = self._h2d_evt = torch.cuda.Event()
                evt.record()
        tp = mark("shard.bm25", tp)
        out = topk(scores, k)
        mark("shard.bm25topk", tp)
        self._bm25_scores_buf = scores
        return out

    def search(self, queries_terms: list[np.ndarray],
               query_emb: torch.Tensor | None, k: int = 100,
               scores_buf: torch.Tensor | None = None,
               phase_t: dict | None = None) -> ShardHits:
        """Score all queries against this shard; returns fixed-size
        top-k with global ids (the per-peer result cap analogue,
        reference p2p/routing.py:48)."""
        import time as _time
        from ..ops import kernels as K

        def mark(name, t0):
            if phase_t is None:
                return t0
            torch.cuda.synchronize()
            t1 = _time.perf_counter()
            phase_t[name] = phase_t.get(name, 0.0) + (t1 - t0)
            return t1

        B = len(queries_terms)
        N = self.n_docs
        assert N > 0, "shard is empty"
        k = min(k, N)
        tp = _time.perf_counter()
        bm_vals, bm_idx = self.search_bm25(queries_terms, k,
                                           scores_buf=scores_buf,
                                           mark=mark)
        tp = _time.perf_counter()

        # --- dense plane ---
        if query_emb is not None and self.embeddings is not None:
            dn_vals, dn_idx = self.search_dense(query_emb, k, mark=mark)
        else:
            dn_vals = torch.full((B, k), -float("inf"), device=dev)
            dn_idx = torch.full((B, k), -1, device=dev, dtype=torch.int32)

        return ShardHits(bm25_scores=bm_vals,
                         bm25_ids=self.to_global(bm_idx),
                         dense_scores=dn_vals,
                         dense_ids=self.to_global(dn_idx))

    def search_dense(self, query_emb: torch.Tensor, k: int,
                     mark=None) -> tuple[torch.Tensor, torch.Tensor]:
        """Dense (cosine) plane; runs after the query embedding exists."""
        import time as _time
        from ..ops import kernels as K
        if mark is None:
            def mark(name, t0):
                return t0
        B = query_emb.shape[0]
        N = self.n_docs
        k = min(k, N)
        tp = _time.perf_counter()
        tk = self._get_topk_dense()
        # hipGraph capture of the whole plane (GEMM + top-k passes, all
        # fixed shapes) removes ~10 launch gaps per batch. Only when the
        # caller runs deferred top-k verification (the query plane does):
        # the eager path reads the overflow flag (D2H), which is illegal
        # inside a capture.
        if self.device.type == "cuda" and tk.defer_check:
            key = (B, k)
            graphs = getattr(self, "_dense_graphs", None)
            if graphs is None:
                graphs = self._dense_graphs = {}
            entry = graphs.get(key)
            if entry is None:
                from ..ops.graphs import GraphedCallable

                def _plane(emb, _B=B, _k=k):
                    d = K.gemm_nt(emb.unsqueeze(0), self.embeddings,
                                  out_f32=True,
                                  out=self._get_dense_buf(_B)
                                  ).reshape(_B, self.n_docs)
                    return tk(d, _k)
                g = GraphedCallable(_plane)
                vals, idx = g(query_emb.bfloat16())  # captures
                # pin the capture-time workspace: replays write THIS
                # tensor even if tk._ws is later reallocated
                entry = (g, tk._ws)
                graphs[key] = entry
            else:
                g, ws = entry
                vals, idx = g(query_emb.bfloat16())
                # Graph replay skips the wrapper's python, so re-arm
                # the deferred overflow check by hand (same static
                # workspace/layout as at capture; need=0 -> exact).
                cnt_off = B * 512 + 3 * B
                flag_off = B * 512 + 4 * B
                tk._pending.append((ws, cnt_off, flag_off, B, 0))
            # detach from the graph's static outputs (next replay
            # overwrites them)
            out = (vals.clone(), idx.clone())
            mark("shard.dense+topk", tp)
            return out
        d_scores = K.gemm_nt(query_emb.bfloat16().unsqueeze(0),
                             self.embeddings,
                             out_f32=True,
                             out=self._get_dense_buf(B)).reshape(B, N)
        tp = mark("shard.dense", tp)
        # Always exact select. The sampled-threshold variant was measured
        # a net loss at every shard size: the candidate slack (~Kp*stride)
        # inflates the final bitonic sort by more than the 2 saved passes
        # (~80us/pass at 1.25M docs), and with an 8192-sample + 8192-cap
        # it is only statistically sound for N <~ 2.7M anyway.
        out = tk(d_scores, k)
        mark("shard.densetopk", tp)
        return out

    def _get_dense_buf(self, B: int) -> torch.Tensor:
        """Persistent dense scores buffer, separate from the BM25 one
        (which may still be feeding its top-k on another stream). Reuse
        avoids a multi-GB alloc/free per batch at large N."""
        N = self.n_docs
        if (getattr(self, "_dense_scores_buf", None) is None
                or self._dense_scores_buf.shape != (1, B, N)):
            self._dense_scores_buf = torch.empty(
                1, B, N, device=self.device, dtype=torch.float32)
        return self._dense_scores_buf

    def to_global(self, idx: torch.Tensor) -> torch.Tensor:
        safe = idx.clamp(min=0).long()
        g = self.global_ids[safe]
        return torch.where(idx >= 0, g, torch.full_like(g, -1))


class CpuShard(GpuShard):
    """CPU shard with identical semantics, scored by plain torch ops.

    Used on machines without a GPU (tests, the CPU-plumbing config, and
    multi-process gloo tests of the query plane). On a GPU box the HIP
    path is always taken — this class is never a silent GPU fallback."""

    def __init__(self, vocab: int = BM25_VOCAB):
        super().__init__(device="cpu", vocab=vocab)

    def search(self, queries_terms, query_emb, k: int = 100,
               scores_buf=None, phase_t=None) -> ShardHits:
        B = len(queries_terms)
        N = self.n_docs
        assert N > 0, "shard is empty"
        k = min(k, N)
        offs = self.offsets.numpy()
        doc_ids = self.doc_ids.numpy()
        tfs = self.tfs.numpy().astype(np.float32)
        tfs = np.where(tfs < 0, tfs + 65536.0, tfs)  # stored as i16
        norm = self.doc_norm.numpy()
        scores = np.zeros((B, N), dtype=np.float32)
        for qi, terms in enumerate(queries_terms):
            for t in np.unique(terms):
                t = int(t)
                b, e = int(offs[t]), int(offs[t + 1])
                if b == e:
                    continue
                idf = self._idf(t)
                d = doc_ids[b:e]
                tf = tfs[b:e]
                scores[qi, d] += idf * tf * (BM25_K1 + 1) / (tf + norm[d])
        st = torch.from_numpy(scores)
        bm_vals, bm_idx = torch.topk(st, k, dim=1)
        if query_emb is not None and self.embeddings is not None:
            d_scores = query_emb.float().cpu() @ self.embeddings.float().T
            dn_vals, dn_idx = torch.topk(d_scores, k, dim=1)
        else:
            dn_vals = torch.full((B, k), -float("inf"))
            dn_idx = torch.full((B, k), -1, dtype=torch.int64)

        def to_global(idx):
            safe = idx.clamp(min=0).long()
            g = self.global_ids[safe]
            return torch.where(idx >= 0, g, torch.full_like(g, -1))

        return ShardHits(bm25_scores=bm_vals, bm25_ids=to_global(bm_idx),
                         dense_scores=dn_vals.float(),
                         dense_ids=to_global(dn_idx))
