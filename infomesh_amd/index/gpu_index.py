"""GPU hybrid index shard: segmented CSR inverted index (BM25) + dense
embeddings (cosine) resident in HBM3E, scored by hand-written CDNA4
kernels.

Replaces intra-node: SQLite FTS5 MATCH+bm25() (reference
infomesh/index/local_store.py:316-332) and ChromaDB HNSW
(infomesh/index/vector_store.py:92-254). One shard per GPU; documents
hash-partitioned across shards; the query plane (parallel/query_plane.py)
fans out and all-gathers top-k.

Segmented design (the FTS5 append/optimize analogue, reference
local_store.py:528-541): a flush appends ONE new posting segment built
only from the new docs — O(new), no host pull-back of old postings —
and `optimize()` merges segments GPU-side (torch sort over (term, doc)
keys). BM25 stays exact across segments because the per-posting doc
length travels packed with tf and the norm is computed in-kernel from
the CURRENT global avgdl; idf comes from the global df table summed
over segments. Segments partition the doc-id axis, so per-segment
score kernels write disjoint column ranges of the [B, N] score matrix
(each column exactly once — no zero-fill, no atomics across segments).

Shard sizing: 1.25M docs × (≈120 postings × 8 B + 384 × 2 B embedding)
≈ 2.2 GB — far under the 288 GB HBM budget, so shards scale to 100M+
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

# auto-merge threshold: beyond this many live segments a flush triggers
# optimize() (GPU-side merge) — the FTS5 automerge analogue
MAX_SEGMENTS = 16

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


@dataclass
class PostingSegment:
    """One immutable CSR posting segment covering the doc-id range
    [doc_base, doc_base + n_docs)."""
    offsets: torch.Tensor   # [V+1] i64 (device)
    doc_ids: torch.Tensor   # [P] i32, SEGMENT-local ids, asc per term
    tfdl: torch.Tensor      # [P] i32: tf | (dl << 16)
    doc_base: int
    n_docs: int
    h_offs: np.ndarray      # host copy of offsets (query-time chunking)

    def hbm_bytes(self) -> int:
        return sum(t.numel() * t.element_size()
                   for t in (self.offsets, self.doc_ids, self.tfdl))


class GpuShard:
    """One GPU's slice of the hybrid index."""

    def __init__(self, device: str = "cuda", vocab: int = BM25_VOCAB,
                 emb_dtype: str = "bf16"):
        assert emb_dtype in ("bf16", "fp8")
        self.device = torch.device(device)
        self.vocab = vocab
        # "fp8" stores embeddings as OCP e4m3 (half the HBM + half the
        # dense-plane read traffic; ~0.4% relative score error on
        # unit-norm vectors) — opt-in via config gpu.dtype
        self.emb_dtype = emb_dtype
        self.n_docs = 0
        self.segments: list[PostingSegment] = []
        self.df: np.ndarray = np.zeros(vocab, dtype=np.int64)  # global df
        self._doc_lens = np.zeros(0, dtype=np.int64)           # host [N]
        self.avgdl = 1.0
        # dense plane: capacity-growth device buffers ([:n_docs] live)
        self._emb_buf: torch.Tensor | None = None   # [cap, D] bf16 (unit)
        self._gid_buf: torch.Tensor | None = None   # [cap] i64
        self._topk = None
        self._h_idf: np.ndarray | None = None
        # pending (un-built) batch buffers — the ingest side-buffer;
        # GPU visibility flips at build() (epoch-style).
        self._pend_tokens: list[np.ndarray] = []
        self._pend_emb: list[torch.Tensor] = []
        self._pend_gids: list[int] = []

    # ------------------------------------------------- derived views
    @property
    def embeddings(self) -> torch.Tensor | None:
        if self._emb_buf is None or self.n_docs == 0:
            return None
        return self._emb_buf[:self.n_docs]

    @property
    def global_ids(self) -> torch.Tensor | None:
        if self._gid_buf is None:
            return None
        return self._gid_buf[:self.n_docs]

    @property
    def doc_norm(self) -> torch.Tensor:
        """BM25 length norm per doc (host-derived; kernels compute this
        in-flight from packed dl — kept for the CPU oracle/tests)."""
        dl = self._doc_lens.astype(np.float32)
        return torch.from_numpy(
            BM25_K1 * (1 - BM25_B + BM25_B * dl / self.avgdl))

    # ------------------------------------------------------------ build
    def add_document(self, global_id: int, term_ids: np.ndarray,
                     embedding: torch.Tensor | None) -> None:
        self._pend_tokens.append(term_ids.astype(np.int64))
        self._pend_gids.append(global_id)
        if embedding is not None:
            self._pend_emb.append(embedding.reshape(1, -1))

    def build(self) -> None:
        """Make pending docs searchable by appending ONE new posting
        segment — O(new docs), never O(corpus). Triggers a GPU-side
        optimize() merge when the segment count exceeds MAX_SEGMENTS
        (the FTS5 automerge analogue)."""
        if not self._pend_tokens:
            return
        token_lists = self._pend_tokens
        gids = list(self._pend_gids)
        embs = self._pend_emb
        self._pend_tokens, self._pend_gids, self._pend_emb = [], [], []
        new_emb = torch.cat(embs, 0) if embs else None
        self._append_segment(token_lists, gids, new_emb)
        if len(self.segments) > MAX_SEGMENTS:
            self.optimize()

    def merged_with(self, token_lists: list[np.ndarray],
                    gids: list[int],
                    new_emb: torch.Tensor | None) -> "GpuShard":
        """Return a NEW shard = this shard + the given docs, sharing
        the immutable segments/buffers with this one (epoch-flip:
        readers of the old shard never see the new docs because every
        read is bounded by the old n_docs). Linear use only — append to
        the RETURNED shard, not to this one, afterwards."""
        out = type(self)() if type(self).__init__ is not GpuShard.__init__ \
            else GpuShard(str(self.device), vocab=self.vocab,
                          emb_dtype=self.emb_dtype)
        out.vocab = self.vocab
        out.emb_dtype = self.emb_dtype
        out.n_docs = self.n_docs
        out.segments = list(self.segments)
        out.df = self.df.copy()
        out._doc_lens = self._doc_lens
        out.avgdl = self.avgdl
        out._emb_buf = self._emb_buf
        out._gid_buf = self._gid_buf
        out._append_segment(token_lists, gids, new_emb)
        return out

    def _append_segment(self, token_lists: list[np.ndarray],
                        gids: list, new_emb: torch.Tensor | None) -> None:
        lens2 = np.array([max(len(t), 1) for t in token_lists],
                         dtype=np.int64)
        flat_terms = (np.concatenate(
            [t.astype(np.int64) for t in token_lists])
            if any(len(t) for t in token_lists)
            else np.zeros(0, np.int64))
        flat_docs = np.repeat(np.arange(len(token_lists), dtype=np.int64),
                              [len(t) for t in token_lists])
        self._install_segment(flat_terms, flat_docs, lens2,
                              np.asarray(gids, dtype=np.int64), new_emb)

    def build_from_arrays(self, flat_terms: np.ndarray,
                          flat_docs: np.ndarray, doc_lens: np.ndarray,
                          global_ids: np.ndarray,
                          embeddings: torch.Tensor | None) -> None:
        """Bulk build from flat (term, doc-local-id) pairs — installs a
        single segment (doc ids must be 0..len(doc_lens)-1)."""
        self._install_segment(flat_terms, flat_docs,
                              np.asarray(doc_lens, dtype=np.int64),
                              np.asarray(global_ids, dtype=np.int64),
                              embeddings)

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
            return t.clone()
        pinned = t.pin_memory()
        with torch.cuda.stream(stream):
            return pinned.to(self.device, non_blocking=True)

    def _grow(self, buf: torch.Tensor | None, need: int,
              row_shape: tuple, dtype: torch.dtype) -> torch.Tensor:
        """Capacity-doubling device buffer growth (amortized O(1) per
        append; the old tensor stays valid for prior-epoch readers)."""
        if buf is None:
            cap = max(need, 256)
            return torch.empty((cap, *row_shape), device=self.device,
                               dtype=dtype)
        if buf.shape[0] >= need:
            return buf
        cap = max(need, 2 * buf.shape[0])
        new = torch.empty((cap, *row_shape), device=self.device,
                          dtype=dtype)
        new[:self.n_docs] = buf[:self.n_docs]
        return new

    def _install_segment(self, flat_terms: np.ndarray,
                         flat_docs: np.ndarray, doc_lens: np.ndarray,
                         global_ids: np.ndarray,
                         embeddings: torch.Tensor | None) -> None:
        n_new = len(doc_lens)
        if n_new == 0:
            return
        if self.n_docs > 0:
            assert (embeddings is not None) == (self._emb_buf is not None), \
                "dense/sparse mode must be consistent across appends"
        on_gpu = self.device.type == "cuda"
        stream = torch.cuda.Stream(self.device) if on_gpu else None
        if stream is not None:
            # new_emb may have been produced on the compute stream
            # (encoder output): order the side-stream copies after it
            stream.wait_stream(torch.cuda.current_stream(self.device))
        if on_gpu and len(flat_terms) > 2_000_000:
            # bulk builds aggregate ON-DEVICE: the host (term,doc) key
            # sort took ~120 s for the 10M-doc corpus (567M postings);
            # torch's GPU radix sort + unique_consecutive do it in
            # seconds. Small appends keep the host path (cheaper than
            # the round trip).
            seg, df_new = self._aggregate_device(
                flat_terms, flat_docs, doc_lens, n_new)
        else:
            terms_u, docs_u, tf_u = self._aggregate(flat_terms,
                                                    flat_docs, n_new)
            dl_u = np.minimum(doc_lens[docs_u.astype(np.int64)], 65535)
            tfdl = (tf_u.astype(np.uint32)
                    | (dl_u.astype(np.uint32) << np.uint32(16))
                    ).view(np.int32)
            df_new = np.bincount(terms_u,
                                 minlength=self.vocab).astype(np.int64)
            offsets = np.zeros(self.vocab + 1, dtype=np.int64)
            np.cumsum(df_new, out=offsets[1:])
            seg = PostingSegment(
                offsets=self._upload(offsets, stream),
                doc_ids=self._upload(docs_u.astype(np.int32), stream),
                tfdl=self._upload(tfdl, stream),
                doc_base=self.n_docs, n_docs=n_new, h_offs=offsets)
        # dense-plane buffers: append rows [n_docs : n_docs+n_new]
        need = self.n_docs + n_new
        self._gid_buf = self._grow(self._gid_buf, need, (), torch.int64)
        gid_t = self._upload(global_ids.astype(np.int64), stream)
        if embeddings is not None:
            e = embeddings
            store_dtype = (torch.float8_e4m3fn if self.emb_dtype == "fp8"
                           else torch.bfloat16)
            if e.dtype not in (torch.bfloat16, torch.float8_e4m3fn):
                e = torch.nn.functional.normalize(e.float(), dim=-1)
            if e.dtype != store_dtype:
                e = e.to(store_dtype)
            assert e.shape[0] == n_new, \
                "dense shard requires embeddings for every pending doc"
            self._emb_buf = self._grow(self._emb_buf, need,
                                       (e.shape[1],), store_dtype)
            if e.device.type == "cpu" and self.device.type == "cuda":
                e = self._upload_t(e, stream)
            if stream is not None:
                with torch.cuda.stream(stream):
                    self._emb_buf[self.n_docs:need].copy_(e)
                    self._gid_buf[self.n_docs:need].copy_(gid_t)
            else:
                self._emb_buf[self.n_docs:need].copy_(e)
                self._gid_buf[self.n_docs:need].copy_(gid_t)
        else:
            if stream is not None:
                with torch.cuda.stream(stream):
                    self._gid_buf[self.n_docs:need].copy_(gid_t)
            else:
                self._gid_buf[self.n_docs:need].copy_(gid_t)
        if stream is not None:
            # epoch flip: the segment becomes visible only after the
            # side stream's uploads complete on the compute stream.
            torch.cuda.current_stream(self.device).wait_stream(stream)
        self.segments.append(seg)
        self.df = self.df + df_new
        self._doc_lens = np.concatenate([self._doc_lens, doc_lens])
        self.avgdl = float(self._doc_lens.mean())
        self.n_docs = need
        self._invalidate_query_caches()

    def _aggregate_device(self, flat_terms: np.ndarray,
                          flat_docs: np.ndarray, doc_lens: np.ndarray,
                          n_new: int):
        """GPU bulk aggregation: (term, doc) pairs -> term-sorted CSR
        postings with packed tf|dl, entirely on-device."""
        dev = self.device
        t = torch.from_numpy(np.ascontiguousarray(flat_terms)).to(dev)
        d = torch.from_numpy(np.ascontiguousarray(flat_docs)).to(dev)
        key, _ = torch.sort(t * np.int64(max(n_new, 1)) + d)
        del t, d
        uniq, counts = torch.unique_consecutive(key, return_counts=True)
        del key
        terms_u = uniq // max(n_new, 1)
        docs_u = (uniq % max(n_new, 1)).to(torch.int32)
        del uniq
        tf = counts.clamp(max=65535).to(torch.int32)
        dl_dev = torch.from_numpy(
            np.minimum(doc_lens, 65535).astype(np.int64)).to(dev)
        dl = dl_dev[docs_u.long()].to(torch.int32)
        tfdl = (tf | (dl << 16)).contiguous()
        df_dev = torch.bincount(terms_u, minlength=self.vocab)
        offsets_dev = torch.zeros(self.vocab + 1, dtype=torch.int64,
                                  device=dev)
        torch.cumsum(df_dev, 0, out=offsets_dev[1:])
        df_new = df_dev.cpu().numpy().astype(np.int64)
        seg = PostingSegment(
            offsets=offsets_dev, doc_ids=docs_u.contiguous(),
            tfdl=tfdl, doc_base=self.n_docs, n_docs=n_new,
            h_offs=offsets_dev.cpu().numpy())
        return seg, df_new

    def _upload_t(self, t: torch.Tensor, stream) -> torch.Tensor:
        pinned = t.contiguous().pin_memory()
        with torch.cuda.stream(stream):
            return pinned.to(self.device, non_blocking=True)

    def _invalidate_query_caches(self) -> None:
        self._h_idf = None
        # dense hipGraphs captured the old embeddings view/shape
        self._dense_graphs = {}
        self._dense_bufs = {}
        self._bm25_hists = {}

    def optimize(self) -> None:
        """Merge all posting segments into one, entirely on-device:
        reconstruct (term, global-doc) keys, single torch.sort, gather
        tfdl — the FTS5 `optimize()` analogue (reference
        local_store.py:528-541). Docs never repeat across segments, so
        no tf re-aggregation is needed."""
        if len(self.segments) <= 1:
            return
        dev = self.device
        V = self.vocab
        term_parts, doc_parts, tfdl_parts = [], [], []
        arangeV = torch.arange(V, dtype=torch.int64, device=dev)
        for seg in self.segments:
            dfg = seg.offsets.diff()
            term_parts.append(torch.repeat_interleave(arangeV, dfg))
            doc_parts.append(seg.doc_ids.to(torch.int64) + seg.doc_base)
            tfdl_parts.append(seg.tfdl)
        terms = torch.cat(term_parts)
        docs = torch.cat(doc_parts)
        tfdl = torch.cat(tfdl_parts)
        key = terms * self.n_docs + docs
        key, order = torch.sort(key)
        docs_sorted = (key % self.n_docs).to(torch.int32)
        tfdl_sorted = tfdl[order].contiguous()
        offsets = torch.zeros(V + 1, dtype=torch.int64, device=dev)
        torch.cumsum(torch.from_numpy(self.df).to(dev), 0,
                     out=offsets[1:])
        self.segments = [PostingSegment(
            offsets=offsets, doc_ids=docs_sorted, tfdl=tfdl_sorted,
            doc_base=0, n_docs=self.n_docs,
            h_offs=offsets.cpu().numpy())]
        self._invalidate_query_caches()

    def hbm_bytes(self) -> int:
        total = sum(s.hbm_bytes() for s in self.segments)
        for t in (self._emb_buf, self._gid_buf):
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
        df = float(self.df[term])
        if df <= 0:
            return 0.0
        return math.log(1.0 + (self.n_docs - df + 0.5) / (df + 0.5))

    def _idf_table(self) -> np.ndarray:
        """Cached idf[V] f32 from the GLOBAL df (sum over segments)."""
        if self._h_idf is None:
            df = self.df.astype(np.float64)
            with np.errstate(divide="ignore"):
                idf = np.log(1.0 + (self.n_docs - df + 0.5) / (df + 0.5))
            self._h_idf = np.where(df > 0, idf, 0.0).astype(np.float32)
        return self._h_idf

    @staticmethod
    def dedupe_terms(queries_terms: list[np.ndarray]
                     ) -> tuple[np.ndarray, np.ndarray]:
        """Vectorized per-query term dedupe -> (qrows, terms), sorted by
        query row (the obvious per-query np.unique loop costs ~1 ms of
        host time at B=128, serializing the side-stream launch ahead of
        the encoder)."""
        B = len(queries_terms)
        if B == 0:
            return np.zeros(0, np.int64), np.zeros(0, np.int64)
        lens = np.fromiter((len(t) for t in queries_terms), np.int64, B)
        T = int(lens.max())
        if T == 0:
            return np.zeros(0, np.int64), np.zeros(0, np.int64)
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
        return qrows.astype(np.int64), srt[qrows, cols]

    def _pick_bd(self, B: int) -> int:
        """Doc-block size: 8K docs (32 KB LDS -> 5 workgroups/CU;
        measured 405 us vs 471/713 us for 16K/32K at 1.25M docs B=128,
        scripts/bm25_probe.py) unless that underfills the chip."""
        for bd in (8192, 4096):
            blocks = sum((s.n_docs + bd - 1) // bd for s in self.segments)
            if blocks * max(B, 1) >= 2048:
                return bd
        return 4096

    def _bounds_ws(self, n: int) -> torch.Tensor:
        """Device i32 workspace for the bounds pre-pass (reused across
        segments/batches; launches are stream-ordered so one suffices)."""
        ws = getattr(self, "_bounds_buf", None)
        if ws is None or ws.numel() < n:
            ws = self._bounds_buf = torch.empty(
                max(n, 1 << 16), device=self.device, dtype=torch.int32)
        return ws

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
        side stream to overlap with query encoding. One kernel launch
        per posting segment; the segments' disjoint doc ranges cover
        every column of the score matrix exactly once."""
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
        else:
            scores = torch.empty(B, N, device=dev, dtype=torch.float32)
        qrows, terms = self.dedupe_terms(queries_terms)
        # queries share Zipf-common terms: dedupe ACROSS queries so the
        # bounds pre-pass searches each term's postings once per block
        uterms, qt_ut = np.unique(terms, return_inverse=True)
        idf = self._idf_table()[terms]
        qt_off = np.zeros(B + 1, dtype=np.int64)
        np.cumsum(np.bincount(qrows, minlength=B), out=qt_off[1:])
        bd = self._pick_bd(B)
        U = len(uterms)
        # fused topk pass 1: the block kernel histograms every score it
        # writes, so the top-k select skips one full read of the [B, N]
        # array (~0.9 ms/batch at 10M docs)
        hists = getattr(self, "_bm25_hists", None)
        if hists is None:
            hists = self._bm25_hists = {}
        hist = hists.get(B)
        if hist is None:
            hist = hists[B] = torch.zeros(
                B * 256, device=dev, dtype=torch.int32)
        else:
            hist.zero_()
        tp = mark("shard.chunks", tp)
        # don't overwrite the pinned staging buffers while a prior
        # step's async H2D copy could still be in flight
        evt = getattr(self, "_h2d_evt", None)
        if evt is not None:
            evt.synchronize()
        qt_off_d = self._h2d("qt_off", qt_off, torch.int32)
        qt_ut_d = self._h2d("qt_ut", qt_ut, torch.int32)
        qt_idf_d = self._h2d("qt_idf", idf, torch.float32)
        for si, seg in enumerate(self.segments):
            nblocks = (seg.n_docs + bd - 1) // bd
            bounds = self._bounds_ws(U * nblocks * 2)
            K.bm25_block(
                seg.doc_ids, seg.tfdl, qt_off_d, qt_ut_d, qt_idf_d,
                self._h2d(f"qb{si}", seg.h_offs[uterms], torch.int64),
                self._h2d(f"qe{si}", seg.h_offs[uterms + 1], torch.int64),
                bounds, scores, seg.doc_base, seg.n_docs, bd,
                self.avgdl, k1=BM25_K1, b=BM25_B, hist1=hist)
        if dev.type == "cuda":
            if evt is None:
                evt = self._h2d_evt = torch.cuda.Event()
            evt.record()
        tp = mark("shard.bm25", tp)
        out = topk(scores, k, ext_hist1=hist)
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

        def mark(name, t0):
            if phase_t is None:
                return t0
            torch.cuda.synchronize()
            t1 = _time.perf_counter()
            phase_t[name] = phase_t.get(name, 0.0) + (t1 - t0)
            return t1

        B = len(queries_terms)
        N = self.n_docs
        dev = self.device
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
        emb = self.embeddings
        # hipGraph capture of the whole plane (GEMM + top-k passes, all
        # fixed shapes) removes ~10 launch gaps per batch. Only when the
        # caller runs deferred top-k verification (the query plane does):
        # the eager path reads the overflow flag (D2H), which is illegal
        # inside a capture. Graphs are invalidated on every append
        # (the captured embeddings view goes stale).
        if self.device.type == "cuda" and tk.defer_check:
            key = (B, k)
            graphs = getattr(self, "_dense_graphs", None)
            if graphs is None:
                graphs = self._dense_graphs = {}
            entry = graphs.get(key)
            if entry is None:
                from ..ops.graphs import GraphedCallable

                def _plane(e, _B=B, _k=k, _emb=emb):
                    d = self._dense_gemm(e, _emb, _B, self.n_docs)
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
                tk._pending.append((ws, K.topk_cnt_off(B),
                                    K.topk_flag_off(B), B, 0))
            # detach from the graph's static outputs (next replay
            # overwrites them)
            out = (vals.clone(), idx.clone())
            mark("shard.dense+topk", tp)
            return out
        d_scores = self._dense_gemm(query_emb, emb, B, N)
        tp = mark("shard.dense", tp)
        # Always exact select. The sampled-threshold variant was measured
        # a net loss at every shard size: the candidate slack (~Kp*stride)
        # inflates the final bitonic sort by more than the 2 saved passes
        # (~80us/pass at 1.25M docs), and with an 8192-sample + 8192-cap
        # it is only statistically sound for N <~ 2.7M anyway.
        out = tk(d_scores, k)
        mark("shard.densetopk", tp)
        return out

    def _dense_gemm(self, query_emb: torch.Tensor, emb: torch.Tensor,
                    B: int, N: int) -> torch.Tensor:
        """Dense cosine scores [B, N] on the stored-embedding dtype:
        bf16 -> generic MFMA tile; fp8 -> streaming fp8 kernel (M
        chunks of <=128; query quantized per batch)."""
        from ..ops import kernels as K
        buf = self._get_dense_buf(B)
        if self.emb_dtype == "fp8":
            qa = query_emb
            if qa.dtype != torch.float8_e4m3fn:
                qa = qa.float().to(torch.float8_e4m3fn)
            out2d = buf.reshape(B, N)
            for m0 in range(0, B, 128):
                m1 = min(m0 + 128, B)
                r = K.dense_scores_fp8(qa[m0:m1].contiguous(), emb,
                                       out=out2d[m0:m1])
                assert r is not None, "fp8 dense plane shape ineligible"
            return out2d
        return K.gemm_nt(query_emb.bfloat16().unsqueeze(0), emb,
                         out_f32=True, out=buf).reshape(B, N)

    def _get_dense_buf(self, B: int) -> torch.Tensor:
        """Persistent dense scores buffer, separate from the BM25 one
        (which may still be feeding its top-k on another stream). Keyed
        per batch size: captured hipGraphs keep writing the buffer they
        recorded, so a shared buffer reallocated for a different B
        would leave older graphs writing freed memory (the serving
        batcher sends pow2-bucketed batch sizes, so this cache is
        bounded). Reuse avoids a multi-GB alloc/free per batch."""
        N = self.n_docs
        bufs = getattr(self, "_dense_bufs", None)
        if bufs is None:
            bufs = self._dense_bufs = {}
        buf = bufs.get(B)
        if buf is None or buf.shape != (1, B, N):
            buf = bufs[B] = torch.empty(
                1, B, N, device=self.device, dtype=torch.float32)
        return buf

    def to_global(self, idx: torch.Tensor) -> torch.Tensor:
        safe = idx.clamp(min=0).long()
        g = self.global_ids[safe]
        return torch.where(idx >= 0, g, torch.full_like(g, -1))


class CpuShard(GpuShard):
    """CPU shard with identical semantics, scored by plain torch/numpy.

    Used on machines without a GPU (tests, the CPU-plumbing config, and
    multi-process gloo tests of the query plane). On a GPU box the HIP
    path is always taken — this class is never a silent GPU fallback."""

    def __init__(self, vocab: int = BM25_VOCAB, emb_dtype: str = "bf16"):
        super().__init__(device="cpu", vocab=vocab, emb_dtype=emb_dtype)

    def search(self, queries_terms, query_emb, k: int = 100,
               scores_buf=None, phase_t=None) -> ShardHits:
        B = len(queries_terms)
        N = self.n_docs
        assert N > 0, "shard is empty"
        k = min(k, N)
        norm = self.doc_norm.numpy()
        scores = np.zeros((B, N), dtype=np.float32)
        for seg in self.segments:
            offs = seg.h_offs
            doc_ids = seg.doc_ids.numpy()
            tfdl = seg.tfdl.numpy().view(np.uint32)
            tf_all = (tfdl & np.uint32(0xFFFF)).astype(np.float32)
            for qi, terms in enumerate(queries_terms):
                for t in np.unique(terms):
                    t = int(t)
                    b, e = int(offs[t]), int(offs[t + 1])
                    if b == e:
                        continue
                    idf = self._idf(t)
                    d = doc_ids[b:e].astype(np.int64) + seg.doc_base
                    tf = tf_all[b:e]
                    scores[qi, d] += (idf * tf * (BM25_K1 + 1)
                                      / (tf + norm[d]))
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
