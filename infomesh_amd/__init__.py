"""infomesh_amd — MI355X-native hybrid search/RAG engine.

A brand-new framework with the capabilities of dotnetpower/infomesh
(reference at /root/reference), re-designed for a single 8×MI355X node:

- Inverted (BM25) + dense (cosine) indexes sharded across GPUs, query
  fan-out and top-k merge via RCCL over xGMI (torch.distributed).
- Scoring hot paths — embedding encoder, cosine top-k, cross-encoder
  reranker, BM25 posting-list scorer, SimHash dedup, summarizer decode —
  are hand-written CDNA4 HIP kernels (MFMA + LDS tiling, gfx950).
- CPU-side capability surface (crawler, credits, trust, MCP/CLI/HTTP,
  LocalStore snapshot format) kept behavior-compatible with the
  reference (see SURVEY.md §2 for the component inventory).
"""

__version__ = "0.1.0"
