"""Search layer: query orchestration, RRF merge, NLP, passages, cache,
reranker, RAG (reference parity: infomesh/search/)."""
