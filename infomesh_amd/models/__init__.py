"""Model families (random-init weights, MI355X HIP-kernel forward paths):

- encoder: bge-small-en-v1.5-shaped text embedding encoder (dense index)
- reranker: bge-reranker-base-shaped cross-encoder (top-100 -> 10)
- phi3: Phi-3-mini-shaped decoder (in-process RAG summarizer)

These replace the reference's external models: sentence-transformers
MiniLM (infomesh/index/vector_store.py:26), LLM-prompt reranker
(infomesh/search/reranker.py:86) and Ollama/llama.cpp/vLLM HTTP backends
(infomesh/summarizer/engine.py:111-318).
"""
