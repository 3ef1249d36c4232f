"""In-process summarizer subsystem (reference parity: infomesh/summarizer/
— the external Ollama/llama.cpp/vLLM HTTP backends are replaced by an
in-process Phi-3-mini-shaped decoder on CDNA4 kernels)."""
