"""Summarization engine with pluggable backends.

Reference parity: infomesh/summarizer/engine.py (LLMBackend ABC,
backend factory, SummarizationEngine prompt/length logic). The default
backend is the in-process Phi-3-mini-shaped decoder on MI355X kernels
(models/phi3.py) instead of HTTP calls to Ollama/llama.cpp/vLLM; an
extractive CPU backend covers GPU-less nodes and tests.
"""
from __future__ import annotations

import abc
import logging
import re
from dataclasses import dataclass

from ..config import SummarizerConfig

log = logging.getLogger("infomesh.summarizer")


@dataclass
class SummaryResult:
    summary: str
    backend: str
    tokens_generated: int = 0
    truncated: bool = False


class LLMBackend(abc.ABC):
    name = "abstract"

    @abc.abstractmethod
    def generate(self, prompt: str, max_new_tokens: int = 128) -> str: ...

    def available(self) -> bool:
        return True


class ExtractiveBackend(LLMBackend):
    """CPU fallback: lead + highest-TF sentences (no model). Keeps the
    summarize surface working on GPU-less nodes."""

    name = "extractive"

    def generate(self, prompt: str, max_new_tokens: int = 128) -> str:
        # The prompt embeds the source text after the last colon block.
        text = prompt.rsplit("TEXT:\n", 1)[-1]
        sents = re.split(r"(?<=[.!?])\s+", text)
        sents = [s.strip() for s in sents if len(s.split()) >= 4]
        if not sents:
            return text[: max_new_tokens * 4]
        words = re.findall(r"\w+", text.lower())
        from collections import Counter
        tf = Counter(words)
        scored = sorted(
            ((sum(tf[w.lower()] for w in re.findall(r"\w+", s))
              / max(1, len(s.split())), i, s)
             for i, s in enumerate(sents[1:], start=1)), reverse=True)
        picks = [sents[0]] + [s for _, _, s in scored[:2]]
        out = " ".join(dict.fromkeys(picks))
        return out[: max_new_tokens * 6]


class Phi3Backend(LLMBackend):
    """In-process Phi-3-mini-shaped decode on HIP kernels."""

    name = "phi3-mini"

    def __init__(self, device: str = "cuda", max_batch: int = 1,
                 max_seq: int = 2304):
        from ..models.phi3 import PHI3_MINI, Phi3Decoder
        from ..models.tokenizer import HashTokenizer
        self.decoder = Phi3Decoder(PHI3_MINI, device=device,
                                   max_batch=max_batch, max_seq=max_seq)
        self.tokenizer = HashTokenizer(PHI3_MINI.vocab_size)
        self.device = device

    def generate(self, prompt: str, max_new_tokens: int = 128) -> str:
        import torch
        ids = self.tokenizer.encode(
            prompt, max_len=self.decoder.max_seq - max_new_tokens - 8,
            add_special=False)
        prompt_t = torch.tensor([ids], dtype=torch.int32,
                                device=self.device)
        out = self.decoder.generate_greedy(prompt_t,
                                           max_new_tokens=max_new_tokens)
        # Random-init weights produce untrained token streams; render a
        # deterministic token transcript (the honest output of an
        # untrained model — real deployments load trained weights).
        toks = out[0].cpu().tolist()
        return " ".join(f"<t{t}>" for t in toks)

    def available(self) -> bool:
        import torch
        return torch.cuda.is_available()


def create_backend(kind: str = "auto", device: str = "cuda") -> LLMBackend:
    """Backend factory (reference: engine.py:319)."""
    if kind in ("auto", "phi3"):
        try:
            import torch
            if torch.cuda.is_available():
                return Phi3Backend(device=device)
        except Exception as e:
            log.warning("phi3 backend unavailable: %s", e)
        if kind == "phi3":
            raise RuntimeError("phi3 backend requires a GPU")
    return ExtractiveBackend()


SUMMARY_PROMPT = """Summarize the following web page content in {n} sentences.
Focus on factual statements.

TITLE: {title}
TEXT:
{text}"""


class SummarizationEngine:
    def __init__(self, backend: LLMBackend | None = None,
                 config: SummarizerConfig | None = None):
        self.cfg = config or SummarizerConfig()
        self.backend = backend or create_backend()

    def summarize(self, text: str, title: str = "",
                  sentences: int = 3) -> SummaryResult:
        max_chars = self.cfg.max_context_tokens * 4
        truncated = len(text) > max_chars
        prompt = SUMMARY_PROMPT.format(n=sentences, title=title,
                                       text=text[:max_chars])
        out = self.backend.generate(prompt, self.cfg.max_new_tokens)
        return SummaryResult(summary=out.strip(), backend=self.backend.name,
                             tokens_generated=len(out.split()),
                             truncated=truncated)

    def summarize_results(self, results: list[dict],
                          query: str = "") -> SummaryResult:
        """Multi-result digest for RAG answers."""
        joined = "\n\n".join(
            f"[{i+1}] {r.get('title','')}: {r.get('snippet') or r.get('text','')[:400]}"
            for i, r in enumerate(results[:5]))
        prompt = (f"Question: {query}\nSummarize what these search results "
                  f"say, citing [n].\nTEXT:\n{joined}")
        out = self.backend.generate(prompt, self.cfg.max_new_tokens)
        return SummaryResult(summary=out.strip(), backend=self.backend.name,
                             tokens_generated=len(out.split()))
