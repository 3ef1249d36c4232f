"""RAG output formatting, answer extraction, entity extraction.

Reference parity: infomesh/search/rag.py (RAGChunk/RAGOutput,
format_rag_output chunking, answer extraction with confidence,
multi-result summary, entity extraction, toxicity filter).
"""
from __future__ import annotations

import re
from dataclasses import dataclass, field

from .passage import split_passages, score_passage, classify_intent

_WORD_RE = re.compile(r"\w+")
_TOXIC = frozenset("kill murder bomb terrorist nazi rape".split())


@dataclass
class RAGChunk:
    text: str
    url: str
    title: str
    score: float
    chunk_index: int = 0


@dataclass
class RAGOutput:
    chunks: list[RAGChunk] = field(default_factory=list)
    answer: str = ""
    confidence: float = 0.0
    entities: list[str] = field(default_factory=list)
    summary: str = ""


def chunk_results(query: str, results: list[dict],
                  chunk_size: int = 512, max_chunks: int = 8
                  ) -> list[RAGChunk]:
    """Split result texts into query-scored chunks of ~chunk_size chars
    (reference: rag.py:62-120)."""
    terms = _WORD_RE.findall(query)
    chunks: list[RAGChunk] = []
    for r in results:
        text = r.get("text") or r.get("snippet") or ""
        url = r.get("url", "")
        title = r.get("title", "")
        for i, p in enumerate(split_passages(
                text, target_words=chunk_size // 6,
                max_words=chunk_size // 4)):
            chunks.append(RAGChunk(
                text=p.text[:chunk_size], url=url, title=title,
                score=score_passage(terms, p.text), chunk_index=i))
    chunks.sort(key=lambda c: -c.score)
    return chunks[:max_chunks]


def extract_answer(query: str, results: list[dict]
                   ) -> tuple[str, float]:
    """Best-passage answer + confidence (reference answer extraction)."""
    chunks = chunk_results(query, results, max_chunks=3)
    if not chunks:
        return "", 0.0
    best = chunks[0]
    confidence = min(1.0, best.score)
    if classify_intent(query) == "question" and confidence > 0:
        confidence = min(1.0, confidence * 1.2)
    return best.text, round(confidence, 3)


_ENTITY_RE = re.compile(
    r"\b([A-Z][a-zA-Z0-9]+(?:\s+[A-Z][a-zA-Z0-9]+){0,3})\b")


def extract_entities(text: str, max_entities: int = 10) -> list[str]:
    """Capitalized-span entity extraction with frequency ranking."""
    from collections import Counter
    counts = Counter()
    for m in _ENTITY_RE.finditer(text):
        span = m.group(1)
        if len(span) > 2 and not span.isupper() or len(span.split()) > 1:
            counts[span] += 1
    # drop sentence-initial one-word commons that appear once
    out = [e for e, c in counts.most_common(max_entities * 2)
           if c > 1 or len(e.split()) > 1]
    return out[:max_entities]


def toxicity_filter(text: str) -> bool:
    """True when text is acceptable (very light heuristic filter)."""
    words = {w.lower() for w in _WORD_RE.findall(text)}
    return len(words & _TOXIC) < 2


def format_rag_output(query: str, results: list[dict],
                      chunk_size: int = 512,
                      answer_mode: bool = False,
                      summarizer=None) -> RAGOutput:
    out = RAGOutput(chunks=chunk_results(query, results, chunk_size))
    if answer_mode:
        out.answer, out.confidence = extract_answer(query, results)
    all_text = " ".join(c.text for c in out.chunks)
    out.entities = extract_entities(all_text)
    if summarizer is not None and results:
        out.summary = summarizer.summarize_results(results, query).summary
    return out
