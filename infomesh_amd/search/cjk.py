"""CJK query handling.

Reference parity: infomesh/search/cjk.py (CJK detection, bigram/trigram
expansion for FTS matching with the trigram/unicode61 tokenizers,
tokenizer recommendation, query preprocessing).
"""
from __future__ import annotations

import re

_CJK_RANGES = (
    (0x4E00, 0x9FFF),    # CJK Unified Ideographs
    (0x3400, 0x4DBF),    # CJK Extension A
    (0x3040, 0x309F),    # Hiragana
    (0x30A0, 0x30FF),    # Katakana
    (0xAC00, 0xD7AF),    # Hangul syllables
    (0xF900, 0xFAFF),    # CJK Compatibility Ideographs
)


def is_cjk_char(ch: str) -> bool:
    cp = ord(ch)
    return any(lo <= cp <= hi for lo, hi in _CJK_RANGES)


def cjk_ratio(text: str) -> float:
    if not text:
        return 0.0
    letters = [c for c in text if not c.isspace()]
    if not letters:
        return 0.0
    return sum(1 for c in letters if is_cjk_char(c)) / len(letters)


def contains_cjk(text: str) -> bool:
    return any(is_cjk_char(c) for c in text)


def ngram_expand(run: str, n: int = 2) -> list[str]:
    """Overlapping n-grams of a contiguous CJK run (bigrams by default)."""
    if not run:
        return []
    if len(run) <= n:
        return [run]
    return [run[i:i + n] for i in range(len(run) - n + 1)]


_CJK_RUN_RE = re.compile(
    "[" + "".join(f"{chr(lo)}-{chr(hi)}" for lo, hi in _CJK_RANGES) + "]+")


def tokenize_query_cjk(query: str, n: int = 2) -> str:
    """Split CJK runs into n-grams so unicode61-tokenized FTS can match
    them; non-CJK spans pass through unchanged
    (reference: cjk.py:76-185)."""
    if not contains_cjk(query):
        return query
    out: list[str] = []
    pos = 0
    for m in _CJK_RUN_RE.finditer(query):
        before = query[pos:m.start()].strip()
        if before:
            out.append(before)
        out.extend(ngram_expand(m.group(0), n))
        pos = m.end()
    tail = query[pos:].strip()
    if tail:
        out.append(tail)
    return " ".join(out)


def recommend_tokenizer(sample_texts: list[str]) -> str:
    """Recommend an FTS tokenizer from corpus content
    (reference: cjk.py:167)."""
    if not sample_texts:
        return "unicode61"
    avg = sum(cjk_ratio(t) for t in sample_texts) / len(sample_texts)
    return "trigram" if avg > 0.3 else "unicode61"
