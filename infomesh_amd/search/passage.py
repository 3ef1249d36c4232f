"""Passage selection and snippet enhancement.

Reference parity: infomesh/search/passage.py (passage splitting, TF
coverage+density scoring, select_best_passage, <b> highlighting, intent
classification). The GPU batch passage scorer is optional
(SURVEY.md §2.9); this CPU path is authoritative.
"""
from __future__ import annotations

import functools as _functools
import re
from dataclasses import dataclass

_SENT_SPLIT_RE = re.compile(r"(?<=[.!?。！？])\s+")
_WORD_RE = re.compile(r"\w+", re.UNICODE)

PASSAGE_TARGET_WORDS = 60
PASSAGE_MAX_WORDS = 100


@dataclass
class Passage:
    text: str
    start: int          # char offset in source
    score: float = 0.0


def split_passages(text: str, target_words: int = PASSAGE_TARGET_WORDS,
                   max_words: int = PASSAGE_MAX_WORDS) -> list[Passage]:
    """Split text into sentence-aligned passages of roughly target_words."""
    if not text:
        return []
    passages: list[Passage] = []
    cur: list[str] = []
    cur_words = 0
    cur_start = 0
    offset = 0
    for sent in _SENT_SPLIT_RE.split(text):
        nwords = len(_WORD_RE.findall(sent))
        if cur and cur_words + nwords > max_words:
            passages.append(Passage(" ".join(cur), cur_start))
            cur, cur_words = [], 0
            cur_start = offset
        cur.append(sent)
        cur_words += nwords
        offset += len(sent) + 1
        if cur_words >= target_words:
            passages.append(Passage(" ".join(cur), cur_start))
            cur, cur_words = [], 0
            cur_start = offset
    if cur:
        passages.append(Passage(" ".join(cur), cur_start))
    return passages


def score_passage(query_terms: list[str], passage: str) -> float:
    """TF coverage + density scoring (reference: passage.py:143-180).

    coverage = fraction of distinct query terms present;
    density   = query-term hits per passage word (saturated)."""
    if not query_terms:
        return 0.0
    words = [w.lower() for w in _WORD_RE.findall(passage)]
    if not words:
        return 0.0
    wordset = set(words)
    terms = [t.lower() for t in query_terms]
    covered = sum(1 for t in set(terms) if t in wordset)
    coverage = covered / len(set(terms))
    hits = sum(1 for w in words if w in set(terms))
    density = hits / len(words)
    return 0.7 * coverage + 0.3 * min(1.0, 5.0 * density)


def select_best_passage(query: str, text: str,
                        max_chars: int = 300) -> str:
    """Best passage for a query, truncated to max_chars
    (reference: passage.py:183-227)."""
    terms = _WORD_RE.findall(query)
    passages = split_passages(text)
    if not passages:
        return text[:max_chars]
    for p in passages:
        p.score = score_passage(terms, p.text)
    best = max(passages, key=lambda p: p.score)
    out = best.text
    if len(out) > max_chars:
        out = out[:max_chars].rsplit(" ", 1)[0] + "…"
    return out


@_functools.lru_cache(maxsize=512)
def _snippet_pattern(query: str) -> "re.Pattern | None":
    terms = sorted({t for t in _WORD_RE.findall(query.lower())
                    if len(t) > 1}, key=len, reverse=True)
    if not terms:
        return None
    return re.compile(
        r"\b(" + "|".join(re.escape(t) for t in terms) + r")\b", re.I)


def fast_snippet(query: str, text: str, width: int = 200) -> str:
    """Cheap snippet for the batched serving path: a window around the
    first query-term occurrence, terms highlighted. The FTS5 snippet()
    analogue (reference local_store.py:253-352 uses SQL snippet());
    full passage scoring (select_best_passage) stays on the unbatched
    local path where per-query CPU time is not the bottleneck. The
    compiled pattern is cached per query — a serving batch snippets
    ~10 docs per query."""
    pattern = _snippet_pattern(query)
    if pattern is None or not text:
        return highlight(text[:width], query)
    m = pattern.search(text)
    if m is None:
        return text[:width]
    lo = max(0, m.start() - width // 3)
    window = text[lo:lo + width]
    if lo > 0:
        window = "…" + window.lstrip()
    return pattern.sub(lambda mm: f"<b>{mm.group(0)}</b>", window)


def highlight(text: str, query: str, tag: str = "b") -> str:
    """Wrap query terms in <b>…</b> (reference: passage.py:233)."""
    terms = sorted({t for t in _WORD_RE.findall(query.lower()) if len(t) > 1},
                   key=len, reverse=True)
    if not terms:
        return text
    pattern = re.compile(
        r"\b(" + "|".join(re.escape(t) for t in terms) + r")\b", re.I)
    return pattern.sub(lambda m: f"<{tag}>{m.group(0)}</{tag}>", text)


# ------------------------------------------------------ intent classes

INTENTS = ("informational", "navigational", "transactional", "question")

_QUESTION_WORDS = frozenset(
    "what why how when where who which can does is are should".split())
_NAV_HINTS = frozenset("login homepage official site download github docs".split())
_TRANS_HINTS = frozenset("buy price order install download purchase cheap deal".split())


def classify_intent(query: str) -> str:
    """Lightweight query-intent classification
    (reference: passage.py:328-373)."""
    terms = [t.lower() for t in _WORD_RE.findall(query)]
    if not terms:
        return "informational"
    if terms[0] in _QUESTION_WORDS or query.rstrip().endswith("?"):
        return "question"
    if any(t in _TRANS_HINTS for t in terms):
        return "transactional"
    if any(t in _NAV_HINTS for t in terms) or ("." in query and " " not in query.strip()):
        return "navigational"
    return "informational"
