"""Summary verification: key-fact extraction + contradiction/support
scoring against the source text.

Reference parity: infomesh/summarizer/verify.py (key-fact extraction,
contradiction detection, self/cross-validation scores). CPU-side.
"""
from __future__ import annotations

import re
from dataclasses import dataclass, field

_SENT_RE = re.compile(r"(?<=[.!?])\s+")
_WORD_RE = re.compile(r"\w+")
_NUM_RE = re.compile(r"\b\d[\d,.]*\b")
_NEGATIONS = frozenset("not never no cannot n't without none".split())


@dataclass
class FactCheck:
    fact: str
    supported: bool
    evidence: str = ""
    score: float = 0.0


@dataclass
class VerificationReport:
    support_score: float            # 0..1 fraction of facts supported
    facts: list[FactCheck] = field(default_factory=list)
    contradictions: list[str] = field(default_factory=list)
    numbers_ok: bool = True


def extract_key_facts(summary: str, max_facts: int = 10) -> list[str]:
    sents = [s.strip() for s in _SENT_RE.split(summary) if s.strip()]
    return [s for s in sents if len(s.split()) >= 4][:max_facts]


def _content_words(text: str) -> set[str]:
    stop = frozenset("the a an is are was were be been of in on at to for "
                     "and or it this that with by from as".split())
    return {w.lower() for w in _WORD_RE.findall(text)
            if len(w) > 2 and w.lower() not in stop}


def _fact_support(fact: str, source: str) -> tuple[float, str]:
    """Best sentence-overlap support score + the evidence sentence."""
    fw = _content_words(fact)
    if not fw:
        return 0.0, ""
    best, best_sent = 0.0, ""
    for sent in _SENT_RE.split(source):
        sw = _content_words(sent)
        if not sw:
            continue
        overlap = len(fw & sw) / len(fw)
        if overlap > best:
            best, best_sent = overlap, sent.strip()
    return best, best_sent


def _negation_mismatch(fact: str, evidence: str) -> bool:
    f_neg = bool(_NEGATIONS & {w.lower() for w in _WORD_RE.findall(fact)})
    e_neg = bool(_NEGATIONS & {w.lower() for w in _WORD_RE.findall(evidence)})
    return f_neg != e_neg


def verify_summary(summary: str, source: str,
                   support_threshold: float = 0.5) -> VerificationReport:
    facts = extract_key_facts(summary)
    checks: list[FactCheck] = []
    contradictions: list[str] = []
    for fact in facts:
        score, evidence = _fact_support(fact, source)
        supported = score >= support_threshold
        if supported and _negation_mismatch(fact, evidence):
            supported = False
            contradictions.append(fact)
        checks.append(FactCheck(fact=fact, supported=supported,
                                evidence=evidence, score=round(score, 3)))
    # numeric claims must appear in the source
    src_nums = set(_NUM_RE.findall(source))
    sum_nums = set(_NUM_RE.findall(summary))
    numbers_ok = sum_nums <= src_nums or not sum_nums
    n = len(checks)
    support = (sum(1 for c in checks if c.supported) / n) if n else 1.0
    return VerificationReport(support_score=round(support, 3), facts=checks,
                              contradictions=contradictions,
                              numbers_ok=numbers_ok)


def cross_validate_summaries(summaries: list[str]) -> list[float]:
    """Pairwise agreement score per summary (peer-summary validation,
    reference verify.py cross-validation)."""
    if len(summaries) <= 1:
        return [1.0] * len(summaries)
    word_sets = [_content_words(s) for s in summaries]
    out = []
    for i, ws in enumerate(word_sets):
        sims = []
        for j, other in enumerate(word_sets):
            if i == j or not ws or not other:
                continue
            sims.append(len(ws & other) / len(ws | other))
        out.append(round(sum(sims) / len(sims), 3) if sims else 0.0)
    return out
