"""Data-quality grading: freshness/trust grades, citations, fact
cross-referencing (reference parity: infomesh/data_quality.py)."""
from __future__ import annotations

import time
from dataclasses import dataclass

from ..index.ranking import freshness_score


@dataclass
class QualityGrade:
    grade: str            # A..F
    freshness: float
    trust: float
    completeness: float


def grade_document(crawled_at: float, trust: float, text_len: int,
                   has_title: bool, now: float | None = None
                   ) -> QualityGrade:
    fresh = freshness_score(crawled_at, now)
    completeness = min(1.0, text_len / 2000.0) * (1.0 if has_title else 0.7)
    score = 0.4 * fresh + 0.35 * trust + 0.25 * completeness
    grade = ("A" if score >= 0.8 else "B" if score >= 0.6 else
             "C" if score >= 0.4 else "D" if score >= 0.2 else "F")
    return QualityGrade(grade, round(fresh, 3), round(trust, 3),
                        round(completeness, 3))


def format_citation(url: str, title: str, crawled_at: float) -> str:
    date = time.strftime("%Y-%m-%d", time.localtime(crawled_at)) \
        if crawled_at else "n.d."
    return f"{title or url}. Retrieved {date}. {url}"


def cross_reference(claim_support: dict[str, float],
                    min_sources: int = 2,
                    threshold: float = 0.5) -> dict:
    """Aggregate per-source support scores into a verdict
    (reference fact cross-reference)."""
    supporting = [u for u, s in claim_support.items() if s >= threshold]
    return {
        "verdict": ("corroborated" if len(supporting) >= min_sources else
                    "single-source" if len(supporting) == 1 else
                    "unsupported"),
        "supporting_sources": supporting,
        "n_sources": len(claim_support),
    }
