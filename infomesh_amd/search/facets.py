"""Facets, clustering, near-dup removal over result lists.

Reference parity: infomesh/search/facets.py (domain/language/date facet
counts, domain clustering, Jaccard near-dup removal).
"""
from __future__ import annotations

import re
import time
from collections import defaultdict
from typing import Any, Sequence

_WORD_RE = re.compile(r"\w+")


def _get(r: Any, key: str, default=""):
    if isinstance(r, dict):
        return r.get(key, default)
    return getattr(r, key, default)


def compute_facets(results: Sequence[Any]) -> dict:
    domains: dict[str, int] = defaultdict(int)
    languages: dict[str, int] = defaultdict(int)
    dates: dict[str, int] = defaultdict(int)
    now = time.time()
    for r in results:
        d = _get(r, "domain") or ""
        if d:
            domains[d] += 1
        l = _get(r, "language") or ""
        if l:
            languages[l] += 1
        ts = _get(r, "crawled_at", 0.0) or 0.0
        if ts:
            age_d = (now - ts) / 86400
            bucket = ("today" if age_d < 1 else
                      "this_week" if age_d < 7 else
                      "this_month" if age_d < 30 else "older")
            dates[bucket] += 1
    return {"domains": dict(sorted(domains.items(), key=lambda p: -p[1])),
            "languages": dict(languages), "dates": dict(dates)}


def cluster_by_domain(results: Sequence[Any],
                      max_per_domain: int = 2) -> list[Any]:
    """Diversify: cap results per domain, preserving order."""
    seen: dict[str, int] = defaultdict(int)
    out, overflow = [], []
    for r in results:
        d = _get(r, "domain") or _get(r, "url")
        if seen[d] < max_per_domain:
            out.append(r)
            seen[d] += 1
        else:
            overflow.append(r)
    return out + overflow


def jaccard(a: str, b: str) -> float:
    wa = set(_WORD_RE.findall(a.lower()))
    wb = set(_WORD_RE.findall(b.lower()))
    if not wa or not wb:
        return 0.0
    return len(wa & wb) / len(wa | wb)


def remove_near_duplicates(results: Sequence[Any],
                           threshold: float = 0.85) -> list[Any]:
    """Drop results whose snippet+title is near-identical to a kept one."""
    kept: list[Any] = []
    for r in results:
        text = f"{_get(r, 'title')} {_get(r, 'snippet')}"
        if any(jaccard(text, f"{_get(k, 'title')} {_get(k, 'snippet')}")
               >= threshold for k in kept):
            continue
        kept.append(r)
    return kept
