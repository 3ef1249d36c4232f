"""Score-breakdown explanations for the `explain` search mode.

Reference parity: infomesh/search/explain.py (per-result component
breakdown rendered for humans).
"""
from __future__ import annotations

from typing import Any

from ..index.local_store import LocalStore
from ..index.ranking import (W_AUTHORITY, W_BM25,
                             W_FRESHNESS, W_TITLE, W_TRUST, W_URL_PATH,
                             rank_local_results)


def explain_search(store: LocalStore, query: str, limit: int = 5,
                   authority_fn=None, trust_fn=None,
                   boost_fn=None) -> list[dict[str, Any]]:
    hits = store.search(query, limit=limit * 2)
    explained = rank_local_results(query, hits, authority_fn=authority_fn,
                                   trust_fn=trust_fn, explain=True,
                                   boost_fn=boost_fn)
    out = []
    for hit, bd in explained[:limit]:
        out.append({
            "url": hit.url,
            "title": hit.title,
            "total": round(hit.score, 4),
            "feedback_boost": round(boost_fn(hit.url), 4) if boost_fn
            else 0.0,
            "components": {
                "bm25": {"value": round(bd.bm25, 4), "weight": W_BM25},
                "freshness": {"value": round(bd.freshness, 4),
                              "weight": W_FRESHNESS},
                "trust": {"value": round(bd.trust, 4), "weight": W_TRUST},
                "authority": {"value": round(bd.authority, 4),
                              "weight": W_AUTHORITY},
                "title_match": {"value": round(bd.title_match, 4),
                                "weight": W_TITLE},
                "url_path": {"value": round(bd.url_path, 4),
                             "weight": W_URL_PATH},
            },
        })
    return out


def render_explanation(explained: list[dict]) -> str:
    lines = []
    for i, e in enumerate(explained, 1):
        lines.append(f"{i}. {e['title'] or e['url']}  (score {e['total']})")
        lines.append(f"   {e['url']}")
        for name, c in e["components"].items():
            contrib = c["value"] * c["weight"]
            lines.append(f"   {name:12s} {c['value']:.3f} × {c['weight']:.2f}"
                         f" = {contrib:.4f}")
    return "\n".join(lines)
