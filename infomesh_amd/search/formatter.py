"""Result renderers: text and JSON views of search responses.

Reference parity: infomesh/search/formatter.py (text/JSON renderers for
FTS/hybrid/distributed results).
"""
from __future__ import annotations

import json
from typing import Any


def _get(r: Any, key: str, default=""):
    if isinstance(r, dict):
        return r.get(key, default)
    return getattr(r, key, default)


def result_to_dict(r: Any) -> dict:
    return {
        "url": _get(r, "url"),
        "title": _get(r, "title"),
        "snippet": _get(r, "snippet"),
        "score": round(float(_get(r, "score", 0.0) or 0.0), 4),
        "sources": _get(r, "sources", None) or [_get(r, "source", "fts")],
        "domain": _get(r, "domain", ""),
        "language": _get(r, "language", ""),
    }


def format_json(response: Any) -> str:
    return json.dumps({
        "query": _get(response, "query"),
        "effective_query": _get(response, "effective_query"),
        "mode": _get(response, "mode"),
        "elapsed_ms": round(float(_get(response, "elapsed_ms", 0.0)), 2),
        "degraded": bool(_get(response, "degraded", False)),
        "results": [result_to_dict(r) for r in _get(response, "results", [])],
    }, ensure_ascii=False, indent=2)


def format_text(response: Any, max_snippet: int = 200) -> str:
    results = _get(response, "results", [])
    lines = [f"# {_get(response, 'query')}  "
             f"({_get(response, 'mode')}, "
             f"{float(_get(response, 'elapsed_ms', 0.0)):.1f} ms, "
             f"{len(results)} results)"]
    if _get(response, "degraded", False):
        lines.append("! degraded mode: local results only")
    for i, r in enumerate(results, 1):
        d = result_to_dict(r)
        snippet = d["snippet"].replace("<b>", "").replace("</b>", "")
        lines.append(f"\n{i}. {d['title'] or d['url']}")
        lines.append(f"   {d['url']}")
        if snippet:
            lines.append(f"   {snippet[:max_snippet]}")
        lines.append(f"   score={d['score']} sources={','.join(d['sources'])}")
    return "\n".join(lines)
