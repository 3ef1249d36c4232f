"""Structured-data extraction: JSON-LD + OpenGraph + code blocks/tables.

Reference parity: infomesh/crawler/structured.py + content_extract.py.
"""
from __future__ import annotations

import json
import re
from dataclasses import dataclass, field

_JSONLD_RE = re.compile(
    r'<script[^>]+type=["\']application/ld\+json["\'][^>]*>(.*?)</script>',
    re.S | re.I)
_OG_RE = re.compile(
    r'<meta[^>]+property=["\']og:([a-z:_]+)["\'][^>]+content=["\']([^"\']*)',
    re.I)
_OG_RE2 = re.compile(
    r'<meta[^>]+content=["\']([^"\']*)["\'][^>]+property=["\']og:([a-z:_]+)',
    re.I)
_CODE_RE = re.compile(r"<pre[^>]*>(.*?)</pre>|<code[^>]*>(.*?)</code>",
                      re.S | re.I)
_TABLE_RE = re.compile(r"<table[^>]*>(.*?)</table>", re.S | re.I)
_TAG_RE = re.compile(r"<[^>]+>")


@dataclass
class StructuredData:
    json_ld: list[dict] = field(default_factory=list)
    open_graph: dict[str, str] = field(default_factory=dict)
    code_blocks: list[str] = field(default_factory=list)
    tables: list[list[list[str]]] = field(default_factory=list)


def extract_json_ld(html: str, max_items: int = 10) -> list[dict]:
    out = []
    for m in _JSONLD_RE.finditer(html):
        try:
            data = json.loads(m.group(1).strip())
        except json.JSONDecodeError:
            continue
        items = data if isinstance(data, list) else [data]
        out.extend(d for d in items if isinstance(d, dict))
        if len(out) >= max_items:
            break
    return out[:max_items]


def extract_open_graph(html: str) -> dict[str, str]:
    og = {}
    for prop, content in _OG_RE.findall(html):
        og.setdefault(prop.lower(), content)
    for content, prop in _OG_RE2.findall(html):
        og.setdefault(prop.lower(), content)
    return og


def extract_code_blocks(html: str, max_blocks: int = 20) -> list[str]:
    """String view over content_extract's typed extraction (one
    implementation; content_extract.py carries language detection,
    markdown fences and the reference's richer surface)."""
    from .content_extract import extract_code_blocks as _rich
    out = []
    for cb in _rich(html, max_blocks=max_blocks):
        if len(cb.code) >= 20:
            out.append(cb.code[:5000])
    return out[:max_blocks]


def extract_tables(html: str, max_tables: int = 5) -> list[list[list[str]]]:
    """Rows-of-cells view over content_extract's typed tables (headers
    included as the first row when present)."""
    from .content_extract import extract_tables as _rich
    out: list[list[list[str]]] = []
    for t in _rich(html, max_tables=max_tables):
        rows = ([list(t.headers)] if t.headers else []) \
            + [list(r) for r in t.rows]
        if rows:
            out.append(rows)
    return out[:max_tables]


def extract_structured(html: str) -> StructuredData:
    return StructuredData(
        json_ld=extract_json_ld(html),
        open_graph=extract_open_graph(html),
        code_blocks=extract_code_blocks(html),
        tables=extract_tables(html),
    )
