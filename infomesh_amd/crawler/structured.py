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
    out = []
    for m in _CODE_RE.finditer(html):
        code = _TAG_RE.sub("", m.group(1) or m.group(2) or "").strip()
        if len(code) >= 20:
            out.append(code[:5000])
        if len(out) >= max_blocks:
            break
    return out


def extract_tables(html: str, max_tables: int = 5) -> list[list[list[str]]]:
    tables = []
    for m in _TABLE_RE.finditer(html):
        rows = []
        for row_html in re.findall(r"<tr[^>]*>(.*?)</tr>", m.group(1),
                                   re.S | re.I):
            cells = [_TAG_RE.sub("", c).strip() for c in re.findall(
                r"<t[hd][^>]*>(.*?)</t[hd]>", row_html, re.S | re.I)]
            if cells:
                rows.append(cells)
        if rows:
            tables.append(rows)
        if len(tables) >= max_tables:
            break
    return tables


def extract_structured(html: str) -> StructuredData:
    return StructuredData(
        json_ld=extract_json_ld(html),
        open_graph=extract_open_graph(html),
        code_blocks=extract_code_blocks(html),
        tables=extract_tables(html),
    )
