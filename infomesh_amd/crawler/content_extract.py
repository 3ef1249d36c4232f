"""Structured content extraction: code blocks + tables.

Reference parity: infomesh/crawler/content_extract.py:1-196 —
extract_code_blocks (HTML <pre><code> with language detection + fenced
markdown), extract_tables (HTML <table> → rows with to_csv /
to_dict_list). Pure-regex like the reference; no HTML parser dep.
"""
from __future__ import annotations

import html
import re
from dataclasses import dataclass

_PRE_CODE_RE = re.compile(
    r"<pre[^>]*>\s*<code([^>]*)>(.*?)</code>\s*</pre>",
    re.IGNORECASE | re.DOTALL)
_PRE_BARE_RE = re.compile(r"<pre([^>]*)>(.*?)</pre>",
                          re.IGNORECASE | re.DOTALL)
_FENCE_RE = re.compile(r"```([A-Za-z0-9_+-]*)\n(.*?)```", re.DOTALL)
_LANG_ATTR_RE = re.compile(
    r"(?:class|data-lang(?:uage)?)=[\"'][^\"']*?"
    r"(?:language-|lang-)?([A-Za-z0-9_+#-]+)[\"']")
_TABLE_RE = re.compile(r"<table[^>]*>(.*?)</table>", re.IGNORECASE | re.DOTALL)
_ROW_RE = re.compile(r"<tr[^>]*>(.*?)</tr>", re.IGNORECASE | re.DOTALL)
_CELL_RE = re.compile(r"<t[hd][^>]*>(.*?)</t[hd]>", re.IGNORECASE | re.DOTALL)
_TAG_RE = re.compile(r"<[^>]+>")

_KNOWN_LANGS = frozenset(
    "python c cpp c++ rust go java javascript js typescript ts bash sh "
    "shell sql html css json yaml toml ruby php kotlin swift scala r "
    "perl lua haskell zig hip cuda".split())


@dataclass(frozen=True)
class CodeBlock:
    code: str
    language: str = ""


@dataclass(frozen=True)
class ExtractedTable:
    headers: tuple[str, ...]
    rows: tuple[tuple[str, ...], ...]

    def to_csv(self) -> str:
        def esc(c: str) -> str:
            if any(ch in c for ch in ',"\n'):
                return '"' + c.replace('"', '""') + '"'
            return c
        lines = []
        if self.headers:
            lines.append(",".join(esc(h) for h in self.headers))
        for r in self.rows:
            lines.append(",".join(esc(c) for c in r))
        return "\n".join(lines)

    def to_dict_list(self) -> list[dict[str, str]]:
        if not self.headers:
            return [dict(enumerate(r)) for r in self.rows]  # type: ignore
        return [{h: (r[i] if i < len(r) else "")
                 for i, h in enumerate(self.headers)}
                for r in self.rows]


def _clean(fragment: str) -> str:
    return html.unescape(_TAG_RE.sub("", fragment)).strip()


def extract_code_blocks(text: str, max_blocks: int = 50) -> list[CodeBlock]:
    """<pre><code class="language-x"> blocks, bare <pre> blocks, plus
    markdown fences."""
    out: list[CodeBlock] = []
    spans: list[tuple[int, int]] = []
    for m in _PRE_CODE_RE.finditer(text):
        attrs, body = m.group(1), m.group(2)
        lang = ""
        lm = _LANG_ATTR_RE.search(attrs)
        if lm and lm.group(1).lower() in _KNOWN_LANGS:
            lang = lm.group(1).lower()
        code = html.unescape(_TAG_RE.sub("", body)).strip("\n")
        if code.strip():
            out.append(CodeBlock(code=code, language=lang))
            spans.append(m.span())
        if len(out) >= max_blocks:
            return out
    for m in _PRE_BARE_RE.finditer(text):
        if any(a <= m.start() < b for a, b in spans):
            continue   # already captured as <pre><code>
        code = html.unescape(_TAG_RE.sub("", m.group(2))).strip("\n")
        if code.strip():
            out.append(CodeBlock(code=code, language=""))
        if len(out) >= max_blocks:
            return out
    for m in _FENCE_RE.finditer(text):
        lang = m.group(1).lower()
        code = m.group(2).strip("\n")
        if code.strip():
            out.append(CodeBlock(
                code=code,
                language=lang if lang in _KNOWN_LANGS else lang))
        if len(out) >= max_blocks:
            break
    return out


def extract_tables(text: str, max_tables: int = 20,
                   max_rows: int = 500) -> list[ExtractedTable]:
    out: list[ExtractedTable] = []
    for tm in _TABLE_RE.finditer(text):
        rows: list[tuple[str, ...]] = []
        headers: tuple[str, ...] = ()
        for i, rm in enumerate(_ROW_RE.finditer(tm.group(1))):
            if i >= max_rows:
                break
            cells = tuple(_clean(c) for c in _CELL_RE.findall(rm.group(1)))
            if not cells:
                continue
            if not headers and not rows and "<th" in rm.group(1).lower():
                headers = cells
            else:
                rows.append(cells)
        if headers or rows:
            out.append(ExtractedTable(headers=headers, rows=tuple(rows)))
        if len(out) >= max_tables:
            break
    return out
