"""Minimal PDF text extraction (no pdf wheel in this image).

Reference parity: infomesh/crawler/pdf.py (PDF text for indexing).
Handles the common case: FlateDecode (zlib) content streams with
Tj/TJ/' text-showing operators and literal `(…)`` strings. Scanned or
exotic-encoding PDFs yield little/no text — callers treat empty output
as "no substantial text" (the crawl worker's thin-page path).
"""
from __future__ import annotations

import re
import zlib

_STREAM_RE = re.compile(rb"stream\r?\n(.*?)endstream", re.S)
_TEXT_OP_RE = re.compile(
    rb"\((?P<lit>(?:\\.|[^\\()])*)\)\s*(?:Tj|')"
    rb"|\[(?P<arr>(?:\\.|[^\]])*)\]\s*TJ", re.S)
_ARR_LIT_RE = re.compile(rb"\((?:\\.|[^\\()])*\)", re.S)

_ESCAPES = {b"\\n": b"\n", b"\\r": b"\r", b"\\t": b"\t",
            b"\\(": b"(", b"\\)": b")", b"\\\\": b"\\"}


def _unescape(lit: bytes) -> bytes:
    for k, v in _ESCAPES.items():
        lit = lit.replace(k, v)
    return re.sub(rb"\\(\d{1,3})",
                  lambda m: bytes([int(m.group(1), 8) & 0xFF]), lit)


def extract_pdf_text(data: bytes, max_chars: int = 500_000) -> str:
    """Best-effort text from a PDF byte string."""
    if not data.startswith(b"%PDF"):
        return ""
    chunks: list[str] = []
    total = 0
    for m in _STREAM_RE.finditer(data):
        raw = m.group(1)
        for candidate in (raw,):
            try:
                content = zlib.decompress(candidate)
            except zlib.error:
                content = candidate  # maybe uncompressed
            found = False
            for tm in _TEXT_OP_RE.finditer(content):
                if tm.group("lit") is not None:
                    text = _unescape(tm.group("lit"))
                else:
                    text = b"".join(
                        _unescape(x[1:-1])
                        for x in _ARR_LIT_RE.findall(tm.group("arr")))
                decoded = text.decode("latin-1", errors="replace").strip()
                if decoded:
                    chunks.append(decoded)
                    total += len(decoded)
                    found = True
                if total >= max_chars:
                    break
            if found or total >= max_chars:
                break
        if total >= max_chars:
            break
    out = " ".join(chunks)
    return re.sub(r"\s{2,}", " ", out)[:max_chars]


def looks_like_pdf(data: bytes, content_type: str = "") -> bool:
    return data[:5] == b"%PDF-" or "application/pdf" in content_type
