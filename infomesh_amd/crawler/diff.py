"""Content diffing + WARC export.

Reference parity: infomesh/crawler/diff.py (content diff between crawl
versions + WARC archive export).
"""
from __future__ import annotations

import difflib
import time
import uuid
from dataclasses import dataclass
from pathlib import Path


@dataclass
class ContentDiff:
    added_lines: int
    removed_lines: int
    changed_ratio: float
    summary: list[str]


def diff_content(old: str, new: str, max_summary: int = 20) -> ContentDiff:
    old_lines = old.splitlines()
    new_lines = new.splitlines()
    sm = difflib.SequenceMatcher(a=old_lines, b=new_lines)
    added = removed = 0
    summary: list[str] = []
    for tag, i1, i2, j1, j2 in sm.get_opcodes():
        if tag in ("replace", "delete"):
            removed += i2 - i1
            if len(summary) < max_summary:
                summary.extend(f"- {l}" for l in old_lines[i1:i2][:3])
        if tag in ("replace", "insert"):
            added += j2 - j1
            if len(summary) < max_summary:
                summary.extend(f"+ {l}" for l in new_lines[j1:j2][:3])
    return ContentDiff(added_lines=added, removed_lines=removed,
                       changed_ratio=round(1.0 - sm.ratio(), 4),
                       summary=summary[:max_summary])


def significant_change(old: str, new: str, threshold: float = 0.1) -> bool:
    return diff_content(old, new).changed_ratio >= threshold


def warc_export(path: str | Path, records: list[dict]) -> int:
    """Append WARC `resource` records ({url, content, content_type}).
    Minimal WARC/1.0 writer for archival export."""
    path = Path(path)
    path.parent.mkdir(parents=True, exist_ok=True)
    n = 0
    with open(path, "ab") as f:
        for rec in records:
            content = rec.get("content", "").encode("utf-8",
                                                    errors="replace")
            headers = (
                "WARC/1.0\r\n"
                "WARC-Type: resource\r\n"
                f"WARC-Record-ID: <urn:uuid:{uuid.uuid4()}>\r\n"
                f"WARC-Target-URI: {rec.get('url', '')}\r\n"
                f"WARC-Date: "
                f"{time.strftime('%Y-%m-%dT%H:%M:%SZ', time.gmtime())}\r\n"
                f"Content-Type: {rec.get('content_type', 'text/plain')}\r\n"
                f"Content-Length: {len(content)}\r\n\r\n")
            f.write(headers.encode())
            f.write(content)
            f.write(b"\r\n\r\n")
            n += 1
    return n
