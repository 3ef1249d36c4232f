"""`.infomesh-snapshot` index snapshot format (byte-layout compatible).

Reference parity: infomesh/index/snapshot.py:10-15,39-44,58-263 —
layout `[4B big-endian header len][zstd(JSON metadata)][zstd(msgpack doc
array)]`, format_version=1, import guards (≤1 GB file, ≤10 MB header,
≤100k docs), import dedups by text_hash.

This is the portable state container: GPU shards warm-start by importing
a snapshot into LocalStore and re-uploading.
"""
from __future__ import annotations

import json
import struct
import time
from pathlib import Path
from typing import Any, Callable

import msgpack

from .. import compression
from ..hashing import content_hash
from ..errors import InfoMeshError
from .local_store import Document, LocalStore

FORMAT_VERSION = 1
MAX_SNAPSHOT_BYTES = 1 * 1024 * 1024 * 1024   # 1 GB
MAX_HEADER_BYTES = 10 * 1024 * 1024           # 10 MB
MAX_DOCS = 100_000
SUFFIX = ".infomesh-snapshot"


def export_snapshot(store: LocalStore, path: str | Path,
                    max_docs: int = MAX_DOCS,
                    node_name: str = "",
                    level: int = compression.LEVEL_SNAPSHOT) -> dict[str, Any]:
    """Export the store to a snapshot file; returns the metadata header."""
    path = Path(path)
    docs: list[dict[str, Any]] = []
    for doc in store.export_documents():
        docs.append({
            "url": doc.url, "title": doc.title, "text": doc.text,
            "language": doc.language,
            "raw_html_hash": doc.raw_hash,  # reference field name
            "text_hash": doc.text_hash,
            "crawled_at": doc.crawled_at,
        })
        if len(docs) >= max_docs:
            break
    header = {
        "format_version": FORMAT_VERSION,
        "created_at": time.time(),
        # the reference reads "document_count" (snapshot.py:85); keep
        # "doc_count" too for our own earlier snapshots
        "document_count": len(docs),
        "doc_count": len(docs),
        "node": node_name,
        "generator": "infomesh-amd",
    }
    comp = compression.Compressor(level)   # config index.snapshot_compression_level
    header_z = comp.compress(json.dumps(header).encode("utf-8"))
    docs_z = comp.compress(msgpack.packb(docs, use_bin_type=True))
    path.parent.mkdir(parents=True, exist_ok=True)
    tmp = path.with_suffix(path.suffix + ".tmp")
    with open(tmp, "wb") as f:
        f.write(struct.pack(">I", len(header_z)))
        f.write(header_z)
        f.write(docs_z)
    tmp.replace(path)
    return header


def read_snapshot_header(path: str | Path) -> dict[str, Any]:
    path = Path(path)
    size = path.stat().st_size
    if size > MAX_SNAPSHOT_BYTES:
        raise InfoMeshError("IDX002", f"file is {size} bytes")
    with open(path, "rb") as f:
        raw = f.read(4)
        if len(raw) != 4:
            raise InfoMeshError("IDX002", "truncated header length")
        (hlen,) = struct.unpack(">I", raw)
        if hlen > MAX_HEADER_BYTES:
            raise InfoMeshError("IDX002", f"header is {hlen} bytes")
        header_z = f.read(hlen)
    if len(header_z) != hlen:
        raise InfoMeshError("IDX002", "truncated header")
    try:
        header = json.loads(compression.decompress(header_z))
    except (ValueError, json.JSONDecodeError) as e:
        raise InfoMeshError("IDX002", f"bad header: {e}") from e
    if header.get("format_version") != FORMAT_VERSION:
        raise InfoMeshError(
            "IDX002", f"format_version {header.get('format_version')!r}")
    return header


def import_snapshot(store: LocalStore, path: str | Path,
                    progress: Callable[[int, int], None] | None = None,
                    on_document: Callable[[Document], None] | None = None,
                    ) -> dict[str, Any]:
    """Import a snapshot into the store; skips text_hash duplicates.

    `on_document` fires for each NEW doc (GPU-shard ingest hook).
    Returns {"imported": n, "skipped": m, "header": …}."""
    path = Path(path)
    header = read_snapshot_header(path)
    with open(path, "rb") as f:
        (hlen,) = struct.unpack(">I", f.read(4))
        f.seek(4 + hlen)
        docs_z = f.read(MAX_SNAPSHOT_BYTES)
    try:
        docs = msgpack.unpackb(
            compression.decompress(docs_z), raw=False,
            max_array_len=MAX_DOCS + 1, max_map_len=64)
    except (ValueError, msgpack.UnpackException) as e:
        raise InfoMeshError("IDX002", f"bad doc payload: {e}") from e
    if not isinstance(docs, list) or len(docs) > MAX_DOCS:
        raise InfoMeshError("IDX002", "doc array invalid or oversized")
    imported = skipped = 0
    total = len(docs)
    for i, d in enumerate(docs):
        if not isinstance(d, dict) or "url" not in d:
            skipped += 1
            continue
        doc = Document(
            url=str(d.get("url", "")),
            title=str(d.get("title", "")),
            text=str(d.get("text", "")),
            language=str(d.get("language", "")),
            text_hash=str(d.get("text_hash", "")) or content_hash(str(d.get("text", ""))),
            raw_hash=str(d.get("raw_html_hash", d.get("raw_hash", ""))),
            crawled_at=float(d.get("crawled_at", 0.0) or 0.0),
        )
        rowid = store.add_document(doc)
        if rowid is None:
            skipped += 1
        else:
            imported += 1
            doc.doc_id = rowid
            if on_document is not None:
                on_document(doc)
        if progress is not None and (i + 1) % 1000 == 0:
            progress(i + 1, total)
    if progress is not None:
        progress(total, total)
    return {"imported": imported, "skipped": skipped, "header": header}
