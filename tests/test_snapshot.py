"""Snapshot format tests (reference parity: tests/test_snapshot.py)."""
from __future__ import annotations

import struct

import pytest

from infomesh_amd.errors import InfoMeshError
from infomesh_amd.index.local_store import LocalStore
from infomesh_amd.index.snapshot import (export_snapshot, import_snapshot,
                                         read_snapshot_header, FORMAT_VERSION)


def test_export_import_roundtrip(seeded_store, tmp_path):
    p = tmp_path / "x.infomesh-snapshot"
    header = export_snapshot(seeded_store, p, node_name="n1")
    assert header["doc_count"] == seeded_store.count()
    assert header["format_version"] == FORMAT_VERSION

    dst = LocalStore(":memory:")
    res = import_snapshot(dst, p)
    assert res["imported"] == seeded_store.count()
    assert res["skipped"] == 0
    assert dst.search("python tutorial")
    dst.close()


def test_import_dedups_by_text_hash(seeded_store, tmp_path):
    p = tmp_path / "x.infomesh-snapshot"
    export_snapshot(seeded_store, p)
    res = import_snapshot(seeded_store, p)  # import into same store
    assert res["imported"] == 0
    assert res["skipped"] == seeded_store.count()


def test_header_read(seeded_store, tmp_path):
    p = tmp_path / "x.infomesh-snapshot"
    export_snapshot(seeded_store, p, node_name="node-7")
    h = read_snapshot_header(p)
    assert h["node"] == "node-7"


def test_corrupt_file_rejected(tmp_path):
    p = tmp_path / "bad.infomesh-snapshot"
    p.write_bytes(struct.pack(">I", 100) + b"garbage" * 5)
    with pytest.raises(InfoMeshError):
        read_snapshot_header(p)


def test_oversized_header_rejected(tmp_path):
    p = tmp_path / "bad2.infomesh-snapshot"
    p.write_bytes(struct.pack(">I", 999_999_999) + b"x")
    with pytest.raises(InfoMeshError) as ei:
        read_snapshot_header(p)
    assert ei.value.code == "IDX002"


def test_on_document_hook(seeded_store, tmp_path):
    p = tmp_path / "x.infomesh-snapshot"
    export_snapshot(seeded_store, p)
    dst = LocalStore(":memory:")
    seen = []
    import_snapshot(dst, p, on_document=seen.append)
    assert len(seen) == seeded_store.count()
    assert all(d.doc_id is not None for d in seen)
    dst.close()


def test_snapshot_reference_field_parity(tmp_path):
    """Byte-level interop with the reference format: header carries
    `document_count` and docs carry `raw_html_hash` (reference
    snapshot.py:82-90, local_store.py:487-498)."""
    import json
    import struct

    import msgpack

    from infomesh_amd import compression
    from infomesh_amd.index.local_store import Document
    from infomesh_amd.index.snapshot import export_snapshot

    store = LocalStore(tmp_path / "s.db")
    store.add_document(Document(url="https://p/1", title="t",
                                text="body " * 20, language="en",
                                raw_hash="rawhash123"))
    path = tmp_path / "p.infomesh-snapshot"
    export_snapshot(store, path, node_name="n")
    raw = path.read_bytes()
    hlen = struct.unpack(">I", raw[:4])[0]
    comp = compression.Compressor()
    header = json.loads(comp.decompress(raw[4:4 + hlen]))
    assert header["document_count"] == 1
    assert header["format_version"] == FORMAT_VERSION
    docs = msgpack.unpackb(comp.decompress(raw[4 + hlen:]), raw=False)
    assert docs[0]["raw_html_hash"] == "rawhash123"
    assert set(docs[0]) >= {"url", "title", "text", "language",
                            "raw_html_hash", "text_hash", "crawled_at"}
    store.close()
