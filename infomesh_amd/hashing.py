"""Content hashing — the identity of every doc/URL/keyword.

Reference parity: infomesh/hashing.py:13-40 (SHA-256 hex digests).
"""
from __future__ import annotations

import hashlib


def content_hash(data: bytes | str) -> str:
    """SHA-256 hex digest of content (text is UTF-8 encoded first)."""
    if isinstance(data, str):
        data = data.encode("utf-8", errors="replace")
    return hashlib.sha256(data).hexdigest()


def short_hash(data: bytes | str, length: int = 12) -> str:
    """Truncated content hash for display/IDs."""
    return content_hash(data)[:length]


def shard_of(key: str, n_shards: int) -> int:
    """Deterministic hash-partition of a document key across GPU shards.

    Replaces the reference's Kademlia-XOR URL→peer assignment
    (crawler/url_assigner.py:26-151) with an intra-node shard hash."""
    if n_shards <= 1:
        return 0
    h = hashlib.sha256(key.encode("utf-8", errors="replace")).digest()
    return int.from_bytes(h[:8], "big") % n_shards


def hash64(data: bytes | str) -> int:
    """Stable 64-bit hash (first 8 bytes of SHA-256), for token/shingle ids."""
    if isinstance(data, str):
        data = data.encode("utf-8", errors="replace")
    return int.from_bytes(hashlib.sha256(data).digest()[:8], "big")
