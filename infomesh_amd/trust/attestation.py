"""Content attestations: signed claims that a node crawled a URL and
observed specific content.

Reference parity: infomesh/trust/attestation.py (ContentAttestation
{url, raw_hash, text_hash, peer_id, ts, signature} create/verify/serde,
Merkle-root verification over attestation batches).
"""
from __future__ import annotations

import json
import time
from dataclasses import dataclass, asdict

from .keys import KeyPair
from .merkle import MerkleTree


@dataclass
class ContentAttestation:
    url: str
    raw_hash: str
    text_hash: str
    node_id: str
    ts: float
    public_key: str = ""
    signature: str = ""

    def payload(self) -> bytes:
        return json.dumps({
            "url": self.url, "raw_hash": self.raw_hash,
            "text_hash": self.text_hash, "node_id": self.node_id,
            "ts": self.ts}, sort_keys=True).encode()

    def to_dict(self) -> dict:
        return asdict(self)

    @classmethod
    def from_dict(cls, d: dict) -> "ContentAttestation":
        return cls(**{k: d[k] for k in
                      ("url", "raw_hash", "text_hash", "node_id", "ts",
                       "public_key", "signature")})


def create_attestation(kp: KeyPair, url: str, raw_hash: str,
                       text_hash: str,
                       sign: bool = True) -> ContentAttestation:
    """sign=False defers the ~3 ms pure-python Ed25519 signature —
    ingest stores the claim unsigned and sign_attestation() completes
    it on first serve (attestations are write-mostly: the reference
    publishes them to the DHT on demand, not per crawl)."""
    att = ContentAttestation(url=url, raw_hash=raw_hash,
                             text_hash=text_hash, node_id=kp.node_id,
                             ts=time.time(), public_key=kp.public.hex())
    if sign:
        att.signature = kp.sign(att.payload()).hex()
    return att


def sign_attestation(kp: KeyPair, att: ContentAttestation) -> ContentAttestation:
    """Complete a deferred signature (idempotent)."""
    if not att.signature:
        att.signature = kp.sign(att.payload()).hex()
    return att


def verify_attestation(att: ContentAttestation) -> bool:
    try:
        pub = bytes.fromhex(att.public_key)
        sig = bytes.fromhex(att.signature)
    except ValueError:
        return False
    from ..hashing import content_hash
    if content_hash(pub)[:32] != att.node_id:
        return False
    return KeyPair.verify(pub, att.payload(), sig)


def attestation_batch_root(atts: list[ContentAttestation]) -> bytes:
    """Merkle root over a batch (reference: attestation.py:218-244)."""
    return MerkleTree.from_items(
        [a.payload() for a in atts]).root
