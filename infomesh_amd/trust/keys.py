"""Node identity keys (Ed25519) with on-disk persistence and rotation.

Reference parity: infomesh/p2p/keys.py (KeyPair at ~/.infomesh/keys/,
peer_id from pubkey hash, ensure_keys, dual-signed rotation records).
"""
from __future__ import annotations

import json
import time
from dataclasses import dataclass
from pathlib import Path

from ..hashing import content_hash
from . import ed25519


@dataclass
class KeyPair:
    seed: bytes
    public: bytes

    @property
    def node_id(self) -> str:
        """Stable node identity = short hash of the public key."""
        return content_hash(self.public)[:32]

    def sign(self, message: bytes) -> bytes:
        return ed25519.sign(self.seed, message)

    @staticmethod
    def verify(public: bytes, message: bytes, signature: bytes) -> bool:
        return ed25519.verify(public, message, signature)

    @classmethod
    def generate(cls) -> "KeyPair":
        seed = ed25519.generate_seed()
        return cls(seed=seed, public=ed25519.public_key(seed))


def keys_dir(data_dir: Path) -> Path:
    return data_dir / "keys"


def ensure_keys(data_dir: Path) -> KeyPair:
    """Load or create the node key (reference: keys.py:193)."""
    kd = keys_dir(data_dir)
    seed_file = kd / "node.seed"
    if seed_file.exists():
        seed = bytes.fromhex(seed_file.read_text().strip())
        return KeyPair(seed=seed, public=ed25519.public_key(seed))
    kp = KeyPair.generate()
    kd.mkdir(parents=True, exist_ok=True)
    tmp = seed_file.with_suffix(".tmp")
    tmp.write_text(kp.seed.hex())
    tmp.chmod(0o600)
    tmp.replace(seed_file)
    (kd / "node.pub").write_text(kp.public.hex())
    return kp


def rotate_keys(data_dir: Path, old: KeyPair) -> tuple[KeyPair, dict]:
    """Dual-signed rotation record: old key endorses the new one
    (reference: keys.py:229-346)."""
    new = KeyPair.generate()
    payload = {
        "type": "key_rotation",
        "old_pub": old.public.hex(),
        "new_pub": new.public.hex(),
        "ts": time.time(),
    }
    blob = json.dumps(payload, sort_keys=True).encode()
    record = {
        **payload,
        "old_sig": old.sign(blob).hex(),
        "new_sig": new.sign(blob).hex(),
    }
    kd = keys_dir(data_dir)
    kd.mkdir(parents=True, exist_ok=True)
    (kd / "node.seed").write_text(new.seed.hex())
    (kd / "node.pub").write_text(new.public.hex())
    with open(kd / "rotations.jsonl", "a") as f:
        f.write(json.dumps(record) + "\n")
    return new, record


def verify_rotation(record: dict) -> bool:
    payload = {k: record[k] for k in ("type", "old_pub", "new_pub", "ts")}
    blob = json.dumps(payload, sort_keys=True).encode()
    try:
        old_pub = bytes.fromhex(record["old_pub"])
        new_pub = bytes.fromhex(record["new_pub"])
        return (ed25519.verify(old_pub, blob, bytes.fromhex(record["old_sig"]))
                and ed25519.verify(new_pub, blob,
                                   bytes.fromhex(record["new_sig"])))
    except (KeyError, ValueError):
        return False
