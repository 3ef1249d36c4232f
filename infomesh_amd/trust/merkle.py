"""SHA-256 Merkle tree over document hashes, with membership proofs.

Reference parity: infomesh/trust/merkle.py (build/prove/verify with L/R
path, serde). Used by credit verification (sampled-entry proofs) and
Merkle-proof audits.
"""
from __future__ import annotations

import hashlib
from dataclasses import dataclass, field


def _h(data: bytes) -> bytes:
    return hashlib.sha256(data).digest()


def _leaf(data: bytes) -> bytes:
    return _h(b"\x00" + data)


def _node(left: bytes, right: bytes) -> bytes:
    return _h(b"\x01" + left + right)


@dataclass
class MerkleProof:
    leaf_index: int
    leaf_hash: bytes
    path: list[tuple[str, bytes]]   # ("L"|"R", sibling hash) bottom-up

    def to_dict(self) -> dict:
        return {"leaf_index": self.leaf_index,
                "leaf_hash": self.leaf_hash.hex(),
                "path": [[d, h.hex()] for d, h in self.path]}

    @classmethod
    def from_dict(cls, d: dict) -> "MerkleProof":
        return cls(leaf_index=int(d["leaf_index"]),
                   leaf_hash=bytes.fromhex(d["leaf_hash"]),
                   path=[(p[0], bytes.fromhex(p[1])) for p in d["path"]])


@dataclass
class MerkleTree:
    leaves: list[bytes] = field(default_factory=list)
    _levels: list[list[bytes]] = field(default_factory=list)

    @classmethod
    def from_items(cls, items: list[bytes | str]) -> "MerkleTree":
        leaves = [_leaf(i.encode() if isinstance(i, str) else i)
                  for i in items]
        t = cls(leaves=leaves)
        t._build()
        return t

    def _build(self) -> None:
        if not self.leaves:
            self._levels = [[]]
            return
        levels = [list(self.leaves)]
        while len(levels[-1]) > 1:
            cur = levels[-1]
            nxt = []
            for i in range(0, len(cur), 2):
                left = cur[i]
                right = cur[i + 1] if i + 1 < len(cur) else cur[i]
                nxt.append(_node(left, right))
            levels.append(nxt)
        self._levels = levels

    @property
    def root(self) -> bytes:
        if not self._levels or not self._levels[-1]:
            return b"\x00" * 32
        return self._levels[-1][0]

    def prove(self, index: int) -> MerkleProof:
        if not (0 <= index < len(self.leaves)):
            raise IndexError(index)
        path: list[tuple[str, bytes]] = []
        i = index
        for level in self._levels[:-1]:
            sib = i ^ 1
            if sib >= len(level):
                sib = i  # duplicated odd node
            path.append(("L" if sib < i else "R", level[sib]))
            i //= 2
        return MerkleProof(index, self.leaves[index], path)

    @staticmethod
    def verify_proof(root: bytes, proof: MerkleProof,
                     item: bytes | str | None = None) -> bool:
        h = proof.leaf_hash
        if item is not None:
            data = item.encode() if isinstance(item, str) else item
            if _leaf(data) != h:
                return False
        for direction, sibling in proof.path:
            h = _node(sibling, h) if direction == "L" else _node(h, sibling)
        return h == root
