"""Credit-ledger spot-check proofs: a signed Merkle root over all
entries plus randomly sampled entries with membership proofs.

Reference parity: infomesh/credits/verification.py (CreditProofBuilder,
peer spot-checks via sampled Merkle proofs).
"""
from __future__ import annotations

import random

from ..trust.keys import KeyPair
from ..trust.merkle import MerkleProof, MerkleTree
from .ledger import CreditLedger


class CreditProofBuilder:
    def __init__(self, ledger: CreditLedger, kp: KeyPair):
        self.ledger = ledger
        self.kp = kp

    def _entry_blobs(self) -> list[str]:
        rows = self.ledger.execute(
            "SELECT entry_hash FROM credit_entries ORDER BY id").fetchall()
        return [r["entry_hash"] for r in rows]

    def build_proof(self, n_samples: int = 3,
                    rng: random.Random | None = None) -> dict:
        rng = rng or random.Random()
        blobs = self._entry_blobs()
        tree = MerkleTree.from_items(blobs)
        root = tree.root
        signed = self.kp.sign(root).hex()
        samples = []
        if blobs:
            for idx in rng.sample(range(len(blobs)),
                                  min(n_samples, len(blobs))):
                samples.append({
                    "index": idx,
                    "entry_hash": blobs[idx],
                    "proof": tree.prove(idx).to_dict(),
                })
        return {
            "root": root.hex(),
            "signature": signed,
            "public_key": self.kp.public.hex(),
            "n_entries": len(blobs),
            "balance": self.ledger.balance(),
            "samples": samples,
        }

    @staticmethod
    def verify_proof(proof: dict) -> bool:
        try:
            root = bytes.fromhex(proof["root"])
            pub = bytes.fromhex(proof["public_key"])
            sig = bytes.fromhex(proof["signature"])
        except (KeyError, ValueError):
            return False
        if not KeyPair.verify(pub, root, sig):
            return False
        for s in proof.get("samples", []):
            mp = MerkleProof.from_dict(s["proof"])
            if not MerkleTree.verify_proof(root, mp, s["entry_hash"]):
                return False
        return True
