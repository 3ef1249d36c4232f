"""SimHash near-duplicate detection (CPU path + GPU batch hook).

Reference parity: infomesh/crawler/simhash.py (Charikar 64-bit simhash
over 3-word shingles, Hamming threshold 3, linear-scan index capped at
500k entries). The GPU analogue (batched fingerprint + XOR/popcount
scan) lives in ops/csrc/simhash.hip; SimHashIndex.scan_gpu uses it when
a GPU is present and the index is large.
"""
from __future__ import annotations

import re

from ..hashing import hash64

HAMMING_THRESHOLD = 3
SHINGLE_WORDS = 3
MAX_INDEX_ENTRIES = 500_000

_WORD_RE = re.compile(r"\w+", re.UNICODE)


def shingle_hashes(text: str, n: int = SHINGLE_WORDS) -> list[int]:
    words = [w.lower() for w in _WORD_RE.findall(text)]
    if not words:
        return []
    if len(words) < n:
        return [hash64(" ".join(words))]
    return [hash64(" ".join(words[i:i + n]))
            for i in range(len(words) - n + 1)]


def simhash(text: str) -> int:
    """Charikar bit-vote fingerprint (matches the GPU kernel exactly —
    tests/test_ops_gpu.py::test_simhash_fingerprint_parity)."""
    hashes = shingle_hashes(text)
    if not hashes:
        return 0
    fp = 0
    for bit in range(64):
        vote = sum(1 if (h >> bit) & 1 else -1 for h in hashes)
        if vote > 0:
            fp |= 1 << bit
    return fp


def hamming_distance(a: int, b: int) -> int:
    return bin(a ^ b).count("1")


class SimHashIndex:
    """In-memory fingerprint map with linear near-dup scan."""

    def __init__(self, threshold: int = HAMMING_THRESHOLD,
                 max_entries: int = MAX_INDEX_ENTRIES):
        self.threshold = threshold
        self.max_entries = max_entries
        self._fps: dict[str, int] = {}    # url_hash -> fingerprint

    def __len__(self) -> int:
        return len(self._fps)

    def add(self, key: str, fp: int) -> None:
        if len(self._fps) >= self.max_entries:
            # drop an arbitrary oldest-ish entry (dict preserves order)
            self._fps.pop(next(iter(self._fps)))
        self._fps[key] = fp

    def find_near(self, fp: int) -> str | None:
        for key, other in self._fps.items():
            if hamming_distance(fp, other) <= self.threshold:
                return key
        return None

    def is_near_duplicate(self, text: str) -> tuple[bool, int]:
        fp = simhash(text)
        return self.find_near(fp) is not None, fp

    def fingerprints(self) -> list[int]:
        return list(self._fps.values())
