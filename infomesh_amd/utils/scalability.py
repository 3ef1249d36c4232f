"""Scalability helpers: Bloom filter, batched ingest, connection pooling.

Reference parity: infomesh/scalability.py (connection pool / batch
ingest / Bloom filter / incremental rebuild helpers).
"""
from __future__ import annotations

import math
from typing import Callable, Iterable, Sequence

from ..hashing import hash64


class BloomFilter:
    """Classic double-hashing Bloom filter over strings."""

    def __init__(self, capacity: int = 1_000_000, error_rate: float = 0.01):
        self.capacity = capacity
        m = max(8, int(-capacity * math.log(error_rate) / (math.log(2) ** 2)))
        self.m = m
        self.k = max(1, round(m / capacity * math.log(2)))
        self._bits = bytearray((m + 7) // 8)
        self.count = 0

    def _positions(self, item: str):
        h = hash64(item)
        h1 = h & 0xFFFFFFFF
        h2 = (h >> 32) | 1
        for i in range(self.k):
            yield (h1 + i * h2) % self.m

    def add(self, item: str) -> None:
        for p in self._positions(item):
            self._bits[p >> 3] |= 1 << (p & 7)
        self.count += 1

    def __contains__(self, item: str) -> bool:
        return all(self._bits[p >> 3] & (1 << (p & 7))
                   for p in self._positions(item))

    def fill_ratio(self) -> float:
        ones = sum(bin(b).count("1") for b in self._bits)
        return ones / self.m


def batch_ingest(items: Iterable, process_batch: Callable[[Sequence], int],
                 batch_size: int = 1000) -> int:
    """Feed items to process_batch in fixed-size chunks; returns total."""
    total = 0
    batch: list = []
    for item in items:
        batch.append(item)
        if len(batch) >= batch_size:
            total += process_batch(batch)
            batch = []
    if batch:
        total += process_batch(batch)
    return total


class RoundRobinPool:
    """Tiny connection/resource pool with round-robin checkout."""

    def __init__(self, factory: Callable[[], object], size: int = 4):
        self._items = [factory() for _ in range(size)]
        self._i = 0

    def get(self):
        item = self._items[self._i % len(self._items)]
        self._i += 1
        return item

    def close_all(self) -> None:
        for it in self._items:
            close = getattr(it, "close", None)
            if close:
                try:
                    close()
                except Exception:
                    pass
