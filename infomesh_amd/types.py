"""Typed protocols decoupling optional components.

Reference parity: infomesh/types.py:17-81 (KeyPairLike,
VectorStoreLike, AuthorityFn protocols).
"""
from __future__ import annotations

from typing import Callable, Protocol, runtime_checkable

AuthorityFn = Callable[[str], float]   # url -> [0,1]
TrustFn = Callable[[str], float]       # domain -> [0,1]


@runtime_checkable
class KeyPairLike(Protocol):
    public: bytes

    @property
    def node_id(self) -> str: ...

    def sign(self, message: bytes) -> bytes: ...


@runtime_checkable
class DenseSearcherLike(Protocol):
    """Any dense retriever (GPU engine adapter or a fake)."""

    def search(self, query: str, limit: int = 10) -> list: ...


@runtime_checkable
class ShardLike(Protocol):
    """A scoreable index shard (GpuShard / CpuShard)."""
    n_docs: int

    def search(self, queries_terms, query_emb, k: int = 100,
               scores_buf=None, phase_t=None): ...


@runtime_checkable
class SummarizerLike(Protocol):
    def summarize(self, text: str, title: str = "",
                  sentences: int = 3): ...
