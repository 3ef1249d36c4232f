"""LangChain / LlamaIndex / Haystack adapters.

Reference parity: infomesh/integrations/ (InfoMeshRetriever,
InfoMeshReader, InfoMeshDocumentStore). These duck-type the framework
interfaces — `get_relevant_documents`, `load_data`, `write_documents`/
`filter_documents` — so they plug in when the frameworks are installed
and remain usable standalone (none of them ship in this image).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

from ..sdk.client import InfoMeshClient


@dataclass
class SimpleDocument:
    page_content: str
    metadata: dict[str, Any] = field(default_factory=dict)


class InfoMeshRetriever:
    """LangChain-style retriever over a running node (or an in-process
    AppContext when `ctx` is given)."""

    def __init__(self, base_url: str = "http://127.0.0.1:8080",
                 api_key: str = "", k: int = 5, mode: str = "auto",
                 ctx=None):
        self.k = k
        self.mode = mode
        self.ctx = ctx
        self.client = None if ctx is not None else \
            InfoMeshClient(base_url, api_key)

    def _search(self, query: str) -> list[dict]:
        if self.ctx is not None:
            resp = self.ctx.search(query, limit=self.k, mode=self.mode)
            from ..search.formatter import result_to_dict
            return [result_to_dict(r) for r in resp.results]
        return self.client.search(query, limit=self.k, mode=self.mode)

    def get_relevant_documents(self, query: str) -> list[SimpleDocument]:
        return [SimpleDocument(
            page_content=r.get("snippet") or r.get("title", ""),
            metadata={"url": r.get("url"), "title": r.get("title"),
                      "score": r.get("score")})
            for r in self._search(query)]

    # LangChain 0.2+ invoke-style alias
    def invoke(self, query: str, **_) -> list[SimpleDocument]:
        return self.get_relevant_documents(query)


class InfoMeshReader:
    """LlamaIndex-style reader: load_data(query) -> documents."""

    def __init__(self, **kw):
        self.retriever = InfoMeshRetriever(**kw)

    def load_data(self, query: str, limit: int = 5) -> list[SimpleDocument]:
        self.retriever.k = limit
        return self.retriever.get_relevant_documents(query)


class InfoMeshDocumentStore:
    """Haystack-style document store over an in-process AppContext."""

    def __init__(self, ctx):
        self.ctx = ctx

    def write_documents(self, documents: list[dict]) -> int:
        from ..index.local_store import Document
        n = 0
        for d in documents:
            rid = self.ctx.index_document(Document(
                url=d.get("id") or d.get("url") or f"doc://{n}",
                title=d.get("meta", {}).get("title", ""),
                text=d.get("content", "")), attest=False, credit=False)
            if rid is not None:
                n += 1
        return n

    def filter_documents(self, query: str | None = None,
                         limit: int = 10) -> list[SimpleDocument]:
        if not query:
            docs = list(self.ctx.store.export_documents())[:limit]
            return [SimpleDocument(d.text, {"url": d.url, "title": d.title})
                    for d in docs]
        resp = self.ctx.search(query, limit=limit)
        return [SimpleDocument(getattr(r, "snippet", ""),
                               {"url": getattr(r, "url", "")})
                for r in resp.results]

    def count_documents(self) -> int:
        return self.ctx.store.count()
