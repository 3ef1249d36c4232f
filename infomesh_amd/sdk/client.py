"""SDK clients: sync + async access to a running node's admin API.

Reference parity: infomesh/sdk/client.py (InfoMeshClient search/crawl/
suggest/status over the local HTTP API).
"""
from __future__ import annotations

from typing import Any

import httpx

DEFAULT_BASE = "http://127.0.0.1:8080"


class InfoMeshClient:
    def __init__(self, base_url: str = DEFAULT_BASE, api_key: str = "",
                 timeout: float = 30.0,
                 transport: httpx.BaseTransport | None = None):
        headers = {"x-api-key": api_key} if api_key else {}
        self._client = httpx.Client(base_url=base_url, headers=headers,
                                    timeout=timeout, transport=transport)

    def search(self, query: str, limit: int = 10,
               mode: str = "auto") -> list[dict[str, Any]]:
        r = self._client.get("/search",
                             params={"q": query, "limit": limit, "mode": mode})
        r.raise_for_status()
        return r.json()["results"]

    def status(self) -> dict[str, Any]:
        r = self._client.get("/status")
        r.raise_for_status()
        return r.json()

    def index_stats(self) -> dict[str, Any]:
        r = self._client.get("/index/stats")
        r.raise_for_status()
        return r.json()

    def credits(self) -> dict[str, Any]:
        r = self._client.get("/credits/balance")
        r.raise_for_status()
        return r.json()

    def feedback(self, url: str, signal: str, query: str = "") -> bool:
        r = self._client.post("/feedback", params={
            "url": url, "signal": signal, "q": query})
        r.raise_for_status()
        return bool(r.json().get("recorded"))

    def health(self) -> bool:
        try:
            return bool(self._client.get("/health").json().get("ok"))
        except (httpx.HTTPError, ValueError):
            return False

    def close(self) -> None:
        self._client.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
        return False


class AsyncInfoMeshClient:
    def __init__(self, base_url: str = DEFAULT_BASE, api_key: str = "",
                 timeout: float = 30.0,
                 transport: httpx.AsyncBaseTransport | None = None):
        headers = {"x-api-key": api_key} if api_key else {}
        self._client = httpx.AsyncClient(base_url=base_url, headers=headers,
                                         timeout=timeout,
                                         transport=transport)

    async def search(self, query: str, limit: int = 10,
                     mode: str = "auto") -> list[dict[str, Any]]:
        r = await self._client.get(
            "/search", params={"q": query, "limit": limit, "mode": mode})
        r.raise_for_status()
        return r.json()["results"]

    async def status(self) -> dict[str, Any]:
        r = await self._client.get("/status")
        r.raise_for_status()
        return r.json()

    async def close(self) -> None:
        await self._client.aclose()

    async def __aenter__(self):
        return self

    async def __aexit__(self, *exc):
        await self.close()
        return False
