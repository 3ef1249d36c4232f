"""robots.txt checking with caching.

Reference parity: infomesh/crawler/robots.py (RobotsChecker: fetch/cache
robots.txt, is_allowed, get_crawl_delay, get_sitemaps).
"""
from __future__ import annotations

import time
import urllib.robotparser
from urllib.parse import urlparse, urlunparse

import httpx

CACHE_TTL_S = 3600.0


class RobotsChecker:
    def __init__(self, user_agent: str = "infomesh-amd",
                 client: httpx.AsyncClient | None = None,
                 ttl_s: float = CACHE_TTL_S):
        self.user_agent = user_agent
        self._client = client
        self._ttl = ttl_s
        # host -> (fetched_at, parser|None, sitemaps)
        self._cache: dict[str, tuple[float, urllib.robotparser.RobotFileParser | None, list[str]]] = {}

    def _robots_url(self, url: str) -> tuple[str, str]:
        p = urlparse(url)
        host = f"{p.scheme}://{p.netloc}"
        return host, urlunparse((p.scheme, p.netloc, "/robots.txt", "", "", ""))

    async def _get(self, url: str) -> tuple[urllib.robotparser.RobotFileParser | None, list[str]]:
        host, robots_url = self._robots_url(url)
        cached = self._cache.get(host)
        if cached and time.time() - cached[0] < self._ttl:
            return cached[1], cached[2]
        parser: urllib.robotparser.RobotFileParser | None = None
        sitemaps: list[str] = []
        try:
            client = self._client or httpx.AsyncClient(timeout=10.0)
            try:
                resp = await client.get(robots_url, follow_redirects=True)
            finally:
                if self._client is None:
                    await client.aclose()
            if resp.status_code == 200 and len(resp.text) < 1_000_000:
                parser = urllib.robotparser.RobotFileParser()
                lines = resp.text.splitlines()
                parser.parse(lines)
                sitemaps = [ln.split(":", 1)[1].strip()
                            for ln in lines
                            if ln.lower().startswith("sitemap:")]
            elif resp.status_code in (401, 403):
                # restricted robots => disallow all (conservative)
                parser = urllib.robotparser.RobotFileParser()
                parser.parse(["User-agent: *", "Disallow: /"])
        except (httpx.HTTPError, OSError):
            parser = None  # unreachable robots => allow (reference behavior)
        self._cache[host] = (time.time(), parser, sitemaps)
        return parser, sitemaps

    async def is_allowed(self, url: str) -> bool:
        parser, _ = await self._get(url)
        if parser is None:
            return True
        return parser.can_fetch(self.user_agent, url)

    async def get_crawl_delay(self, url: str) -> float | None:
        parser, _ = await self._get(url)
        if parser is None:
            return None
        delay = parser.crawl_delay(self.user_agent)
        if delay is None:
            delay = parser.crawl_delay("*")
        return float(delay) if delay is not None else None

    async def get_sitemaps(self, url: str) -> list[str]:
        _, sitemaps = await self._get(url)
        return sitemaps

    def load_parsed(self, host_url: str, robots_text: str) -> None:
        """Inject pre-fetched robots content (tests / offline)."""
        host, _ = self._robots_url(host_url)
        parser = urllib.robotparser.RobotFileParser()
        lines = robots_text.splitlines()
        parser.parse(lines)
        sitemaps = [ln.split(":", 1)[1].strip() for ln in lines
                    if ln.lower().startswith("sitemap:")]
        self._cache[host] = (time.time(), parser, sitemaps)
