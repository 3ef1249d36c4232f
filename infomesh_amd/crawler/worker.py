"""Crawl worker: async fetch -> parse -> dedup pipeline.

Reference parity: infomesh/crawler/worker.py (SSRF validation with DNS,
URL dedup check, robots + per-domain Crawl-delay, retry on 5xx with
exponential backoff (2 retries), content-type/size guards, extraction,
canonical-URL redirect handling, exact-hash + SimHash dedup, BFS link
extraction + scheduling, RSS feed discovery). The DHT crawl-lock step
is dropped: intra-node shard ownership makes it unnecessary
(SURVEY.md §2.10 axis 4).
"""
from __future__ import annotations

import asyncio
import logging
from dataclasses import dataclass, field

import httpx

from ..config import CrawlConfig
from ..errors import InfoMeshError
from ..security import validate_url, validate_url_post_redirect
from .dedup import DeduplicatorDB, normalize_url
from .parser import ParsedPage, extract_content, looks_like_js_app
from .robots import RobotsChecker
from .scheduler import Scheduler

log = logging.getLogger("infomesh.crawler")

RETRY_BACKOFF_S = (1.0, 4.0)
TEXT_CONTENT_TYPES = ("text/html", "application/xhtml", "text/plain")


@dataclass
class FetchedResponse:
    """What crawl_url needs from an HTTP response (body already capped
    by the streaming reader)."""
    status_code: int
    headers: dict[str, str]
    content: bytes
    encoding: str


@dataclass
class CrawlResult:
    url: str
    status: str                     # ok | skipped | error
    reason: str = ""
    page: ParsedPage | None = None
    links_scheduled: int = 0
    feeds: list[str] = field(default_factory=list)
    http_status: int = 0
    etag: str = ""
    last_modified: str = ""
    not_modified: bool = False


class CrawlWorker:
    def __init__(self, config: CrawlConfig | None = None,
                 scheduler: Scheduler | None = None,
                 dedup: DeduplicatorDB | None = None,
                 robots: RobotsChecker | None = None,
                 client: httpx.AsyncClient | None = None,
                 retry_backoff_s: tuple[float, ...] = RETRY_BACKOFF_S,
                 resolve_dns: bool = True):
        self.cfg = config or CrawlConfig()
        self.retry_backoff_s = retry_backoff_s
        self.resolve_dns = resolve_dns
        self.scheduler = scheduler or Scheduler(
            self.cfg.politeness_delay_s, self.cfg.max_urls_per_hour,
            self.cfg.max_depth)
        self.dedup = dedup or DeduplicatorDB()
        self.robots = robots or RobotsChecker(self.cfg.user_agent)
        self._client = client
        self._own_client = client is None
        self.stats = {"crawled": 0, "skipped": 0, "errors": 0}

    async def _get_client(self) -> httpx.AsyncClient:
        if self._client is None:
            self._client = httpx.AsyncClient(
                timeout=self.cfg.timeout_s,
                headers={"User-Agent": self.cfg.user_agent},
                follow_redirects=False,
                limits=httpx.Limits(
                    max_connections=self.cfg.max_concurrent * 2))
        return self._client

    async def close(self) -> None:
        if self._client is not None and self._own_client:
            await self._client.aclose()
            self._client = None

    async def _fetch(self, url: str, etag: str = "",
                     last_modified: str = "") -> "FetchedResponse":
        """GET with redirect re-validation, 5xx retries, and a STREAMING
        size guard: at most max_response_bytes are ever read off the
        wire, so a huge/hostile body cannot spike memory."""
        client = await self._get_client()
        headers = {}
        if etag:
            headers["If-None-Match"] = etag
        if last_modified:
            headers["If-Modified-Since"] = last_modified
        current = url
        resp = None
        for attempt in range(self.cfg.retries + 1):
            redirects = 0
            while True:
                async with client.stream("GET", current,
                                         headers=headers) as r:
                    if r.status_code in (301, 302, 303, 307, 308):
                        redirects += 1
                        if redirects > 5:
                            raise InfoMeshError("CRWL003",
                                                "too many redirects")
                        location = r.headers.get("location", "")
                        current = str(httpx.URL(current).join(location))
                        # DNS-resolve the redirect target: a hostname
                        # resolving to a private/internal IP is the
                        # classic SSRF/DNS-rebinding hop (reference
                        # security.py validate_url_post_redirect forces
                        # resolve_dns=True for exactly this)
                        validate_url_post_redirect(
                            current, resolve_dns=self.resolve_dns)
                        continue
                    body = b""
                    if r.status_code == 200:
                        cap = self.cfg.max_response_bytes
                        async for chunk in r.aiter_bytes():
                            body += chunk
                            if len(body) >= cap:
                                body = body[:cap]
                                break
                    resp = FetchedResponse(
                        status_code=r.status_code,
                        headers=dict(r.headers),
                        content=body,
                        encoding=r.encoding or "utf-8")
                    break
            if resp.status_code >= 500 and attempt < self.cfg.retries:
                bo = self.retry_backoff_s
                await asyncio.sleep(bo[min(attempt, len(bo) - 1)])
                continue
            return resp
        return resp

    async def crawl_url(self, url: str, depth: int = 0,
                        force: bool = False,
                        etag: str = "", last_modified: str = ""
                        ) -> CrawlResult:
        # 1. SSRF guard (DNS resolution included)
        try:
            validate_url(url, resolve_dns=self.resolve_dns)
        except InfoMeshError as e:
            self.stats["skipped"] += 1
            return CrawlResult(url, "skipped", f"ssrf: {e.detail}")

        # 2. URL dedup
        if not force and self.dedup.seen_url(url):
            self.stats["skipped"] += 1
            return CrawlResult(url, "skipped", "url already crawled")

        # 3. robots + crawl delay
        if self.cfg.respect_robots:
            if not await self.robots.is_allowed(url):
                self.stats["skipped"] += 1
                return CrawlResult(url, "skipped", "robots disallow")
            delay = await self.robots.get_crawl_delay(url)
            if delay:
                from urllib.parse import urlparse
                self.scheduler.set_crawl_delay(
                    (urlparse(url).hostname or ""), delay)

        # 4. fetch
        try:
            resp = await self._fetch(url, etag, last_modified)
        except InfoMeshError as e:
            self.stats["errors"] += 1
            return CrawlResult(url, "error", str(e))
        except (httpx.HTTPError, OSError) as e:
            self.stats["errors"] += 1
            return CrawlResult(url, "error", f"fetch: {e}")

        if resp.status_code == 304:
            return CrawlResult(url, "skipped", "not modified",
                               http_status=304, not_modified=True)
        if resp.status_code != 200:
            self.stats["errors"] += 1
            return CrawlResult(url, "error", f"http {resp.status_code}",
                               http_status=resp.status_code)

        ctype = resp.headers.get("content-type", "").lower()
        if ctype and not any(t in ctype for t in TEXT_CONTENT_TYPES):
            self.stats["skipped"] += 1
            return CrawlResult(url, "skipped", f"content-type {ctype}",
                               http_status=200)
        try:
            html = resp.content.decode(resp.encoding, errors="replace")
        except LookupError:
            html = resp.content.decode("utf-8", errors="replace")

        # 5. extract
        page = extract_content(url, html)
        if looks_like_js_app(html, page.text):
            log.info("js-app detected (no renderer in image): %s", url)
        if len(page.text) < 50:
            self.stats["skipped"] += 1
            self.dedup.record(url)  # don't refetch thin pages constantly
            return CrawlResult(url, "skipped", "no substantial text",
                               http_status=200, page=page)

        # 6. canonical redirect handling
        if page.canonical and normalize_url(page.canonical) != normalize_url(url):
            if self.dedup.seen_url(page.canonical):
                self.dedup.record(url, page.text)
                self.stats["skipped"] += 1
                return CrawlResult(url, "skipped",
                                   f"canonical dup {page.canonical}",
                                   page=page)

        # 7. content dedup (exact + near)
        if not force:
            if self.dedup.seen_content(page.text):
                self.dedup.record(url, page.text)
                self.stats["skipped"] += 1
                return CrawlResult(url, "skipped", "exact duplicate",
                                   page=page)
            near, fp = self.dedup.near_duplicate(page.text)
            if near:
                self.dedup.record(url, page.text, fp)
                self.stats["skipped"] += 1
                return CrawlResult(url, "skipped", "near duplicate",
                                   page=page)
        self.dedup.record(url, page.text)

        # 8. BFS link scheduling
        scheduled = 0
        if depth < self.cfg.max_depth:
            for link in page.links:
                try:
                    validate_url(link)
                except InfoMeshError:
                    continue
                if not self.dedup.seen_url(link):
                    if self.scheduler.add_url(link, depth + 1):
                        scheduled += 1

        self.stats["crawled"] += 1
        return CrawlResult(
            url, "ok", page=page, links_scheduled=scheduled,
            feeds=page.feeds, http_status=200,
            etag=resp.headers.get("etag", ""),
            last_modified=resp.headers.get("last-modified", ""))
