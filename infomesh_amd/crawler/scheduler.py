"""Crawl scheduler: politeness-aware URL frontier.

Reference parity: infomesh/crawler/scheduler.py (asyncio queue max 10k,
per-domain pending caps + delays with robots Crawl-delay override capped
at 60 s, global URLs/hour budget, stale-domain pruning at 50k domains).
"""
from __future__ import annotations

import asyncio
import time
from dataclasses import dataclass, field
from urllib.parse import urlparse

MAX_QUEUE = 10_000
MAX_DOMAINS = 50_000
MAX_PENDING_PER_DOMAIN = 100
MAX_DELAY_S = 60.0


@dataclass
class _DomainState:
    last_fetch: float = 0.0
    delay_s: float = 1.0
    pending: int = 0
    touched: float = field(default_factory=time.time)


@dataclass(order=True)
class QueueItem:
    priority: int
    added_at: float
    url: str = field(compare=False)
    depth: int = field(compare=False, default=0)


class Scheduler:
    def __init__(self, politeness_delay_s: float = 1.0,
                 max_urls_per_hour: int = 60, max_depth: int = 3):
        self.default_delay = politeness_delay_s
        self.max_urls_per_hour = max_urls_per_hour
        self.max_depth = max_depth
        self._queue: asyncio.PriorityQueue[QueueItem] = \
            asyncio.PriorityQueue(MAX_QUEUE)
        self._domains: dict[str, _DomainState] = {}
        self._hour_start = time.time()
        self._hour_count = 0
        self.stats = {"added": 0, "rejected": 0, "served": 0}

    def _domain(self, url: str) -> str:
        try:
            return (urlparse(url).hostname or "").lower()
        except ValueError:
            return ""

    def _state(self, domain: str) -> _DomainState:
        st = self._domains.get(domain)
        if st is None:
            if len(self._domains) >= MAX_DOMAINS:
                self._prune()
            st = _DomainState(delay_s=self.default_delay)
            self._domains[domain] = st
        return st

    def _prune(self) -> None:
        """Drop the least-recently-touched half of domain states."""
        items = sorted(self._domains.items(), key=lambda p: p[1].touched)
        for d, _ in items[: len(items) // 2]:
            del self._domains[d]

    def set_crawl_delay(self, domain: str, delay_s: float) -> None:
        self._state(domain.lower()).delay_s = min(
            max(delay_s, self.default_delay), MAX_DELAY_S)

    def add_url(self, url: str, depth: int = 0, priority: int = 5) -> bool:
        if depth > self.max_depth:
            self.stats["rejected"] += 1
            return False
        domain = self._domain(url)
        if not domain:
            self.stats["rejected"] += 1
            return False
        st = self._state(domain)
        if st.pending >= MAX_PENDING_PER_DOMAIN or self._queue.full():
            self.stats["rejected"] += 1
            return False
        st.pending += 1
        st.touched = time.time()
        self._queue.put_nowait(QueueItem(priority, time.time(), url, depth))
        self.stats["added"] += 1
        return True

    def _hour_budget_left(self) -> bool:
        now = time.time()
        if now - self._hour_start > 3600:
            self._hour_start = now
            self._hour_count = 0
        return self._hour_count < self.max_urls_per_hour

    async def get_url(self, timeout: float | None = None
                      ) -> tuple[str, int] | None:
        """Next crawlable URL honoring per-domain delay + hourly budget."""
        deadline = time.time() + timeout if timeout is not None else None
        while True:
            if not self._hour_budget_left():
                await asyncio.sleep(min(5.0, timeout or 5.0))
                if deadline and time.time() > deadline:
                    return None
                continue
            try:
                remaining = None if deadline is None else \
                    max(0.01, deadline - time.time())
                item = await asyncio.wait_for(self._queue.get(),
                                              timeout=remaining)
            except asyncio.TimeoutError:
                return None
            domain = self._domain(item.url)
            st = self._state(domain)
            wait = st.last_fetch + st.delay_s - time.time()
            if wait > 0:
                await asyncio.sleep(min(wait, MAX_DELAY_S))
            st.last_fetch = time.time()
            st.touched = st.last_fetch
            st.pending = max(0, st.pending - 1)
            self._hour_count += 1
            self.stats["served"] += 1
            return item.url, item.depth

    def qsize(self) -> int:
        return self._queue.qsize()
