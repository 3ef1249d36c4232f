"""Freshness classification + trigger-driven priority recrawl queue.

Reference parity: infomesh/crawler/freshness.py:23-212 — freshness
tiers, recrawl triggers (RSS/user/content-change/scheduled), and a
deduplicating min-heap priority queue consumed by the crawl loop ahead
of the regular scheduler. The conditional-GET header plumbing lives in
crawler/worker.py (etag/last_modified pass-through) here.
"""
from __future__ import annotations

import enum
import heapq
import threading
import time
from dataclasses import dataclass, field

HOUR = 3600.0
DAY = 24 * HOUR


class FreshnessTier(str, enum.Enum):
    HOT = "hot"          # changes within hours (news, feeds)
    WARM = "warm"        # changes within days
    COLD = "cold"        # changes within weeks
    STALE = "stale"      # no observed change for a long time


def classify_freshness(change_interval_s: float | None,
                       age_s: float) -> FreshnessTier:
    """Tier from the adaptive change interval (LocalStore recrawl
    metadata) and how long ago we last fetched."""
    if change_interval_s is not None:
        if change_interval_s <= 6 * HOUR:
            return FreshnessTier.HOT
        if change_interval_s <= 3 * DAY:
            return FreshnessTier.WARM
        if change_interval_s <= 21 * DAY:
            return FreshnessTier.COLD
        return FreshnessTier.STALE
    # no change history: age decides
    if age_s <= DAY:
        return FreshnessTier.HOT
    if age_s <= 7 * DAY:
        return FreshnessTier.WARM
    if age_s <= 30 * DAY:
        return FreshnessTier.COLD
    return FreshnessTier.STALE


class RecrawlTrigger(str, enum.Enum):
    USER_REQUEST = "user_request"        # explicit crawl_url(force=True)
    RSS_UPDATE = "rss_update"            # new item seen in a feed
    SITEMAP_UPDATE = "sitemap_update"    # lastmod moved in a sitemap
    CONTENT_CHANGE = "content_change"    # diff detected on refetch
    SCHEDULED = "scheduled"              # adaptive interval elapsed


TRIGGER_PRIORITY: dict[RecrawlTrigger, int] = {
    RecrawlTrigger.USER_REQUEST: 0,
    RecrawlTrigger.RSS_UPDATE: 1,
    RecrawlTrigger.SITEMAP_UPDATE: 2,
    RecrawlTrigger.CONTENT_CHANGE: 3,
    RecrawlTrigger.SCHEDULED: 4,
}


@dataclass(frozen=True, order=True)
class RecrawlItem:
    priority: int
    enqueued_at: float
    url: str = field(compare=False)
    trigger: RecrawlTrigger = field(compare=False)
    source: str = field(default="", compare=False)   # feed url etc.


class PriorityRecrawlQueue:
    """Deduplicating min-heap of recrawl candidates, drained by the
    crawl loop BEFORE the BFS scheduler each tick (so feed-triggered
    refreshes beat breadth-first discovery). Thread-safe: the feed
    monitor enqueues from its own poll task."""

    def __init__(self, max_size: int = 10_000):
        self._heap: list[RecrawlItem] = []
        self._urls: set[str] = set()
        self._max_size = max_size
        self._lock = threading.Lock()
        self.total_enqueued = 0
        self.total_dequeued = 0

    def enqueue(self, url: str, trigger: RecrawlTrigger,
                source: str = "") -> bool:
        """False if duplicate or full (unless the trigger outranks the
        current worst item, which it then evicts)."""
        with self._lock:
            if url in self._urls:
                return False
            item = RecrawlItem(TRIGGER_PRIORITY[trigger], time.time(),
                               url, trigger, source)
            if len(self._heap) >= self._max_size:
                worst = max(self._heap)
                if item.priority >= worst.priority:
                    return False
                self._heap.remove(worst)
                self._urls.discard(worst.url)
                heapq.heapify(self._heap)
            heapq.heappush(self._heap, item)
            self._urls.add(url)
            self.total_enqueued += 1
            return True

    def dequeue(self) -> RecrawlItem | None:
        with self._lock:
            if not self._heap:
                return None
            item = heapq.heappop(self._heap)
            self._urls.discard(item.url)
            self.total_dequeued += 1
            return item

    def dequeue_batch(self, n: int) -> list[RecrawlItem]:
        out = []
        for _ in range(n):
            item = self.dequeue()
            if item is None:
                break
            out.append(item)
        return out

    def peek(self) -> RecrawlItem | None:
        with self._lock:
            return self._heap[0] if self._heap else None

    def discard(self, url: str) -> None:
        with self._lock:
            if url in self._urls:
                self._heap = [i for i in self._heap if i.url != url]
                heapq.heapify(self._heap)
                self._urls.discard(url)

    def __len__(self) -> int:
        return len(self._heap)

    def stats(self) -> dict:
        with self._lock:
            by_trigger: dict[str, int] = {}
            for i in self._heap:
                by_trigger[i.trigger.value] = \
                    by_trigger.get(i.trigger.value, 0) + 1
        return {"size": len(self._heap),
                "enqueued": self.total_enqueued,
                "dequeued": self.total_dequeued,
                "by_trigger": by_trigger}
