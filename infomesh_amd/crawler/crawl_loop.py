"""The background crawl loop: seeds -> frontier -> crawl -> index.

Reference parity: infomesh/crawler/crawl_loop.py (seed_and_crawl_loop,
feed polling, priority recrawl batches, governor backpressure, idle
re-seeding, hourly FTS optimize).
"""
from __future__ import annotations

import asyncio
import logging
import time

from ..services import AppContext
from ..utils.governor import ResourceGovernor
from .rss import parse_feed
from .seeds import load_seeds

log = logging.getLogger("infomesh.crawl_loop")

FEED_POLL_EVERY_S = 60.0
RECRAWL_EVERY_S = 120.0
OPTIMIZE_EVERY_S = 3600.0
ENGINE_FLUSH_EVERY_S = 60.0
IDLE_RESEED_AFTER_S = 600.0


async def seed_and_crawl_loop(ctx: AppContext,
                              governor: ResourceGovernor | None = None,
                              seed_category: str = "quickstart",
                              max_iterations: int | None = None,
                              stop_check=None) -> dict:
    """Main crawl loop. `max_iterations`/`stop_check` bound it for tests
    and foreground runs."""
    assert ctx.worker is not None, "crawl loop needs a worker role"
    governor = governor or getattr(ctx, "governor", None) \
        or ResourceGovernor()
    sched = ctx.worker.scheduler
    # trigger-driven priority queue (reference freshness.py:67-212):
    # drained before the BFS scheduler so feed/user triggers beat
    # breadth-first discovery
    from .freshness import PriorityRecrawlQueue, RecrawlTrigger
    pq = getattr(ctx, "recrawl_queue", None)
    if pq is None:
        pq = ctx.recrawl_queue = PriorityRecrawlQueue()
    for url in load_seeds(seed_category):
        sched.add_url(url, depth=0)
    last_feed = last_recrawl = last_flush = 0.0
    last_optimize = time.time()
    last_activity = time.time()
    iterations = 0
    stats = {"crawled": 0, "indexed": 0, "errors": 0}
    while True:
        if stop_check is not None and stop_check():
            break
        if max_iterations is not None and iterations >= max_iterations:
            break
        iterations += 1
        now = time.time()

        # governor backpressure (crawl_loop.py:183-221)
        if not governor.crawl_allowed():
            await asyncio.sleep(1.0)
            continue

        # feed polling (crawl_loop.py:38-104)
        if now - last_feed > FEED_POLL_EVERY_S and ctx.feeds.feeds:
            last_feed = now
            for mf in ctx.feeds.due(now)[:5]:
                try:
                    # capped streaming read (feeds are small; a hostile
                    # endpoint must not buffer unbounded XML)
                    client = await ctx.worker._get_client()
                    async with client.stream("GET", mf.url) as r:
                        body = b""
                        if r.status_code == 200:
                            async for chunk in r.aiter_bytes():
                                body += chunk
                                if len(body) > 2_000_000:
                                    break
                    feed = parse_feed(
                        mf.url, body.decode("utf-8", errors="replace")) \
                        if body else None
                except Exception:
                    feed = None
                for item in ctx.feeds.record_poll(mf.url, feed, now):
                    pq.enqueue(item.url, RecrawlTrigger.RSS_UPDATE,
                               source=mf.url)

        # adaptive recrawl feeds the SCHEDULED tier of the queue
        if now - last_recrawl > RECRAWL_EVERY_S:
            last_recrawl = now
            for doc in ctx.store.due_for_recrawl(limit=10):
                pq.enqueue(doc.url, RecrawlTrigger.SCHEDULED)

        # drain the priority queue first (reference crawl_loop.py:110-180)
        pitem = pq.dequeue()
        if pitem is not None:
            item = (pitem.url, -1 if pitem.trigger
                    == RecrawlTrigger.USER_REQUEST else 0)
        else:
            item = await sched.get_url(timeout=2.0)
        if item is None:
            if now - last_activity > IDLE_RESEED_AFTER_S:
                for url in load_seeds(seed_category):
                    sched.add_url(url, depth=0)
                last_activity = now
            await asyncio.sleep(0.2)
            continue
        url, depth = item
        last_activity = now
        try:
            out = await ctx.crawl_and_index(url, depth=depth,
                                            force=depth == -1)
            stats["crawled"] += 1
            if out.get("indexed"):
                stats["indexed"] += 1
        except Exception as e:
            stats["errors"] += 1
            log.warning("crawl %s failed: %s", url, e)

        # GPU ingest flush + hourly FTS optimize (crawl_loop.py:485-496)
        if ctx.engine is not None and now - last_flush > ENGINE_FLUSH_EVERY_S \
                and ctx.engine.pending_count:
            last_flush = now
            ctx.flush_engine()
        if now - last_optimize > OPTIMIZE_EVERY_S:
            last_optimize = now
            ctx.store.optimize()
    if ctx.engine is not None and ctx.engine.pending_count:
        ctx.flush_engine()
    return stats
