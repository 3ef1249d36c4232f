"""RSS/Atom feed discovery and parsing.

Reference parity: infomesh/crawler/rss.py (feed autodiscovery from HTML,
RSS/Atom XML parse) + crawler/feed_monitor.py (priority tiers, polling).
"""
from __future__ import annotations

import time
import xml.etree.ElementTree as ET
from dataclasses import dataclass, field

ATOM_NS = "{http://www.w3.org/2005/Atom}"


@dataclass
class FeedItem:
    title: str
    url: str
    published: str = ""


@dataclass
class Feed:
    url: str
    title: str = ""
    items: list[FeedItem] = field(default_factory=list)


def parse_feed(url: str, xml_text: str, max_items: int = 50) -> Feed | None:
    """Parse RSS 2.0 or Atom."""
    try:
        root = ET.fromstring(xml_text)
    except ET.ParseError:
        return None
    feed = Feed(url=url)
    if root.tag in ("rss", "rdf:RDF") or root.tag.endswith("rss"):
        channel = root.find("channel")
        if channel is None:
            return None
        feed.title = (channel.findtext("title") or "").strip()
        for item in channel.findall("item")[:max_items]:
            link = (item.findtext("link") or "").strip()
            if link:
                feed.items.append(FeedItem(
                    title=(item.findtext("title") or "").strip(),
                    url=link,
                    published=(item.findtext("pubDate") or "").strip()))
    elif root.tag == ATOM_NS + "feed":
        feed.title = (root.findtext(ATOM_NS + "title") or "").strip()
        for entry in root.findall(ATOM_NS + "entry")[:max_items]:
            link_el = entry.find(ATOM_NS + "link")
            href = link_el.get("href") if link_el is not None else None
            if href:
                feed.items.append(FeedItem(
                    title=(entry.findtext(ATOM_NS + "title") or "").strip(),
                    url=href,
                    published=(entry.findtext(ATOM_NS + "updated") or "").strip()))
    else:
        return None
    return feed


# Priority tiers in minutes (reference: feed_monitor.py:25)
POLL_TIERS_MIN = (1, 5, 15, 60)


@dataclass
class MonitoredFeed:
    url: str
    tier: int = 3                  # slowest by default
    last_poll: float = 0.0
    last_items: set[str] = field(default_factory=set)
    failures: int = 0


class FeedMonitor:
    """Tracks feeds and reports which are due + which items are new.

    Optionally persistent (path=...): discovered feeds survive restarts
    — the reference's `infomesh feeds list/import` CLI expects that."""

    def __init__(self, path=None):
        self.feeds: dict[str, MonitoredFeed] = {}
        self._path = path
        if path is not None:
            self._load()

    def _load(self) -> None:
        import json
        from pathlib import Path
        p = Path(self._path)
        if not p.exists():
            return
        try:
            data = json.loads(p.read_text())
        except (ValueError, OSError):
            return
        for d in data.get("feeds", []):
            self.feeds[d["url"]] = MonitoredFeed(
                url=d["url"], tier=int(d.get("tier", 3)),
                last_poll=float(d.get("last_poll", 0.0)),
                failures=int(d.get("failures", 0)))

    def save(self) -> None:
        if self._path is None:
            return
        import json
        from pathlib import Path
        p = Path(self._path)
        tmp = p.with_suffix(".tmp")
        tmp.write_text(json.dumps({"feeds": [
            {"url": f.url, "tier": f.tier, "last_poll": f.last_poll,
             "failures": f.failures} for f in self.feeds.values()]}))
        tmp.replace(p)

    def add(self, url: str, tier: int = 3) -> None:
        tier = min(max(tier, 0), len(POLL_TIERS_MIN) - 1)
        if url not in self.feeds:
            self.feeds[url] = MonitoredFeed(url=url, tier=tier)
            self.save()

    def remove(self, url: str) -> None:
        if self.feeds.pop(url, None) is not None:
            self.save()

    def due(self, now: float | None = None) -> list[MonitoredFeed]:
        now = now or time.time()
        return [f for f in self.feeds.values()
                if now - f.last_poll >= POLL_TIERS_MIN[f.tier] * 60]

    def record_poll(self, url: str, feed: Feed | None,
                    now: float | None = None) -> list[FeedItem]:
        """Record a poll result; returns the NEW items. Feeds with fresh
        items are promoted a tier; stale ones demoted."""
        mf = self.feeds.get(url)
        if mf is None:
            return []
        mf.last_poll = now or time.time()
        if feed is None:
            mf.failures += 1
            if mf.failures >= 5:
                mf.tier = len(POLL_TIERS_MIN) - 1
            return []
        mf.failures = 0
        new = [it for it in feed.items if it.url not in mf.last_items]
        mf.last_items = {it.url for it in feed.items}
        if new:
            mf.tier = max(0, mf.tier - 1)
        else:
            mf.tier = min(len(POLL_TIERS_MIN) - 1, mf.tier + 1)
        return new

    def import_opml(self, opml_xml: str) -> int:
        """OPML feed-list import (reference: feed_monitor.py OPML)."""
        try:
            root = ET.fromstring(opml_xml)
        except ET.ParseError:
            return 0
        n = 0
        for outline in root.iter("outline"):
            url = outline.get("xmlUrl")
            if url:
                self.add(url)
                n += 1
        return n
