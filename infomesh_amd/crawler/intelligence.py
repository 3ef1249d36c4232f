"""Adaptive crawl intelligence: per-domain speed tuning from observed
outcomes (reference parity: infomesh/crawler/intelligence.py — robots
cache lives in robots.py; this is the adaptive crawl-speed tuner)."""
from __future__ import annotations

import time
from dataclasses import dataclass, field

MIN_DELAY_S = 0.25
MAX_DELAY_S = 60.0


@dataclass
class DomainSpeed:
    delay_s: float = 1.0
    ok: int = 0
    errors: int = 0
    last_latency_ms: float = 0.0
    updated: float = field(default_factory=time.time)


class AdaptiveCrawlTuner:
    """Speeds up well-behaved domains, backs off on errors/slowness.

    record() feeds fetch outcomes; delay_for() is consumed by the
    Scheduler's per-domain politeness (scheduler.set_crawl_delay)."""

    def __init__(self, base_delay_s: float = 1.0, max_domains: int = 10_000):
        self.base = base_delay_s
        self.max_domains = max_domains
        self._domains: dict[str, DomainSpeed] = {}

    def _get(self, domain: str) -> DomainSpeed:
        d = self._domains.get(domain)
        if d is None:
            if len(self._domains) >= self.max_domains:
                oldest = min(self._domains, key=lambda k: self._domains[k].updated)
                del self._domains[oldest]
            d = DomainSpeed(delay_s=self.base)
            self._domains[domain] = d
        return d

    def record(self, domain: str, ok: bool, latency_ms: float = 0.0,
               status: int = 200) -> None:
        d = self._get(domain)
        d.updated = time.time()
        d.last_latency_ms = latency_ms
        if not ok or status in (429, 503):
            d.errors += 1
            factor = 4.0 if status in (429, 503) else 2.0
            d.delay_s = min(MAX_DELAY_S, d.delay_s * factor)
        else:
            d.ok += 1
            if d.ok % 5 == 0 and latency_ms < 2000:
                d.delay_s = max(MIN_DELAY_S, d.delay_s * 0.8)

    def delay_for(self, domain: str) -> float:
        d = self._domains.get(domain)
        return d.delay_s if d else self.base

    def stats(self) -> dict:
        return {dom: {"delay_s": round(d.delay_s, 2), "ok": d.ok,
                      "errors": d.errors}
                for dom, d in sorted(self._domains.items())}
