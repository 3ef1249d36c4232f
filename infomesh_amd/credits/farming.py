"""Credit-farming detection: statistical anomaly checks + probation.

Reference parity: infomesh/credits/farming.py (anomaly detection + 24 h
probation windows).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field

PROBATION_S = 24 * 3600.0
# plausibility rates (per hour) per action
MAX_RATES = {"crawl": 2000.0, "query_served": 36_000.0,
             "llm_summarize": 600.0, "llm_rerank": 3600.0}
BURST_WINDOW_S = 60.0
BURST_FACTOR = 10.0


@dataclass
class FarmingDetector:
    events: dict[str, list[float]] = field(default_factory=dict)
    probation_until: float = 0.0
    flags: list[str] = field(default_factory=list)

    def record(self, action: str, ts: float | None = None) -> None:
        ts = ts if ts is not None else time.time()
        lst = self.events.setdefault(action, [])
        lst.append(ts)
        cutoff = ts - 3600.0
        while lst and lst[0] < cutoff:
            lst.pop(0)
        self._check(action, ts)

    def _check(self, action: str, now: float) -> None:
        lst = self.events.get(action, [])
        max_rate = MAX_RATES.get(action)
        if max_rate and len(lst) > max_rate:
            self._flag(f"hourly rate exceeded for {action}", now)
        recent = [t for t in lst if t > now - BURST_WINDOW_S]
        if max_rate and len(recent) > max_rate / 3600.0 * BURST_WINDOW_S * BURST_FACTOR \
                and len(recent) > 30:
            self._flag(f"burst anomaly for {action}", now)

    def _flag(self, reason: str, now: float) -> None:
        if reason not in self.flags:
            self.flags.append(reason)
        self.probation_until = now + PROBATION_S

    def on_probation(self, now: float | None = None) -> bool:
        return (now or time.time()) < self.probation_until

    def multiplier(self, now: float | None = None) -> float:
        """Credits earned during probation are zeroed."""
        return 0.0 if self.on_probation(now) else 1.0
