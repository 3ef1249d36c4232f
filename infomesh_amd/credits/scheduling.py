"""Energy-aware scheduling: off-peak windows route LLM-heavy work.

Reference parity: infomesh/credits/scheduling.py (off-peak default
23:00–07:00 local; EnergyAwareScheduler).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field

OFF_PEAK_START_H = 23
OFF_PEAK_END_H = 7


def is_off_peak(ts: float | None = None,
                start_h: int = OFF_PEAK_START_H,
                end_h: int = OFF_PEAK_END_H) -> bool:
    lt = time.localtime(ts if ts is not None else time.time())
    h = lt.tm_hour
    if start_h > end_h:     # wraps midnight
        return h >= start_h or h < end_h
    return start_h <= h < end_h


@dataclass
class EnergyAwareScheduler:
    """Defers deferrable (LLM) work to off-peak windows; urgent work
    runs immediately."""
    start_h: int = OFF_PEAK_START_H
    end_h: int = OFF_PEAK_END_H
    deferred: list[tuple[str, dict]] = field(default_factory=list)
    max_deferred: int = 10_000

    def submit(self, kind: str, payload: dict, urgent: bool = False,
               now: float | None = None) -> bool:
        """Returns True when the task should run NOW."""
        if urgent or is_off_peak(now, self.start_h, self.end_h):
            return True
        if len(self.deferred) < self.max_deferred:
            self.deferred.append((kind, payload))
        return False

    def drain(self, now: float | None = None) -> list[tuple[str, dict]]:
        """Off-peak: hand back everything deferred."""
        if not is_off_peak(now, self.start_h, self.end_h):
            return []
        out, self.deferred = self.deferred, []
        return out
