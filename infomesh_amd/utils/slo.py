"""SLO tracking (reference parity: infomesh/slo.py — rolling windows of
latency/error observations against objectives)."""
from __future__ import annotations

import time
from dataclasses import dataclass, field


@dataclass
class SLO:
    name: str
    target_p95_ms: float | None = None
    target_success_rate: float | None = None


@dataclass
class SLOTracker:
    window_s: float = 3600.0
    slos: dict[str, SLO] = field(default_factory=dict)
    _samples: dict[str, list[tuple[float, float, bool]]] = \
        field(default_factory=dict)

    def define(self, name: str, target_p95_ms: float | None = None,
               target_success_rate: float | None = None) -> None:
        self.slos[name] = SLO(name, target_p95_ms, target_success_rate)

    def record(self, name: str, latency_ms: float, ok: bool = True) -> None:
        lst = self._samples.setdefault(name, [])
        lst.append((time.time(), latency_ms, ok))
        cutoff = time.time() - self.window_s
        while lst and lst[0][0] < cutoff:
            lst.pop(0)

    def report(self) -> dict:
        out = {}
        for name, slo in self.slos.items():
            samples = self._samples.get(name, [])
            if not samples:
                out[name] = {"samples": 0, "met": True}
                continue
            lats = sorted(s[1] for s in samples)
            p95 = lats[min(len(lats) - 1, int(len(lats) * 0.95))]
            success = sum(1 for s in samples if s[2]) / len(samples)
            met = True
            if slo.target_p95_ms is not None and p95 > slo.target_p95_ms:
                met = False
            if slo.target_success_rate is not None and \
                    success < slo.target_success_rate:
                met = False
            out[name] = {"samples": len(samples), "p95_ms": round(p95, 2),
                         "success_rate": round(success, 4), "met": met}
        return out
