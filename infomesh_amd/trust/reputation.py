"""Per-node summarization-quality reputation.

Reference parity: infomesh/trust/reputation.py (per-peer LLM summary
grades feeding the 0.20 summary-quality term of the trust score,
trust/scoring.py). Each verified summary (summarizer/verify.py score in
[0,1]) updates an EWMA grade; grades gate whether a node's summaries
are accepted and feed TrustScorer.summary_quality.
"""
from __future__ import annotations

import json
import time
from dataclasses import dataclass
from pathlib import Path

# EWMA smoothing: recent summaries dominate after ~10 samples.
ALPHA = 0.2
GRADES = [(0.9, "A"), (0.75, "B"), (0.6, "C"), (0.4, "D")]
MIN_SAMPLES_FOR_GRADE = 3
ACCEPT_THRESHOLD = 0.4  # grade D or better keeps summaries accepted


@dataclass
class Reputation:
    node_id: str
    ewma: float
    samples: int
    grade: str
    accept: bool
    updated_at: float


def grade_of(ewma: float, samples: int) -> str:
    if samples < MIN_SAMPLES_FOR_GRADE:
        return "?"
    for thresh, g in GRADES:
        if ewma >= thresh:
            return g
    return "F"


class SummaryReputation:
    """EWMA reputation tracker with optional JSON persistence."""

    def __init__(self, path: Path | None = None, now=time.time):
        self._path = path
        self._now = now
        self._state: dict[str, dict] = {}
        if path is not None and path.exists():
            try:
                self._state = json.loads(path.read_text())
            except (ValueError, OSError):
                self._state = {}

    def record(self, node_id: str, quality: float) -> Reputation:
        """quality in [0,1] from summarizer/verify.py verification."""
        q = max(0.0, min(1.0, float(quality)))
        st = self._state.get(node_id)
        if st is None:
            st = {"ewma": q, "samples": 0}
        else:
            st["ewma"] = (1 - ALPHA) * st["ewma"] + ALPHA * q
        st["samples"] = st.get("samples", 0) + 1
        st["updated_at"] = self._now()
        self._state[node_id] = st
        self._save()
        return self.get(node_id)

    def get(self, node_id: str) -> Reputation:
        st = self._state.get(node_id)
        if st is None:
            return Reputation(node_id, 0.5, 0, "?", True, 0.0)
        g = grade_of(st["ewma"], st["samples"])
        accept = (st["samples"] < MIN_SAMPLES_FOR_GRADE
                  or st["ewma"] >= ACCEPT_THRESHOLD)
        return Reputation(node_id, round(st["ewma"], 4), st["samples"],
                          g, accept, st.get("updated_at", 0.0))

    def summary_quality(self, node_id: str) -> float:
        """The [0,1] term consumed by trust/scoring.py (weight 0.20)."""
        return self.get(node_id).ewma

    def sync_to_trust(self, trust) -> int:
        """Push each node's summary-quality EWMA into the TrustStore's
        0.20-weighted component (the reputation->trust bridge the
        reference runs in its maintenance loop). Returns nodes synced."""
        n = 0
        for node_id in list(self._state):
            rep = self.get(node_id)
            if rep.samples >= 1:
                trust.update_component(node_id, "summary_quality",
                                       rep.ewma)
                n += 1
        return n

    def leaderboard(self) -> list[Reputation]:
        return sorted((self.get(n) for n in self._state),
                      key=lambda r: -r.ewma)

    def _save(self) -> None:
        if self._path is None:
            return
        tmp = self._path.with_suffix(".tmp")
        tmp.write_text(json.dumps(self._state))
        tmp.replace(self._path)
