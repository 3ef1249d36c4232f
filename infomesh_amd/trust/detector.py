"""Malicious-node detection heuristics.

Reference parity: infomesh/trust/detector.py (threat levels +
MaliciousNodeDetector). In the single-node MI355X deployment the
"nodes" under observation are shard peers / snapshot submitters /
same-owner sync partners rather than libp2p peers; the event taxonomy
and escalation ladder are the same so trust decisions stay portable
across snapshot/proof exchanges.
"""
from __future__ import annotations

import enum
import time
from collections import defaultdict, deque
from dataclasses import dataclass, field


class ThreatLevel(enum.IntEnum):
    NONE = 0
    LOW = 1
    MEDIUM = 2
    HIGH = 3
    CRITICAL = 4


# Event kind -> (weight, decay half-life seconds). Weights accumulate
# into a threat score; the ladder below maps score -> level.
EVENT_WEIGHTS: dict[str, tuple[float, float]] = {
    "audit_fail": (3.0, 24 * 3600.0),
    "invalid_signature": (5.0, 7 * 24 * 3600.0),
    "invalid_proof": (4.0, 7 * 24 * 3600.0),
    "fabricated_result": (4.0, 24 * 3600.0),
    "spam": (1.0, 6 * 3600.0),
    "bad_summary": (1.5, 24 * 3600.0),
    "replay": (2.0, 24 * 3600.0),
    "rate_abuse": (1.0, 3600.0),
}

_LADDER = [(12.0, ThreatLevel.CRITICAL), (8.0, ThreatLevel.HIGH),
           (4.0, ThreatLevel.MEDIUM), (1.5, ThreatLevel.LOW)]


@dataclass
class NodeThreat:
    node_id: str
    score: float
    level: ThreatLevel
    events: dict[str, int]
    isolate: bool


@dataclass
class _NodeState:
    events: deque = field(default_factory=lambda: deque(maxlen=512))
    counts: dict = field(default_factory=lambda: defaultdict(int))


class MaliciousNodeDetector:
    """Accumulates weighted, exponentially-decayed misbehavior events
    per node and maps the running score onto a threat ladder.

    `isolate` mirrors the reference semantics: CRITICAL always
    isolates; HIGH isolates after repeat signature/proof forgery
    (non-recoverable classes of misbehavior)."""

    def __init__(self, now=time.time):
        self._now = now
        self._nodes: dict[str, _NodeState] = defaultdict(_NodeState)

    def record(self, node_id: str, kind: str, count: int = 1) -> NodeThreat:
        if kind not in EVENT_WEIGHTS:
            raise ValueError(f"unknown event kind {kind!r}")
        st = self._nodes[node_id]
        t = self._now()
        for _ in range(count):
            st.events.append((t, kind))
        st.counts[kind] += count
        return self.assess(node_id)

    def _score(self, st: _NodeState) -> float:
        t = self._now()
        score = 0.0
        for ts, kind in st.events:
            w, half = EVENT_WEIGHTS[kind]
            score += w * 0.5 ** ((t - ts) / half)
        return score

    def assess(self, node_id: str) -> NodeThreat:
        st = self._nodes[node_id]
        score = self._score(st)
        level = ThreatLevel.NONE
        for thresh, lv in _LADDER:
            if score >= thresh:
                level = lv
                break
        forgery = (st.counts["invalid_signature"]
                   + st.counts["invalid_proof"])
        isolate = (level >= ThreatLevel.CRITICAL
                   or (level >= ThreatLevel.HIGH and forgery >= 2))
        return NodeThreat(node_id=node_id, score=round(score, 3),
                          level=level, events=dict(st.counts),
                          isolate=isolate)

    def threats(self, min_level: ThreatLevel = ThreatLevel.LOW
                ) -> list[NodeThreat]:
        out = [self.assess(n) for n in self._nodes]
        return sorted((x for x in out if x.level >= min_level),
                      key=lambda x: -x.score)

    def clear(self, node_id: str) -> None:
        self._nodes.pop(node_id, None)
