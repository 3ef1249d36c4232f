"""Summarization request queue with admission control.

Reference parity: infomesh/summarizer/peer_handler.py (serve summarize
requests from peers with queueing/rejection). In the single-node build
the "peers" are local clients (API/MCP sessions); the same queueing,
rejection and credit-award semantics apply, and the energy-aware
scheduler can defer non-urgent work to off-peak windows.
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field

from ..credits.ledger import Action, CreditLedger
from .engine import SummarizationEngine, SummaryResult

MAX_QUEUE = 32
MAX_TEXT_CHARS = 100_000


@dataclass
class SummarizeRequest:
    text: str
    title: str = ""
    requester: str = ""
    submitted: float = field(default_factory=time.time)


class PeerSummarizeHandler:
    def __init__(self, engine: SummarizationEngine,
                 ledger: CreditLedger | None = None,
                 max_queue: int = MAX_QUEUE,
                 reputation=None):
        self.engine = engine
        self.ledger = ledger
        self.max_queue = max_queue
        # trust/reputation.SummaryReputation: requesters with failed
        # summary grades are deprioritized; our own verified summary
        # quality feeds back through record_quality().
        self.reputation = reputation
        self._lock = threading.Lock()
        self._active = 0
        self.stats = {"served": 0, "rejected": 0}

    def record_quality(self, node_id: str, quality: float) -> None:
        if self.reputation is not None:
            self.reputation.record(node_id, quality)

    def handle(self, req: SummarizeRequest) -> SummaryResult | None:
        """Serve or reject (None) a summarize request."""
        if not req.text or len(req.text) > MAX_TEXT_CHARS:
            self.stats["rejected"] += 1
            return None
        with self._lock:
            if self._active >= self.max_queue:
                self.stats["rejected"] += 1
                return None
            self._active += 1
        try:
            result = self.engine.summarize(req.text, title=req.title)
            if self.ledger is not None:
                self.ledger.record_action(Action.LLM_SUMMARIZE, 1.0)
            self.stats["served"] += 1
            return result
        finally:
            with self._lock:
                self._active -= 1
