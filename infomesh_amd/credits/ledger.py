"""Credit ledger: C = Σ W·Q·M over contribution actions.

Reference parity: infomesh/credits/ledger.py + credits/types.py —
action weights (crawl 1.0/page, query 0.5, hosting 0.1/hr, uptime
0.5/hr, LLM 1.5–2.0), off-peak 1.5× multiplier for LLM actions, SQLite
entries hash-chained + Ed25519-signed, tiers 1–3 with search costs
0.100/0.050/0.033, 72 h grace then 2× debt mode.
"""
from __future__ import annotations

import enum
import json
import time
from dataclasses import dataclass
from pathlib import Path

from ..db import SQLiteStore
from ..hashing import content_hash
from ..trust.keys import KeyPair


class Action(enum.Enum):
    CRAWL = "crawl"
    QUERY_SERVED = "query_served"
    HOSTING = "hosting"
    UPTIME = "uptime"
    LLM_SUMMARIZE = "llm_summarize"
    LLM_RERANK = "llm_rerank"
    SEARCH_SPEND = "search_spend"


ACTION_WEIGHTS: dict[Action, float] = {
    Action.CRAWL: 1.0,            # per page
    Action.QUERY_SERVED: 0.5,     # per served query
    Action.HOSTING: 0.1,          # per hour
    Action.UPTIME: 0.5,           # per hour
    Action.LLM_SUMMARIZE: 2.0,    # per summary
    Action.LLM_RERANK: 1.5,       # per rerank batch
    Action.SEARCH_SPEND: -1.0,    # quantity = cost
}

LLM_ACTIONS = frozenset({Action.LLM_SUMMARIZE, Action.LLM_RERANK})
OFF_PEAK_MULTIPLIER = 1.5

TIER_THRESHOLDS = (0.0, 100.0, 1000.0)          # tier 1 / 2 / 3 balances
TIER_SEARCH_COST = (0.100, 0.050, 0.033)
GRACE_HOURS = 72.0
DEBT_MULTIPLIER = 2.0


@dataclass
class CreditEntry:
    action: str
    quantity: float
    multiplier: float
    credits: float
    ts: float
    entry_hash: str
    prev_hash: str
    signature: str = ""


class CreditLedger(SQLiteStore):
    SCHEMA = """
    CREATE TABLE IF NOT EXISTS credit_entries (
        id INTEGER PRIMARY KEY,
        action TEXT NOT NULL,
        quantity REAL NOT NULL,
        multiplier REAL NOT NULL,
        credits REAL NOT NULL,
        ts REAL NOT NULL,
        entry_hash TEXT NOT NULL,
        prev_hash TEXT NOT NULL,
        signature TEXT NOT NULL DEFAULT ''
    );
    """

    def __init__(self, path: str | Path = ":memory:",
                 kp: KeyPair | None = None,
                 off_peak_fn=None,
                 crawl_reward: float | None = None,
                 query_reward: float | None = None,
                 search_cost: float | None = None,
                 grace_hours: float | None = None):
        super().__init__(path)
        self.kp = kp
        self._off_peak_fn = off_peak_fn
        self._debt_since: float | None = None
        # config overrides (credits.* section); None = module defaults
        self._weights = dict(ACTION_WEIGHTS)
        if crawl_reward is not None:
            self._weights[Action.CRAWL] = float(crawl_reward)
        if query_reward is not None:
            self._weights[Action.QUERY_SERVED] = float(query_reward)
        self._base_search_cost = search_cost
        self._grace_hours = float(grace_hours) if grace_hours is not None \
            else GRACE_HOURS
        # batched accounting (record_action_async)
        import threading
        self._accum: dict[Action, float] = {}
        self._accum_searches = 0
        self._accum_lock = threading.Lock()
        self._flusher = None
        self._flush_stop = None
        self.flush_interval_s = 0.5

    def close(self) -> None:
        if self._flush_stop is not None:
            self._flush_stop.set()
        try:
            self.flush_pending()
        except Exception:
            pass
        super().close()

    # ------------------------------------------------------------ record
    def _last_hash(self) -> str:
        row = self.execute(
            "SELECT entry_hash FROM credit_entries ORDER BY id DESC LIMIT 1"
        ).fetchone()
        return row["entry_hash"] if row else "genesis"

    def record_action(self, action: Action, quantity: float = 1.0,
                      ts: float | None = None) -> CreditEntry:
        ts = float(ts if ts is not None else time.time())
        quantity = float(quantity)
        mult = 1.0
        if action in LLM_ACTIONS and self._is_off_peak(ts):
            mult = OFF_PEAK_MULTIPLIER
        credits = self._weights[action] * quantity * mult
        prev = self._last_hash()
        payload = json.dumps(
            {"action": action.value, "q": quantity, "m": mult,
             "c": credits, "ts": ts, "prev": prev}, sort_keys=True)
        entry_hash = content_hash(payload)
        sig = self.kp.sign(payload.encode()).hex() if self.kp else ""
        self.execute(
            "INSERT INTO credit_entries (action, quantity, multiplier,"
            " credits, ts, entry_hash, prev_hash, signature)"
            " VALUES (?,?,?,?,?,?,?,?)",
            (action.value, quantity, mult, credits, ts, entry_hash, prev,
             sig))
        self.commit()
        return CreditEntry(action.value, quantity, mult, credits, ts,
                           entry_hash, prev, sig)

    def _is_off_peak(self, ts: float) -> bool:
        if self._off_peak_fn is not None:
            return bool(self._off_peak_fn(ts))
        from .scheduling import is_off_peak
        return is_off_peak(ts)

    # ----------------------------------------------------------- balance
    def balance(self) -> float:
        """Durable entries plus accrued-but-unflushed async credits —
        callers of record_action_async see their contribution
        immediately even though the signed entry lands at the next
        flush interval."""
        row = self.execute(
            "SELECT COALESCE(SUM(credits), 0) AS b FROM credit_entries"
        ).fetchone()
        pending = 0.0
        with self._accum_lock:
            for action, qty in self._accum.items():
                pending += self._weights[action] * qty
            n_search = self._accum_searches
        if n_search:
            pending += self._weights[Action.SEARCH_SPEND] \
                * self._base_cost_now() * n_search
        return float(row["b"]) + pending

    def _base_cost_now(self) -> float:
        """Current per-search cost WITHOUT re-entering balance()."""
        if self._base_search_cost is not None:
            return self._base_search_cost
        return TIER_SEARCH_COST[0]

    def tier(self) -> int:
        b = self.balance()
        t = 1
        for i, thr in enumerate(TIER_THRESHOLDS, start=1):
            if b >= thr:
                t = i
        return t

    def search_cost(self) -> float:
        if self._base_search_cost is not None:
            # config override scales the tier-1 rate; tier discounts
            # keep their published ratios (ledger.py:12-15)
            cost = (self._base_search_cost
                    * TIER_SEARCH_COST[self.tier() - 1]
                    / TIER_SEARCH_COST[0])
        else:
            cost = TIER_SEARCH_COST[self.tier() - 1]
        if self.in_debt_mode():
            cost *= DEBT_MULTIPLIER
        return cost

    def in_debt_mode(self, now: float | None = None) -> bool:
        """Negative balance beyond the 72 h grace window ⇒ 2× costs."""
        now = now or time.time()
        if self.balance() >= 0:
            self._debt_since = None
            return False
        if self._debt_since is None:
            self._debt_since = now
            return False
        return (now - self._debt_since) > self._grace_hours * 3600.0

    def deduct_search_cost(self) -> float:
        cost = self.search_cost()
        self.record_action(Action.SEARCH_SPEND, cost)
        return cost

    # --------------------------------------------- batched accounting
    def record_action_async(self, action: Action,
                            quantity: float = 1.0) -> None:
        """Coalesce high-rate accounting into periodic signed entries:
        ONE hash-chained + Ed25519-signed entry per action type per
        flush interval, with the quantities summed. Pure-python signing
        costs ~3 ms under the GIL — per-request record_action() would
        cap the whole serving node at ~130 QPS. Credit totals are
        identical (C = W·Q·M is linear in Q)."""
        with self._accum_lock:
            self._accum[action] = self._accum.get(action, 0.0) + quantity
            self._ensure_flusher()

    def deduct_search_cost_async(self) -> None:
        """Batched deduct: the per-search cost is evaluated at flush
        time (tier drift within one interval is negligible)."""
        with self._accum_lock:
            self._accum_searches += 1
            self._ensure_flusher()

    def _ensure_flusher(self) -> None:
        import threading
        if self._flusher is None or not self._flusher.is_alive():
            self._flush_stop = threading.Event()
            self._flusher = threading.Thread(
                target=self._flush_loop, name="infomesh-ledger-flush",
                daemon=True)
            self._flusher.start()

    def _flush_loop(self) -> None:
        while not self._flush_stop.wait(self.flush_interval_s):
            try:
                self.flush_pending()
            except Exception:
                return

    def flush_pending(self) -> int:
        """Write one signed entry per accumulated action type."""
        with self._accum_lock:
            accum, self._accum = self._accum, {}
            n_search, self._accum_searches = self._accum_searches, 0
        n = 0
        for action, qty in accum.items():
            self.record_action(action, qty)
            n += 1
        if n_search:
            self.record_action(Action.SEARCH_SPEND,
                               self.search_cost() * n_search)
            n += 1
        return n

    # ------------------------------------------------------ verification
    def verify_chain(self) -> bool:
        """Recompute the hash chain (and signatures when present)."""
        prev = "genesis"
        for row in self.execute(
                "SELECT * FROM credit_entries ORDER BY id").fetchall():
            payload = json.dumps(
                {"action": row["action"], "q": row["quantity"],
                 "m": row["multiplier"], "c": row["credits"],
                 "ts": row["ts"], "prev": prev}, sort_keys=True)
            if content_hash(payload) != row["entry_hash"]:
                return False
            if row["signature"] and self.kp is not None:
                if not KeyPair.verify(self.kp.public, payload.encode(),
                                      bytes.fromhex(row["signature"])):
                    return False
            prev = row["entry_hash"]
        return True

    def entries(self, limit: int = 100) -> list[dict]:
        return [dict(r) for r in self.execute(
            "SELECT * FROM credit_entries ORDER BY id DESC LIMIT ?",
            (limit,)).fetchall()]

    def stats(self) -> dict:
        by_action = {r["action"]: r["total"] for r in self.execute(
            "SELECT action, SUM(credits) AS total FROM credit_entries"
            " GROUP BY action")}
        return {"balance": self.balance(), "tier": self.tier(),
                "search_cost": self.search_cost(), "by_action": by_action,
                "debt_mode": self.in_debt_mode()}
