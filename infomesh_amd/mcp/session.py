"""MCP session store + per-tool analytics.

Reference parity: infomesh/mcp/session.py (session store, analytics
tracker, webhook registry).
"""
from __future__ import annotations

import secrets
import time
from collections import defaultdict
from dataclasses import dataclass, field


@dataclass
class Session:
    session_id: str
    created_at: float = field(default_factory=time.time)
    last_seen: float = field(default_factory=time.time)
    calls: int = 0


class SessionStore:
    def __init__(self, ttl_s: float = 3600.0, max_sessions: int = 1000):
        import threading
        self._lock = threading.Lock()
        self.ttl_s = ttl_s
        self.max_sessions = max_sessions
        self._sessions: dict[str, Session] = {}

    def create(self) -> Session:
        # handlers run on a 256-thread pool: every read-modify of the
        # session dict must hold the lock (iteration during _gc raced
        # with concurrent inserts)
        with self._lock:
            self._gc()
            s = Session(session_id=secrets.token_hex(16))
            self._sessions[s.session_id] = s
            return s

    def get(self, session_id: str) -> Session | None:
        with self._lock:
            s = self._sessions.get(session_id)
            if s is None:
                return None
            if time.time() - s.last_seen > self.ttl_s:
                del self._sessions[session_id]
                return None
            s.last_seen = time.time()
            return s

    def touch(self, session_id: str) -> Session:
        with self._lock:
            s = self._sessions.get(session_id)
            now = time.time()
            if s is None or now - s.last_seen > self.ttl_s:
                s = Session(session_id=session_id)
                self._sessions[session_id] = s
            s.last_seen = now
            s.calls += 1
            return s

    def _gc(self) -> None:
        now = time.time()
        dead = [k for k, s in self._sessions.items()
                if now - s.last_seen > self.ttl_s]
        for k in dead:
            del self._sessions[k]
        while len(self._sessions) >= self.max_sessions:
            oldest = min(self._sessions, key=lambda k: self._sessions[k].last_seen)
            del self._sessions[oldest]

    def count(self) -> int:
        return len(self._sessions)


class AnalyticsTracker:
    """Per-tool call counts + latency percentiles."""

    def __init__(self, max_samples: int = 1000):
        import threading
        self.max_samples = max_samples
        self._lock = threading.Lock()   # recorded from the handler pool
        self._latency: dict[str, list[float]] = defaultdict(list)
        self._errors: dict[str, int] = defaultdict(int)

    def record(self, tool: str, elapsed_ms: float, error: bool = False) -> None:
        with self._lock:
            lst = self._latency[tool]
            lst.append(elapsed_ms)
            if len(lst) > self.max_samples:
                del lst[: len(lst) // 2]
            if error:
                self._errors[tool] += 1

    def report(self) -> dict:
        out = {}
        with self._lock:
            snap = {t: list(v) for t, v in self._latency.items()}
        for tool, lst in snap.items():
            s = sorted(lst)
            n = len(s)
            out[tool] = {
                "calls": n,
                "errors": self._errors.get(tool, 0),
                "avg_ms": round(sum(s) / n, 2),
                "p50_ms": round(s[n // 2], 2),
                "p95_ms": round(s[min(n - 1, int(n * 0.95))], 2),
                "p99_ms": round(s[min(n - 1, int(n * 0.99))], 2),
            }
        return out


class WebhookRegistry:
    """Registered webhooks notified on index events (in-process)."""

    def __init__(self):
        import threading
        self._lock = threading.Lock()
        self._hooks: dict[str, list] = defaultdict(list)

    def register(self, event: str, callback) -> None:
        with self._lock:
            self._hooks[event].append(callback)

    def fire(self, event: str, payload: dict) -> int:
        n = 0
        with self._lock:   # snapshot: register() may run concurrently
            hooks = list(self._hooks.get(event, []))
        for cb in hooks:
            try:
                cb(payload)
                n += 1
            except Exception:
                pass
        return n
