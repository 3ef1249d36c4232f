"""Dynamic micro-batching between serving entry points and the GPU
query plane.

Round-1 gap: the flagship QPS exists only at batch 128
inside bench.py while every MCP/HTTP request ran B=1 — the headline
number was unreachable from the real entry points. The batcher collects
concurrent requests for up to `max_wait_ms` (or until `max_batch`) and
submits ONE collective `engine.search_many`, then scatters the
per-query results back to the waiting callers.

All engine/GPU work happens on the single batcher thread — the engine
is not thread-safe and a single CUDA-stream owner is the right model
anyway. Callers block on an event (thread-per-request servers) — the
MCP HTTP transport dispatches handler calls to a thread pool so
concurrent requests actually overlap here.

Reference flow analogue: infomesh/mcp/handlers.py:382-541 funnels every
entry point through one search path; here that path is the batcher.
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Any, Callable

__all__ = ["QueryBatcher"]


@dataclass
class _Pending:
    query: str
    limit: int
    event: threading.Event = field(default_factory=threading.Event)
    result: Any = None
    error: BaseException | None = None


class QueryBatcher:
    """Micro-batching queue in front of a batch-capable search engine.

    execute(queries, limit) -> list of per-query results; defaults to
    `engine.search_many`."""

    def __init__(self, engine: Any = None, max_batch: int = 128,
                 max_wait_ms: float = 1.5,
                 execute: Callable[[list[str], int], list] | None = None,
                 timeout_s: float = 30.0):
        assert engine is not None or execute is not None
        self.engine = engine
        self.max_batch = max(1, int(max_batch))
        self.max_wait_s = max(0.0, float(max_wait_ms)) / 1e3
        self.timeout_s = timeout_s
        # read self.engine at call time so callers may rebind .engine
        # (AppContext.engine is a mutable field)
        self._execute = execute or (
            lambda qs, limit: self.engine.search_many(qs, limit=limit))
        self._queue: list[_Pending] = []
        self._cond = threading.Condition()
        self._thread: threading.Thread | None = None
        self._closed = False
        # stats (reference routing.py:58-77 spirit)
        self.n_queries = 0
        self.n_batches = 0
        self.max_batch_seen = 0

    # ------------------------------------------------------------ API
    def submit(self, query: str, limit: int = 10):
        """Block until the containing batch executes; returns this
        query's results (or re-raises the batch's error)."""
        p = _Pending(query=query, limit=limit)
        with self._cond:
            if self._closed:
                raise RuntimeError("batcher closed")
            self._ensure_thread()
            self._queue.append(p)
            self._cond.notify_all()
        if not p.event.wait(self.timeout_s):
            raise TimeoutError(f"batched search timed out ({self.timeout_s}s)")
        if p.error is not None:
            raise p.error
        return p.result

    def stats(self) -> dict:
        return {
            "queries": self.n_queries,
            "batches": self.n_batches,
            "avg_batch": round(self.n_queries / self.n_batches, 2)
            if self.n_batches else 0.0,
            "max_batch_seen": self.max_batch_seen,
            "queued": len(self._queue),
        }

    def close(self) -> None:
        with self._cond:
            self._closed = True
            self._cond.notify_all()
        t = self._thread
        if t is not None and t.is_alive():
            t.join(timeout=5)

    # ------------------------------------------------------- internals
    def _ensure_thread(self) -> None:
        if self._thread is None or not self._thread.is_alive():
            self._thread = threading.Thread(
                target=self._loop, name="infomesh-query-batcher",
                daemon=True)
            self._thread.start()

    def _take_batch(self) -> list[_Pending] | None:
        """Wait for work, then give stragglers max_wait to pile on."""
        with self._cond:
            while not self._queue and not self._closed:
                self._cond.wait(0.25)
            if self._closed and not self._queue:
                return None
            deadline = time.perf_counter() + self.max_wait_s
            while (len(self._queue) < self.max_batch
                   and not self._closed):
                remaining = deadline - time.perf_counter()
                if remaining <= 0:
                    break
                self._cond.wait(remaining)
            batch = self._queue[: self.max_batch]
            del self._queue[: len(batch)]
            return batch

    def _loop(self) -> None:
        while True:
            batch = self._take_batch()
            if batch is None:
                return
            queries = [p.query for p in batch]
            limit = max(p.limit for p in batch)
            try:
                results = self._execute(queries, limit)
                assert len(results) == len(batch), \
                    "execute() must return one result per query"
                for p, r in zip(batch, results):
                    p.result = (r[: p.limit]
                                if isinstance(r, list) else r)
                    p.event.set()
            except BaseException as e:  # propagate to every caller
                for p in batch:
                    p.error = e
                    p.event.set()
            self.n_queries += len(batch)
            self.n_batches += 1
            self.max_batch_seen = max(self.max_batch_seen, len(batch))
