"""LRU + TTL query cache.

Reference parity: infomesh/search/cache.py (QueryCache, default 1000
entries / 300 s TTL, keyed by query+filters; wired at mcp/server.py:120-126).
"""
from __future__ import annotations

import json
import threading
import time
from collections import OrderedDict
from typing import Any


class QueryCache:
    def __init__(self, max_entries: int = 1000, ttl_s: float = 300.0):
        self.max_entries = int(max_entries)
        self.ttl_s = float(ttl_s)
        self._data: OrderedDict[str, tuple[float, Any]] = OrderedDict()
        self._lock = threading.Lock()
        self.hits = 0
        self.misses = 0

    @staticmethod
    def make_key(query: str, **filters: Any) -> str:
        return json.dumps({"q": query, **{k: v for k, v in sorted(filters.items())
                                          if v is not None}}, sort_keys=True)

    def get(self, key: str) -> Any | None:
        if self.max_entries <= 0:
            return None
        with self._lock:
            item = self._data.get(key)
            if item is None:
                self.misses += 1
                return None
            ts, value = item
            if time.time() - ts > self.ttl_s:
                del self._data[key]
                self.misses += 1
                return None
            self._data.move_to_end(key)
            self.hits += 1
            return value

    def put(self, key: str, value: Any) -> None:
        if self.max_entries <= 0:
            return
        with self._lock:
            self._data[key] = (time.time(), value)
            self._data.move_to_end(key)
            while len(self._data) > self.max_entries:
                self._data.popitem(last=False)

    def invalidate(self) -> None:
        with self._lock:
            self._data.clear()

    def stats(self) -> dict[str, Any]:
        with self._lock:
            return {"entries": len(self._data), "hits": self.hits,
                    "misses": self.misses, "ttl_s": self.ttl_s,
                    "max_entries": self.max_entries}
