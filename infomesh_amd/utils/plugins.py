"""Plugin hook system (reference parity: infomesh/plugins.py — PRE/POST
hooks for crawl/index/search/rank + custom tokenizer/scorer slots)."""
from __future__ import annotations

import logging
from collections import defaultdict
from typing import Any, Callable

log = logging.getLogger("infomesh.plugins")

HOOK_POINTS = (
    "pre_crawl", "post_crawl",
    "pre_index", "post_index",
    "pre_search", "post_search",
    "pre_rank", "post_rank",
    "tokenizer", "scorer",
)


class PluginManager:
    def __init__(self):
        self._hooks: dict[str, list[Callable]] = defaultdict(list)

    def register(self, point: str, fn: Callable) -> None:
        if point not in HOOK_POINTS:
            raise ValueError(f"unknown hook point {point!r}; "
                             f"valid: {HOOK_POINTS}")
        self._hooks[point].append(fn)

    def hook(self, point: str):
        """Decorator form: @plugins.hook('pre_search')."""
        def deco(fn):
            self.register(point, fn)
            return fn
        return deco

    def run(self, point: str, value: Any, **kw) -> Any:
        """Chain hooks; each may transform and return the value (or
        return None to keep it). Exceptions are isolated."""
        for fn in self._hooks.get(point, []):
            try:
                out = fn(value, **kw)
                if out is not None:
                    value = out
            except Exception as e:
                log.warning("plugin %s at %s failed: %s",
                            getattr(fn, "__name__", fn), point, e)
        return value

    def get_single(self, point: str) -> Callable | None:
        """For replacement slots (tokenizer/scorer): last registration
        wins."""
        hooks = self._hooks.get(point, [])
        return hooks[-1] if hooks else None

    def count(self) -> int:
        return sum(len(v) for v in self._hooks.values())


GLOBAL_PLUGINS = PluginManager()
