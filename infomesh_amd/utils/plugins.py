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


def load_plugin_module(module_path: str,
                       manager: PluginManager | None = None) -> int:
    """Import a plugin module by dotted path (reference dx.py:84-115).

    The module may either call ``register(point, fn)`` on the GLOBAL
    manager at import time, or expose a ``setup(plugins)`` function
    that receives the manager. Returns hooks added."""
    import importlib
    mgr = manager or GLOBAL_PLUGINS
    before = mgr.count()
    mod = importlib.import_module(module_path)
    setup = getattr(mod, "setup", None)
    if callable(setup):
        setup(mgr)
    return mgr.count() - before


def load_plugins_from_config(paths: list[str],
                             manager: PluginManager | None = None) -> int:
    """Best-effort bulk load; a broken plugin never takes the node down."""
    total = 0
    for p in paths:
        try:
            total += load_plugin_module(p, manager)
        except Exception as e:  # noqa: BLE001
            log.warning("plugin module %s failed to load: %s", p, e)
    return total
