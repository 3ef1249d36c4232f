"""Version checks (reference parity: infomesh/version_check.py — PyPI +
peer version checks; offline-first here: compares against a pinned file
and only reaches PyPI when connectivity exists)."""
from __future__ import annotations

import json
import re
import time
from pathlib import Path

from .. import __version__

CHECK_INTERVAL_S = 24 * 3600.0


def parse_version(v: str) -> tuple[int, ...]:
    return tuple(int(x) for x in re.findall(r"\d+", v)[:3] or [0])


def is_newer(candidate: str, current: str = __version__) -> bool:
    return parse_version(candidate) > parse_version(current)


def check_for_update(state_path: Path | None = None,
                     fetch: bool = False) -> dict:
    """Returns {"current", "latest", "update_available", "source"}.
    Offline default reads a pinned `latest_version` file if present."""
    out = {"current": __version__, "latest": __version__,
           "update_available": False, "source": "offline"}
    if state_path is not None and state_path.exists():
        try:
            state = json.loads(state_path.read_text())
            if time.time() - state.get("ts", 0) < CHECK_INTERVAL_S:
                return state["result"]
        except (json.JSONDecodeError, KeyError):
            pass
    if fetch:
        try:
            import httpx
            resp = httpx.get("https://pypi.org/pypi/infomesh-amd/json",
                             timeout=5.0)
            latest = resp.json()["info"]["version"]
            out.update(latest=latest, source="pypi",
                       update_available=is_newer(latest))
        except Exception:
            pass
    if state_path is not None:
        state_path.parent.mkdir(parents=True, exist_ok=True)
        state_path.write_text(json.dumps(
            {"ts": time.time(), "result": out}))
    return out
