"""Starter-index bootstrap.

Reference parity: infomesh/index/starter.py (community-snapshot
download + needs_starter threshold). The target image has no network,
so discovery is local-first: a configured directory of
`.infomesh-snapshot` files (e.g. shipped with a deployment) with the
HTTP fetch available when an URL and connectivity exist.
"""
from __future__ import annotations

import logging
from pathlib import Path
from typing import Callable

from .local_store import LocalStore
from .snapshot import SUFFIX, import_snapshot, read_snapshot_header

log = logging.getLogger("infomesh.starter")

NEEDS_STARTER_BELOW = 100   # docs


def needs_starter(store: LocalStore) -> bool:
    return store.count() < NEEDS_STARTER_BELOW


def find_local_starters(search_dirs: list[Path]) -> list[Path]:
    out: list[Path] = []
    for d in search_dirs:
        if d.is_dir():
            out.extend(sorted(d.glob(f"*{SUFFIX}")))
    return out


def load_starter(store: LocalStore, search_dirs: list[Path],
                 on_document=None,
                 progress: Callable[[int, int], None] | None = None
                 ) -> dict | None:
    """Import the largest available local starter snapshot."""
    candidates = find_local_starters(search_dirs)
    if not candidates:
        return None
    best = max(candidates, key=lambda p: p.stat().st_size)
    header = read_snapshot_header(best)
    log.info("importing starter %s (%s docs)", best, header.get("doc_count"))
    res = import_snapshot(store, best, on_document=on_document,
                          progress=progress)
    res["path"] = str(best)
    return res


def download_starter(url: str, dest: Path, timeout: float = 120.0) -> Path:
    """Fetch a snapshot over HTTP when connectivity exists (offline
    deployments use find_local_starters instead)."""
    import httpx
    dest.parent.mkdir(parents=True, exist_ok=True)
    tmp = dest.with_suffix(".part")
    with httpx.stream("GET", url, timeout=timeout,
                      follow_redirects=True) as resp:
        resp.raise_for_status()
        with open(tmp, "wb") as f:
            for chunk in resp.iter_bytes():
                f.write(chunk)
    tmp.replace(dest)
    return dest
