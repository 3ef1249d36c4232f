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


MAX_STARTER_BYTES = 1 << 30   # snapshot import guard (snapshot.py)


def download_starter(url: str, dest: Path, timeout: float = 120.0,
                     progress: Callable[[int, int], None] | None = None,
                     client=None) -> Path:
    """Fetch a snapshot over HTTP when connectivity exists (offline
    deployments use find_local_starters instead). Streams with a size
    guard and an optional progress(done, total) callback."""
    import httpx
    dest.parent.mkdir(parents=True, exist_ok=True)
    tmp = dest.with_suffix(".part")
    own = client is None
    client = client or httpx.Client(follow_redirects=True, timeout=timeout)
    try:
        with client.stream("GET", url) as resp:
            resp.raise_for_status()
            total = int(resp.headers.get("content-length", 0))
            if total > MAX_STARTER_BYTES:
                raise ValueError(f"starter too large ({total} bytes)")
            done = 0
            with open(tmp, "wb") as f:
                for chunk in resp.iter_bytes():
                    done += len(chunk)
                    if done > MAX_STARTER_BYTES:
                        raise ValueError("starter exceeded size guard")
                    f.write(chunk)
                    if progress:
                        progress(done, total)
    finally:
        if own:
            client.close()
    tmp.replace(dest)
    return dest


def fetch_release_starter(store: LocalStore, dest_dir: Path,
                          repo: str = "dotnetpower/infomesh",
                          progress: Callable[[int, int], None] | None = None,
                          client=None) -> dict | None:
    """Community-release discovery + download + import (reference flow:
    infomesh/index/starter.py:76-192): query the GitHub releases API,
    pick the newest `.infomesh-snapshot` asset, stream it down with the
    size guard, then import into the LocalStore. `client` is injectable
    for offline tests."""
    import httpx
    own = client is None
    client = client or httpx.Client(follow_redirects=True, timeout=30.0)
    try:
        r = client.get(
            f"https://api.github.com/repos/{repo}/releases/latest",
            headers={"Accept": "application/vnd.github+json"})
        if r.status_code != 200:
            log.info("no starter release available (%s)", r.status_code)
            return None
        assets = r.json().get("assets", [])
        asset = next((a for a in assets
                      if a.get("name", "").endswith(SUFFIX)
                      and a.get("size", 0) <= MAX_STARTER_BYTES), None)
        if asset is None:
            return None
        dest = dest_dir / asset["name"]
        download_starter(asset["browser_download_url"], dest,
                         progress=progress, client=client)
        res = import_snapshot(store, dest)
        res["path"] = str(dest)
        res["release_asset"] = asset["name"]
        return res
    finally:
        if own:
            client.close()
