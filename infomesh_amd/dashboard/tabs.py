"""Tabbed terminal dashboard: six tabs over the node's durable state.

Reference parity: infomesh/dashboard/app.py:168 + screens/* — the
reference's Textual TUI has tabs Overview / Crawl / Search / Network /
Credits / Settings. Here the same six surfaces render with rich.Live
plus raw-terminal key handling (1-6/←→ switch, q quits); "Network"
becomes "Shards" (the intra-node GPU fabric replaces the P2P mesh).
All data comes from the heartbeat file + SQLite under WAL, so the
dashboard runs safely in a separate process from the daemon
(reference behavior).
"""
from __future__ import annotations

import time

from rich.layout import Layout
from rich.panel import Panel
from rich.table import Table
from rich.text import Text

from .app import DashboardData, History, _kv_table, _load_and_rss, sparkline

TABS = ("overview", "crawl", "search", "shards", "credits", "settings")


class TabbedData(DashboardData):
    """Extra per-tab queries on top of the overview snapshot."""

    def recent_documents(self, limit: int = 12) -> list[tuple]:
        import sqlite3
        path = self.data_dir / "index.db"
        if not path.exists():
            return []
        try:
            conn = sqlite3.connect(f"file:{path}?mode=ro", uri=True,
                                   timeout=1.0)
            try:
                return conn.execute(
                    "SELECT url, title, crawled_at FROM documents"
                    " ORDER BY crawled_at DESC LIMIT ?",
                    (limit,)).fetchall()
            finally:
                conn.close()
        except sqlite3.Error:
            return []

    def top_domains(self, limit: int = 10) -> list[tuple]:
        import sqlite3
        path = self.data_dir / "index.db"
        if not path.exists():
            return []
        try:
            conn = sqlite3.connect(f"file:{path}?mode=ro", uri=True,
                                   timeout=1.0)
            try:
                return conn.execute(
                    "SELECT domain, COUNT(*) c FROM documents"
                    " GROUP BY domain ORDER BY c DESC LIMIT ?",
                    (limit,)).fetchall()
            finally:
                conn.close()
        except sqlite3.Error:
            return []

    def language_counts(self) -> list[tuple]:
        import sqlite3
        path = self.data_dir / "index.db"
        if not path.exists():
            return []
        try:
            conn = sqlite3.connect(f"file:{path}?mode=ro", uri=True,
                                   timeout=1.0)
            try:
                return conn.execute(
                    "SELECT language, COUNT(*) FROM documents"
                    " GROUP BY language ORDER BY 2 DESC LIMIT 8"
                ).fetchall()
            finally:
                conn.close()
        except sqlite3.Error:
            return []

    def ledger_tail(self, limit: int = 12) -> list[tuple]:
        import sqlite3
        path = self.data_dir / "ledger.db"
        if not path.exists():
            return []
        try:
            conn = sqlite3.connect(f"file:{path}?mode=ro", uri=True,
                                   timeout=1.0)
            try:
                return conn.execute(
                    "SELECT action, quantity, credits, ts FROM"
                    " credit_entries ORDER BY id DESC LIMIT ?",
                    (limit,)).fetchall()
            finally:
                conn.close()
        except sqlite3.Error:
            return []

    def shard_manifests(self) -> list[dict]:
        from ..index.manifest import manifest_info
        out = []
        for p in sorted(self.data_dir.glob("shard*.pt")):
            info = manifest_info(p)
            if info:
                out.append(info)
        return out


# ---------------------------------------------------------- tab renders

def render_overview(data: TabbedData, hist: History) -> Panel:
    s = data.snapshot()
    rt = s["runtime"]
    hist.push("docs", s["docs"])
    load, rss = _load_and_rss()
    hist.push("load", load)
    rows = [
        ("state", rt.get("state", "stopped")),
        ("pid", rt.get("pid", "—")),
        ("documents", s["docs"]),
        ("domains", s["domains"]),
        ("indexed last hour", s["recent_docs"]),
        ("credit balance", f"{s['balance']:.2f}"),
        ("cpu load", f"{load:.2f}  {sparkline(hist.get('load'))}"),
        ("rss", f"{rss:.2f} GB"),
        ("docs trend", sparkline(hist.get("docs"))),
    ]
    return Panel(_kv_table(rows), title="overview")


def render_crawl(data: TabbedData, hist: History) -> Panel:
    t = Table(title=None, expand=True)
    t.add_column("crawled", style="dim", width=9)
    t.add_column("url")
    for url, title, ts in data.recent_documents():
        t.add_row(time.strftime("%H:%M:%S", time.localtime(ts or 0)),
                  url[:80])
    s = data.snapshot()
    grid = Table.grid()
    grid.add_row(_kv_table([("seen URLs", s["seen_urls"]),
                            ("link edges", s["link_edges"])]))
    grid.add_row(t)
    return Panel(grid, title="crawl")


def render_search(data: TabbedData, hist: History) -> Panel:
    t = Table(expand=True)
    t.add_column("domain")
    t.add_column("docs", justify="right")
    for dom, c in data.top_domains():
        t.add_row(dom or "—", str(c))
    lt = Table(expand=True)
    lt.add_column("language")
    lt.add_column("docs", justify="right")
    for lang, c in data.language_counts():
        lt.add_row(lang or "—", str(c))
    grid = Table.grid(expand=True)
    grid.add_column()
    grid.add_column()
    grid.add_row(t, lt)
    return Panel(grid, title="search / index")


def render_shards(data: TabbedData, hist: History) -> Panel:
    manifests = data.shard_manifests()
    if not manifests:
        body = Text("no shard manifests under the data dir\n"
                    "(GPU shards are built in the serving process; "
                    "warm-start manifests appear here)", style="dim")
        return Panel(body, title="shards (GPU fabric)")
    t = Table(expand=True)
    for col in ("rank", "world", "docs", "segments", "bytes", "created"):
        t.add_column(col)
    for m in manifests:
        t.add_row(str(m.get("rank")), str(m.get("world")),
                  str(m.get("n_docs")), str(m.get("n_segments", 1)),
                  f"{m.get('bytes', 0)/1e6:.1f} MB",
                  time.strftime("%m-%d %H:%M",
                                time.localtime(m.get("created_at", 0))))
    return Panel(t, title="shards (GPU fabric)")


def render_credits(data: TabbedData, hist: History) -> Panel:
    s = data.snapshot()
    t = Table(expand=True)
    for col in ("action", "qty", "credits", "at"):
        t.add_column(col)
    for action, qty, credits, ts in data.ledger_tail():
        t.add_row(action, f"{qty:g}", f"{credits:+.3f}",
                  time.strftime("%H:%M:%S", time.localtime(ts)))
    grid = Table.grid()
    grid.add_row(_kv_table([("balance", f"{s['balance']:.2f}"),
                            ("entries", s["ledger_entries"])]))
    grid.add_row(t)
    return Panel(grid, title="credits")


def render_settings(data: TabbedData, hist: History) -> Panel:
    cfg = data.cfg
    rows = [
        ("data dir", str(data.data_dir)),
        ("role", cfg.node.role),
        ("fts tokenizer", cfg.index.fts_tokenizer),
        ("max results", cfg.search.max_results),
        ("batcher", f"max {cfg.search.batch_max} / "
                    f"{cfg.search.batch_wait_ms} ms"),
        ("crawl concurrency", cfg.crawl.max_concurrent),
        ("politeness", f"{cfg.crawl.politeness_delay_s}s"),
        ("credits enabled", cfg.credits.enabled),
    ]
    return Panel(_kv_table(rows), title="settings")


RENDERERS = {
    "overview": render_overview,
    "crawl": render_crawl,
    "search": render_search,
    "shards": render_shards,
    "credits": render_credits,
    "settings": render_settings,
}


def render_tabbed(data: TabbedData, hist: History, active: str) -> Layout:
    header = Text()
    for name in TABS:
        style = "bold reverse" if name == active else "dim"
        header.append(f" {TABS.index(name) + 1}:{name} ", style=style)
    header.append("   (1-6 or ←→ switch, q quits)", style="dim")
    layout = Layout()
    layout.split_column(
        Layout(Panel(header), size=3),
        Layout(RENDERERS[active](data, hist), name="body"),
    )
    return layout


class _RawKeys:
    """Non-blocking single-key reads; inert when stdin is not a TTY."""

    def __init__(self):
        import sys
        self.tty = sys.stdin.isatty()
        self._saved = None

    def __enter__(self):
        if self.tty:
            import sys
            import termios
            import tty
            self._saved = termios.tcgetattr(sys.stdin.fileno())
            tty.setcbreak(sys.stdin.fileno())
        return self

    def __exit__(self, *exc):
        if self._saved is not None:
            import sys
            import termios
            termios.tcsetattr(sys.stdin.fileno(), termios.TCSADRAIN,
                              self._saved)
        return False

    def poll(self) -> str:
        if not self.tty:
            return ""
        import select
        import sys
        r, _, _ = select.select([sys.stdin], [], [], 0)
        return sys.stdin.read(1) if r else ""


def next_tab(active: str, key: str) -> str | None:
    """Tab-switch state machine; None means quit."""
    if key in ("q", "Q", "\x03"):
        return None
    if key.isdigit() and 1 <= int(key) <= len(TABS):
        return TABS[int(key) - 1]
    i = TABS.index(active)
    if key in ("l", "\t", "C"):     # right / tab / arrow-right tail
        return TABS[(i + 1) % len(TABS)]
    if key in ("h", "D"):           # left / arrow-left tail
        return TABS[(i - 1) % len(TABS)]
    return active


def run_tabbed_dashboard(refresh_s: float = 1.0,
                         iterations: int | None = None) -> None:
    from rich.live import Live
    data = TabbedData()
    hist = History()
    active = "overview"
    with _RawKeys() as keys, \
            Live(render_tabbed(data, hist, active), refresh_per_second=8,
                 screen=iterations is None) as live:
        n = 0
        while iterations is None or n < iterations:
            deadline = time.time() + (refresh_s if iterations is None
                                      else 0.01)
            while time.time() < deadline:
                k = keys.poll()
                if k:
                    nxt = next_tab(active, k)
                    if nxt is None:
                        return
                    if nxt != active:
                        active = nxt
                        break
                time.sleep(0.03)
            live.update(render_tabbed(data, hist, active))
            n += 1
