"""Live terminal dashboard (rich.Live).

Reads the same node-state surfaces as the reference TUI
(dashboard/app.py:168 + screens/*): the runtime heartbeat file and the
SQLite stores under WAL (safe concurrent reads from a separate
process), refreshing in place. Panels mirror the reference's tabs:
Overview, Index, GPU engine, Crawl, Credits, Cache/Search.
"""
from __future__ import annotations

import time

from rich.layout import Layout
from rich.live import Live
from rich.panel import Panel
from rich.table import Table
from rich.text import Text

from ..config import Config, load_config
from ..runtime import RuntimeStatus


class DashboardData:
    """Read-only view over a node's durable state (no AppContext —
    works against a running daemon's files)."""

    def __init__(self, cfg: Config | None = None):
        self.cfg = cfg or load_config()
        self.data_dir = self.cfg.data_dir

    def runtime(self) -> dict:
        return RuntimeStatus(self.data_dir).read()

    def _sqlite_scalar(self, db: str, sql: str, default=0):
        import sqlite3
        path = self.data_dir / db
        if not path.exists():
            return default
        try:
            conn = sqlite3.connect(f"file:{path}?mode=ro", uri=True,
                                   timeout=1.0)
            try:
                row = conn.execute(sql).fetchone()
                return row[0] if row else default
            finally:
                conn.close()
        except sqlite3.Error:
            return default

    def snapshot(self) -> dict:
        return {
            "runtime": self.runtime(),
            "docs": self._sqlite_scalar(
                "index.db", "SELECT COUNT(*) FROM documents"),
            "domains": self._sqlite_scalar(
                "index.db", "SELECT COUNT(DISTINCT domain) FROM documents"),
            "recent_docs": self._sqlite_scalar(
                "index.db",
                "SELECT COUNT(*) FROM documents WHERE crawled_at > "
                f"{time.time() - 3600}"),
            "seen_urls": self._sqlite_scalar(
                "dedup.db", "SELECT COUNT(*) FROM seen_urls"),
            "balance": self._sqlite_scalar(
                "ledger.db", "SELECT COALESCE(SUM(credits),0)"
                             " FROM credit_entries", 0.0),
            "ledger_entries": self._sqlite_scalar(
                "ledger.db", "SELECT COUNT(*) FROM credit_entries"),
            "link_edges": self._sqlite_scalar(
                "links.db", "SELECT COUNT(*) FROM links"),
            "trust_subjects": self._sqlite_scalar(
                "trust.db", "SELECT COUNT(*) FROM trust"),
        }


SPARK_CHARS = " ▁▂▃▄▅▆▇█"


def sparkline(values: list[float], width: int = 24) -> str:
    """Unicode sparkline (reference dashboard widgets analogue)."""
    if not values:
        return ""
    vals = values[-width:]
    lo, hi = min(vals), max(vals)
    span = (hi - lo) or 1.0
    return "".join(SPARK_CHARS[1 + int((v - lo) / span * 7)]
                   for v in vals)


class History:
    """In-process sample history for sparklines."""

    def __init__(self, maxlen: int = 60):
        self.maxlen = maxlen
        self.series: dict[str, list[float]] = {}

    def push(self, name: str, value: float) -> None:
        xs = self.series.setdefault(name, [])
        xs.append(float(value))
        del xs[:-self.maxlen]

    def get(self, name: str) -> list[float]:
        return self.series.get(name, [])


def _load_and_rss() -> tuple[float, float]:
    try:
        load = __import__("os").getloadavg()[0]
    except OSError:
        load = 0.0
    rss = 0.0
    try:
        with open("/proc/self/status") as f:
            for line in f:
                if line.startswith("VmRSS"):
                    rss = float(line.split()[1]) / 1e6  # GB
                    break
    except OSError:
        pass
    return load, rss


def _kv_table(rows: list[tuple[str, str]]) -> Table:
    t = Table.grid(padding=(0, 2))
    t.add_column(style="dim")
    t.add_column()
    for k, v in rows:
        t.add_row(k, str(v))
    return t


def render_dashboard(data: DashboardData, hist: History | None = None):
    s = data.snapshot()
    rt = s["runtime"]
    if hist is not None:
        hist.push("docs", s["docs"])
        hist.push("recent", s["recent_docs"])
        load, rss = _load_and_rss()
        hist.push("load", load)
    state = rt.get("state", "stopped")
    color = "green" if state == "running" else "red"
    layout = Layout()
    layout.split_column(
        Layout(Panel(Text.assemble(
            ("infomesh-amd ", "bold"),
            (f"● {state}", color),
            (f"   pid {rt.get('pid', '—')}   "
             f"heartbeat {time.strftime('%H:%M:%S', time.localtime(rt.get('ts', 0)))}"
             if rt.get("ts") else "", "dim")),
            title="overview"), size=3),
        Layout(name="mid"),
        Layout(name="bottom"),
    )
    doc_rows = [
        ("documents", s["docs"]),
        ("domains", s["domains"]),
        ("indexed last hour", s["recent_docs"]),
        ("link edges", s["link_edges"]),
    ]
    if hist is not None and len(hist.get("docs")) > 1:
        doc_rows.append(("docs trend", sparkline(hist.get("docs"))))
        doc_rows.append(("crawl rate", sparkline(hist.get("recent"))))
    eng_rows = [
        ("engine docs", rt.get("engine_docs", "—")),
        ("seen URLs", s["seen_urls"]),
        ("trust subjects", s["trust_subjects"]),
    ]
    if hist is not None and hist.get("load"):
        eng_rows.append(("cpu load", f"{hist.get('load')[-1]:.2f}  "
                         + sparkline(hist.get("load"))))
    layout["mid"].split_row(
        Layout(Panel(_kv_table(doc_rows), title="index")),
        Layout(Panel(_kv_table(eng_rows), title="crawl / engine")),
    )
    layout["bottom"].split_row(
        Layout(Panel(_kv_table([
            ("balance", f"{s['balance']:.2f}"),
            ("ledger entries", s["ledger_entries"]),
        ]), title="credits")),
        Layout(Panel(_kv_table([
            ("data dir", str(data.data_dir)),
            ("role", data.cfg.node.role),
        ]), title="settings")),
    )
    return layout


def run_dashboard(refresh_s: float = 2.0,
                  iterations: int | None = None) -> None:
    data = DashboardData()
    hist = History()
    with Live(render_dashboard(data, hist), refresh_per_second=4,
              screen=iterations is None) as live:
        n = 0
        while iterations is None or n < iterations:
            time.sleep(refresh_s if iterations is None else 0.01)
            live.update(render_dashboard(data, hist))
            n += 1
