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


def _kv_table(rows: list[tuple[str, str]]) -> Table:
    t = Table.grid(padding=(0, 2))
    t.add_column(style="dim")
    t.add_column()
    for k, v in rows:
        t.add_row(k, str(v))
    return t


def render_dashboard(data: DashboardData):
    s = data.snapshot()
    rt = s["runtime"]
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
    layout["mid"].split_row(
        Layout(Panel(_kv_table([
            ("documents", s["docs"]),
            ("domains", s["domains"]),
            ("indexed last hour", s["recent_docs"]),
            ("link edges", s["link_edges"]),
        ]), title="index")),
        Layout(Panel(_kv_table([
            ("engine docs", rt.get("engine_docs", "—")),
            ("seen URLs", s["seen_urls"]),
            ("trust subjects", s["trust_subjects"]),
        ]), title="crawl / engine")),
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
    with Live(render_dashboard(data), refresh_per_second=4,
              screen=iterations is None) as live:
        n = 0
        while iterations is None or n < iterations:
            time.sleep(refresh_s if iterations is None else 0.01)
            live.update(render_dashboard(data))
            n += 1
