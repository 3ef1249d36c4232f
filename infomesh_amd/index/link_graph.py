"""Link graph + domain authority.

Reference parity: infomesh/index/link_graph.py (SQLite edge table,
damped PageRank-style domain-authority propagation — 20 iterations,
damping 0.85; `url_authority` is the authority_fn injected into
ranking).
"""
from __future__ import annotations

from collections import defaultdict
from pathlib import Path

from ..db import SQLiteStore
from .local_store import extract_domain

DAMPING = 0.85
ITERATIONS = 20


class LinkGraph(SQLiteStore):
    SCHEMA = """
    CREATE TABLE IF NOT EXISTS links (
        source_url TEXT NOT NULL,
        target_url TEXT NOT NULL,
        source_domain TEXT NOT NULL,
        target_domain TEXT NOT NULL,
        PRIMARY KEY (source_url, target_url)
    );
    CREATE INDEX IF NOT EXISTS idx_links_sd ON links(source_domain);
    CREATE INDEX IF NOT EXISTS idx_links_td ON links(target_domain);
    """

    def __init__(self, path: str | Path = ":memory:"):
        super().__init__(path)
        self._authority: dict[str, float] = {}
        self._dirty = True

    def add_links(self, source_url: str, target_urls: list[str]) -> int:
        sd = extract_domain(source_url)
        rows = []
        for t in target_urls:
            td = extract_domain(t)
            if td:
                rows.append((source_url, t, sd, td))
        if not rows:
            return 0
        self.executemany(
            "INSERT OR IGNORE INTO links VALUES (?,?,?,?)", rows)
        self.commit()
        self._dirty = True
        return len(rows)

    def edge_count(self) -> int:
        return int(self.execute("SELECT COUNT(*) AS c FROM links").fetchone()["c"])

    def _compute_authority(self) -> dict[str, float]:
        """Damped PageRank over the domain graph."""
        edges = self.execute(
            "SELECT DISTINCT source_domain, target_domain FROM links"
            " WHERE source_domain != target_domain").fetchall()
        out_edges: dict[str, set[str]] = defaultdict(set)
        domains: set[str] = set()
        for r in edges:
            s, t = r["source_domain"], r["target_domain"]
            out_edges[s].add(t)
            domains.add(s)
            domains.add(t)
        if not domains:
            return {}
        n = len(domains)
        rank = {d: 1.0 / n for d in domains}
        for _ in range(ITERATIONS):
            nxt = {d: (1.0 - DAMPING) / n for d in domains}
            for s, targets in out_edges.items():
                share = DAMPING * rank[s] / len(targets)
                for t in targets:
                    nxt[t] += share
            rank = nxt
        mx = max(rank.values())
        if mx > 0:
            rank = {d: v / mx for d, v in rank.items()}
        return rank

    def domain_authority(self, domain: str) -> float:
        if self._dirty:
            self._authority = self._compute_authority()
            self._dirty = False
        return self._authority.get(domain.lower(), 0.0)

    def url_authority(self, url: str) -> float:
        """The authority_fn injected into composite ranking
        (reference: mcp/handlers.py:410). Neutral 0.5 baseline blended
        with the domain's PageRank share."""
        d = extract_domain(url)
        if not d:
            return 0.5
        return 0.3 + 0.7 * self.domain_authority(d)

    def top_domains(self, limit: int = 10) -> list[tuple[str, float]]:
        if self._dirty:
            self._authority = self._compute_authority()
            self._dirty = False
        return sorted(self._authority.items(), key=lambda p: -p[1])[:limit]
