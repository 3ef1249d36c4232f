"""LocalStore — the canonical SQLite FTS5 document store.

Reference parity: infomesh/index/local_store.py (documents table +
external-content FTS5 with sync triggers, tokenizer whitelist, WAL,
url/text_hash dedup, FTS5 MATCH + bm25() search with snippet() and
language/date/domain filters, suggest, optimize, export, recrawl
metadata).

In the MI355X design this store is the durable ground truth; GPU shards
are rebuildable caches over it loaded via the snapshot
path or direct scans.
"""
from __future__ import annotations

import re
import sqlite3
import time
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Iterator
from urllib.parse import urlparse

from ..errors import InfoMeshError
from ..hashing import content_hash

FTS_TOKENIZERS = {
    "unicode61": "unicode61 remove_diacritics 2",
    "ascii": "ascii",
    "porter": "porter unicode61",
    "trigram": "trigram",
}

_SCHEMA = """
CREATE TABLE IF NOT EXISTS documents (
    id INTEGER PRIMARY KEY,
    url TEXT NOT NULL UNIQUE,
    title TEXT NOT NULL DEFAULT '',
    text TEXT NOT NULL DEFAULT '',
    language TEXT NOT NULL DEFAULT '',
    domain TEXT NOT NULL DEFAULT '',
    text_hash TEXT NOT NULL UNIQUE,
    raw_hash TEXT NOT NULL DEFAULT '',
    crawled_at REAL NOT NULL,
    updated_at REAL NOT NULL,
    etag TEXT NOT NULL DEFAULT '',
    last_modified TEXT NOT NULL DEFAULT '',
    recrawl_interval_s REAL NOT NULL DEFAULT 86400.0,
    stale_count INTEGER NOT NULL DEFAULT 0,
    shard INTEGER NOT NULL DEFAULT -1
);
CREATE INDEX IF NOT EXISTS idx_documents_domain ON documents(domain);
CREATE INDEX IF NOT EXISTS idx_documents_crawled ON documents(crawled_at);
"""

_FTS_SCHEMA = """
CREATE VIRTUAL TABLE IF NOT EXISTS documents_fts USING fts5(
    title, text, content='documents', content_rowid='id', tokenize='{tok}'
);
CREATE TRIGGER IF NOT EXISTS documents_ai AFTER INSERT ON documents BEGIN
    INSERT INTO documents_fts(rowid, title, text)
    VALUES (new.id, new.title, new.text);
END;
CREATE TRIGGER IF NOT EXISTS documents_ad AFTER DELETE ON documents BEGIN
    INSERT INTO documents_fts(documents_fts, rowid, title, text)
    VALUES ('delete', old.id, old.title, old.text);
END;
CREATE TRIGGER IF NOT EXISTS documents_au AFTER UPDATE OF title, text ON documents BEGIN
    INSERT INTO documents_fts(documents_fts, rowid, title, text)
    VALUES ('delete', old.id, old.title, old.text);
    INSERT INTO documents_fts(rowid, title, text)
    VALUES (new.id, new.title, new.text);
END;
"""


@dataclass
class Document:
    url: str
    title: str = ""
    text: str = ""
    language: str = ""
    doc_id: int | None = None
    domain: str = ""
    text_hash: str = ""
    raw_hash: str = ""
    crawled_at: float = 0.0
    updated_at: float = 0.0
    etag: str = ""
    last_modified: str = ""
    extra: dict[str, Any] = field(default_factory=dict)


@dataclass
class SearchHit:
    doc_id: int
    url: str
    title: str
    snippet: str
    bm25: float            # positive relevance (negated sqlite bm25())
    language: str = ""
    domain: str = ""
    crawled_at: float = 0.0
    score: float = 0.0     # composite, filled by ranking
    source: str = "fts"


def extract_domain(url: str) -> str:
    try:
        host = urlparse(url).hostname or ""
    except ValueError:
        return ""
    return host.lower()


# FTS5 query sanitizer (reference: search/query.py:54-79): strip operators
# that would make MATCH raise, quote each bareword token.
_TOKEN_RE = re.compile(r"[^\s\"'()*:^]+")


def sanitize_fts_query(query: str, match_any: bool = False) -> str:
    tokens = _TOKEN_RE.findall(query)
    quoted = ['"' + t.replace('"', "") + '"' for t in tokens if t.strip('"')]
    return (" OR " if match_any else " ").join(quoted)


class LocalStore:
    """The canonical FTS5-backed document store."""

    def __init__(self, path: str | Path = ":memory:", tokenizer: str = "unicode61",
                 max_text_chars: int = 500_000):
        if tokenizer not in FTS_TOKENIZERS:
            raise InfoMeshError("IDX003", tokenizer)
        self.path = str(path)
        self.tokenizer = tokenizer
        self.max_text_chars = int(max_text_chars)
        if self.path != ":memory:":
            Path(self.path).parent.mkdir(parents=True, exist_ok=True)
        self.conn = sqlite3.connect(self.path, check_same_thread=False)
        self.conn.row_factory = sqlite3.Row
        cur = self.conn.cursor()
        cur.execute("PRAGMA busy_timeout=5000")
        if self.path != ":memory:":
            cur.execute("PRAGMA journal_mode=WAL")
        cur.execute("PRAGMA synchronous=NORMAL")
        self.conn.executescript(_SCHEMA)
        self.conn.executescript(
            _FTS_SCHEMA.format(tok=FTS_TOKENIZERS[tokenizer]))
        self._migrate()
        self.conn.commit()

    # ------------------------------------------------------------ migrate
    def _migrate(self) -> None:
        """Additive migrations (reference: local_store.py:150-196)."""
        cols = {r["name"] for r in self.conn.execute("PRAGMA table_info(documents)")}
        wanted = {
            "shard": "INTEGER NOT NULL DEFAULT -1",
            "recrawl_interval_s": "REAL NOT NULL DEFAULT 86400.0",
            "stale_count": "INTEGER NOT NULL DEFAULT 0",
        }
        for col, decl in wanted.items():
            if col not in cols:
                self.conn.execute(f"ALTER TABLE documents ADD COLUMN {col} {decl}")

    # ------------------------------------------------------------- ingest
    def add_document(self, doc: Document, force: bool = False) -> int | None:
        """Insert a document; dedup by url and text_hash.

        Returns the rowid, or None when skipped as a duplicate
        (reference: local_store.py:198-251)."""
        now = time.time()
        if len(doc.text) > self.max_text_chars:   # config index.max_text_chars
            doc.text = doc.text[:self.max_text_chars]
        text_hash = doc.text_hash or content_hash(doc.text)
        domain = doc.domain or extract_domain(doc.url)
        existing = self.conn.execute(
            "SELECT id, text_hash FROM documents WHERE url=?",
            (doc.url,)).fetchone()
        if existing is not None:
            if existing["text_hash"] == text_hash and not force:
                # Unchanged content: just refresh the crawl timestamp.
                self.conn.execute(
                    "UPDATE documents SET crawled_at=? WHERE id=?",
                    (now, existing["id"]))
                self.conn.commit()
                return None
            self.conn.execute(
                "UPDATE documents SET title=?, text=?, language=?, domain=?,"
                " text_hash=?, raw_hash=?, crawled_at=?, updated_at=?,"
                " etag=?, last_modified=?, stale_count=0 WHERE id=?",
                (doc.title, doc.text, doc.language, domain, text_hash,
                 doc.raw_hash, now, now, doc.etag, doc.last_modified,
                 existing["id"]))
            self.conn.commit()
            return int(existing["id"])
        dup = self.conn.execute(
            "SELECT id FROM documents WHERE text_hash=?", (text_hash,)).fetchone()
        if dup is not None and not force:
            return None
        cur = self.conn.execute(
            "INSERT INTO documents (url, title, text, language, domain,"
            " text_hash, raw_hash, crawled_at, updated_at, etag, last_modified)"
            " VALUES (?,?,?,?,?,?,?,?,?,?,?)",
            (doc.url, doc.title, doc.text, doc.language, domain,
             text_hash if dup is None else content_hash(doc.text + doc.url),
             doc.raw_hash, doc.crawled_at or now, now, doc.etag,
             doc.last_modified))
        self.conn.commit()
        return int(cur.lastrowid)

    def delete_document(self, doc_id: int) -> bool:
        cur = self.conn.execute("DELETE FROM documents WHERE id=?", (doc_id,))
        self.conn.commit()
        return cur.rowcount > 0

    def delete_by_url(self, url: str) -> bool:
        cur = self.conn.execute("DELETE FROM documents WHERE url=?", (url,))
        self.conn.commit()
        return cur.rowcount > 0

    def delete_by_domain(self, domain: str) -> int:
        """Bulk takedown path for DMCA/GDPR compliance."""
        cur = self.conn.execute(
            "DELETE FROM documents WHERE domain=?", (domain.lower(),))
        self.conn.commit()
        return cur.rowcount

    # ------------------------------------------------------------- search
    def search(self, query: str, limit: int = 10,
               language: str | None = None,
               domain: str | None = None,
               after: float | None = None,
               before: float | None = None,
               match_any: bool = False) -> list[SearchHit]:
        """FTS5 MATCH + bm25() ordering + snippet() with filters
        (reference: local_store.py:253-352). match_any=True ORs the
        terms (fact-check / recall-first queries)."""
        fts_query = sanitize_fts_query(query, match_any=match_any)
        if not fts_query:
            return []
        sql = [
            "SELECT d.id, d.url, d.title, d.language, d.domain, d.crawled_at,",
            " bm25(documents_fts, 2.0, 1.0) AS rank,",
            " snippet(documents_fts, 1, '<b>', '</b>', '…', 24) AS snip",
            " FROM documents_fts JOIN documents d ON d.id = documents_fts.rowid",
            " WHERE documents_fts MATCH ?",
        ]
        params: list[Any] = [fts_query]
        if language:
            sql.append(" AND d.language = ?")
            params.append(language)
        if domain:
            sql.append(" AND d.domain = ?")
            params.append(domain.lower())
        if after is not None:
            sql.append(" AND d.crawled_at >= ?")
            params.append(after)
        if before is not None:
            sql.append(" AND d.crawled_at <= ?")
            params.append(before)
        sql.append(" ORDER BY rank LIMIT ?")
        params.append(max(1, int(limit)))
        try:
            rows = self.conn.execute("".join(sql), params).fetchall()
        except sqlite3.OperationalError as e:
            raise InfoMeshError("SRCH001", str(e)) from e
        return [SearchHit(doc_id=r["id"], url=r["url"], title=r["title"],
                          snippet=r["snip"] or "",
                          bm25=-float(r["rank"]),  # sqlite bm25() is negative
                          language=r["language"], domain=r["domain"],
                          crawled_at=r["crawled_at"]) for r in rows]

    def suggest(self, prefix: str, limit: int = 5) -> list[str]:
        """Title-prefix suggestions (reference: local_store.py:354)."""
        prefix = prefix.strip()
        if not prefix:
            return []
        rows = self.conn.execute(
            "SELECT title FROM documents WHERE title LIKE ? || '%'"
            " ORDER BY crawled_at DESC LIMIT ?", (prefix, limit)).fetchall()
        return [r["title"] for r in rows]

    # ------------------------------------------------------------- access
    def _row_to_doc(self, r: sqlite3.Row) -> Document:
        return Document(url=r["url"], title=r["title"], text=r["text"],
                        language=r["language"], doc_id=r["id"],
                        domain=r["domain"], text_hash=r["text_hash"],
                        raw_hash=r["raw_hash"], crawled_at=r["crawled_at"],
                        updated_at=r["updated_at"], etag=r["etag"],
                        last_modified=r["last_modified"])

    def get_document(self, doc_id: int) -> Document | None:
        r = self.conn.execute(
            "SELECT * FROM documents WHERE id=?", (doc_id,)).fetchone()
        return self._row_to_doc(r) if r else None

    def get_documents(self, ids: list[int]) -> dict[int, "Document"]:
        """Batched lookup (one IN query) — the serving batcher hydrates
        a whole GPU batch's results in one round trip."""
        if not ids:
            return {}
        out: dict[int, Document] = {}
        uniq = list({int(i) for i in ids})
        for i in range(0, len(uniq), 500):   # SQLite var limit safety
            chunk = uniq[i:i + 500]
            q = ("SELECT * FROM documents WHERE id IN (%s)"
                 % ",".join("?" * len(chunk)))
            for r in self.conn.execute(q, chunk).fetchall():
                d = self._row_to_doc(r)
                out[d.doc_id] = d
        return out

    def get_document_by_url(self, url: str) -> Document | None:
        r = self.conn.execute(
            "SELECT * FROM documents WHERE url=?", (url,)).fetchone()
        return self._row_to_doc(r) if r else None

    def count(self) -> int:
        return int(self.conn.execute(
            "SELECT COUNT(*) AS c FROM documents").fetchone()["c"])

    def stats(self) -> dict[str, Any]:
        c = self.count()
        domains = int(self.conn.execute(
            "SELECT COUNT(DISTINCT domain) AS c FROM documents").fetchone()["c"])
        size = 0
        if self.path != ":memory:":
            p = Path(self.path)
            if p.exists():
                size = p.stat().st_size
        return {"documents": c, "domains": domains, "db_bytes": size,
                "tokenizer": self.tokenizer}

    def export_documents(self, batch: int = 1000) -> Iterator[Document]:
        """Full scan for snapshot export (reference: local_store.py:477-498)."""
        last_id = 0
        while True:
            rows = self.conn.execute(
                "SELECT * FROM documents WHERE id > ? ORDER BY id LIMIT ?",
                (last_id, batch)).fetchall()
            if not rows:
                return
            for r in rows:
                last_id = r["id"]
                yield self._row_to_doc(r)

    def iter_for_shard(self, shard: int, n_shards: int,
                       batch: int = 5000) -> Iterator[Document]:
        """Documents owned by one GPU shard (id % n_shards == shard)."""
        last_id = 0
        while True:
            rows = self.conn.execute(
                "SELECT * FROM documents WHERE id > ? AND (id % ?) = ?"
                " ORDER BY id LIMIT ?",
                (last_id, n_shards, shard, batch)).fetchall()
            if not rows:
                return
            for r in rows:
                last_id = r["id"]
                yield self._row_to_doc(r)

    # ------------------------------------------------------------ recrawl
    def update_recrawl(self, url: str, changed: bool,
                       etag: str = "", last_modified: str = "") -> None:
        """Adaptive recrawl metadata (reference: local_store.py:549-634,
        crawler/recrawl.py:70-110): halve the interval on change, grow
        1.5× (cap 30 d) when unchanged."""
        r = self.conn.execute(
            "SELECT recrawl_interval_s, stale_count FROM documents WHERE url=?",
            (url,)).fetchone()
        if r is None:
            return
        interval = float(r["recrawl_interval_s"])
        if changed:
            interval = max(3600.0, interval * 0.5)
            stale = 0
        else:
            interval = min(30 * 86400.0, interval * 1.5)
            stale = int(r["stale_count"]) + 1
        self.conn.execute(
            "UPDATE documents SET recrawl_interval_s=?, stale_count=?,"
            " etag=?, last_modified=?, crawled_at=? WHERE url=?",
            (interval, stale, etag, last_modified, time.time(), url))
        self.conn.commit()

    def due_for_recrawl(self, limit: int = 100) -> list[Document]:
        now = time.time()
        rows = self.conn.execute(
            "SELECT * FROM documents WHERE crawled_at + recrawl_interval_s < ?"
            " ORDER BY crawled_at LIMIT ?", (now, limit)).fetchall()
        return [self._row_to_doc(r) for r in rows]

    # ---------------------------------------------------------- lifecycle
    def optimize(self) -> None:
        """FTS5 segment merge (reference: local_store.py:528-541)."""
        self.conn.execute(
            "INSERT INTO documents_fts(documents_fts) VALUES('optimize')")
        self.conn.commit()

    def close(self) -> None:
        try:
            self.conn.commit()
        except sqlite3.Error:
            pass
        self.conn.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
        return False
