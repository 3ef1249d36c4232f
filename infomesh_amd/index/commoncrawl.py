"""Common Crawl WET import.

Reference parity: infomesh/index/commoncrawl.py (WET-file parser +
streaming importer with size guards). WET files are WARC `conversion`
records with plain-text payloads; this parses local (optionally
gzipped) WET files — there is no network in the target image, so files
arrive on disk.
"""
from __future__ import annotations

import gzip
import io
from dataclasses import dataclass
from pathlib import Path
from typing import Callable, Iterator

from ..hashing import content_hash
from .local_store import Document, LocalStore

MAX_RECORD_BYTES = 1_000_000
MIN_TEXT_CHARS = 100


@dataclass
class WetRecord:
    url: str
    text: str
    language: str = ""


def parse_wet(stream: io.BufferedIOBase,
              max_records: int | None = None) -> Iterator[WetRecord]:
    """Stream WARC/WET records: headers until blank line, then
    Content-Length payload bytes."""
    n = 0
    text_stream = io.TextIOWrapper(stream, encoding="utf-8",
                                   errors="replace", newline="")
    while True:
        # find record start
        line = text_stream.readline()
        if not line:
            return
        if not line.startswith("WARC/"):
            continue
        headers: dict[str, str] = {}
        while True:
            h = text_stream.readline()
            if not h or h.strip() == "":
                break
            if ":" in h:
                k, v = h.split(":", 1)
                headers[k.strip().lower()] = v.strip()
        length = int(headers.get("content-length", "0") or 0)
        if length <= 0 or length > MAX_RECORD_BYTES:
            # skip payload
            text_stream.read(min(length, MAX_RECORD_BYTES * 10))
            continue
        payload = text_stream.read(length)
        if headers.get("warc-type") != "conversion":
            continue
        url = headers.get("warc-target-uri", "")
        if not url or len(payload) < MIN_TEXT_CHARS:
            continue
        yield WetRecord(url=url, text=payload,
                        language=headers.get(
                            "warc-identified-content-language", "")[:2])
        n += 1
        if max_records is not None and n >= max_records:
            return


class CommonCrawlImporter:
    def __init__(self, store: LocalStore,
                 on_document: Callable[[Document], None] | None = None):
        self.store = store
        self.on_document = on_document

    def import_wet(self, path: str | Path,
                   max_records: int | None = None) -> dict:
        path = Path(path)
        opener = gzip.open if path.suffix == ".gz" else open
        imported = skipped = 0
        with opener(path, "rb") as f:
            for rec in parse_wet(f, max_records):
                from .local_store import extract_domain
                if not extract_domain(rec.url):
                    skipped += 1
                    continue
                doc = Document(url=rec.url, title=rec.text.split("\n", 1)[0][:200],
                               text=rec.text, language=rec.language,
                               text_hash=content_hash(rec.text))
                rid = self.store.add_document(doc)
                if rid is None:
                    skipped += 1
                else:
                    imported += 1
                    doc.doc_id = rid
                    if self.on_document:
                        self.on_document(doc)
        return {"imported": imported, "skipped": skipped}
