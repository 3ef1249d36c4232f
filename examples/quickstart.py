#!/usr/bin/env python3
"""Quickstart: index a few documents, search, explain, snapshot.

Runs fully on CPU (the same engine code runs the GPU path on MI355X).
    python examples/quickstart.py
"""
from __future__ import annotations

import json
import sys
import tempfile
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from infomesh_amd.config import Config
from infomesh_amd.index.local_store import Document
from infomesh_amd.services import AppContext


def main() -> None:
    ctx = AppContext.create(config=Config(), with_worker=False,
                            with_engine=False, in_memory=True)
    try:
        docs = [
            ("https://rocm.docs/hip", "HIP kernels on CDNA4",
             "MFMA matrix cores, 64-wide wavefronts and the LDS drive "
             "high-throughput GPU kernels on the MI355X."),
            ("https://rocm.docs/rccl", "RCCL collectives",
             "All-reduce and all-gather over xGMI links scale multi-GPU "
             "training and serving on one node."),
            ("https://example.com/pasta", "Cooking pasta",
             "Boil until al dente and finish in the sauce."),
        ]
        for url, title, text in docs:
            ctx.index_document(Document(url=url, title=title, text=text))

        resp = ctx.search("mfma kernels lds", limit=3)
        print("results:")
        for r in resp.results:
            print(f"  {r.score:.3f}  {r.url}  {r.title}")

        from infomesh_amd.search.explain import explain_search
        exp = explain_search(ctx.store, "mfma kernels", limit=1)
        print("explain:", json.dumps(exp[0]["components"], indent=2))

        with tempfile.TemporaryDirectory() as d:
            from infomesh_amd.index.snapshot import (export_snapshot,
                                                     import_snapshot)
            p = Path(d) / "demo.infomesh-snapshot"
            header = export_snapshot(ctx.store, p)
            print("snapshot:", header["doc_count"], "docs,",
                  p.stat().st_size, "bytes")
            from infomesh_amd.index.local_store import LocalStore
            restored = LocalStore(":memory:")
            import_snapshot(restored, p)
            print("restored docs:", restored.count())
    finally:
        ctx.close()


if __name__ == "__main__":
    main()
