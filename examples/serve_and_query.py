#!/usr/bin/env python3
"""Serve + query example: admin HTTP API on :8080-equivalent test port,
queried through the Python SDK client.

    python examples/serve_and_query.py
"""
from __future__ import annotations

import sys
import threading
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import uvicorn

from infomesh_amd.api.local_api import create_app
from infomesh_amd.config import Config
from infomesh_amd.index.local_store import Document
from infomesh_amd.sdk.client import InfoMeshClient
from infomesh_amd.services import AppContext


def main(port: int = 18480) -> None:
    ctx = AppContext.create(config=Config(), with_worker=False,
                            with_engine=False, in_memory=True)
    try:
        for i, topic in enumerate(["mfma kernels", "rccl collectives",
                                   "hip streams"]):
            ctx.index_document(Document(
                url=f"https://docs.example/{i}", title=topic.title(),
                text=f"All about {topic} on the MI355X GPU. " * 8))

        app = create_app(ctx)
        server = uvicorn.Server(uvicorn.Config(
            app, host="127.0.0.1", port=port, log_level="error"))
        th = threading.Thread(target=server.run, daemon=True)
        th.start()
        for _ in range(100):
            time.sleep(0.05)
            if server.started:
                break

        client = InfoMeshClient(base_url=f"http://127.0.0.1:{port}")
        print("health:", client.health())
        hits = client.search("rccl collectives", limit=2)
        for h in hits:
            print(f"  {h.get('score', 0):.3f}  {h['url']}")
        server.should_exit = True
        th.join(timeout=5)
    finally:
        ctx.close()


if __name__ == "__main__":
    main()
