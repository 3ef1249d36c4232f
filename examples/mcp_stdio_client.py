#!/usr/bin/env python3
"""Drive the MCP stdio transport end to end: initialize, tools/list,
index a doc through crawl-free ingestion, then web_search — all over
JSON-RPC 2.0 on pipes, exactly as an MCP host (Claude Desktop etc.)
would.

    python examples/mcp_stdio_client.py
"""
from __future__ import annotations

import io
import json
import sys
import threading
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from infomesh_amd.config import Config
from infomesh_amd.index.local_store import Document
from infomesh_amd.mcp.server import McpServer
from infomesh_amd.services import AppContext


def main() -> None:
    ctx = AppContext.create(config=Config(), with_worker=False,
                            with_engine=False, in_memory=True)
    try:
        ctx.index_document(Document(
            url="https://rocm.docs/mfma", title="MFMA matrix cores",
            text="MFMA instructions drive matrix math on CDNA4 GPUs. " * 6))

        reqs = [
            {"jsonrpc": "2.0", "id": 1, "method": "initialize",
             "params": {"protocolVersion": "2024-11-05"}},
            {"jsonrpc": "2.0", "id": 2, "method": "tools/list"},
            {"jsonrpc": "2.0", "id": 3, "method": "tools/call",
             "params": {"name": "web_search",
                        "arguments": {"query": "mfma matrix"}}},
        ]
        stdin = io.StringIO("".join(json.dumps(r) + "\n" for r in reqs))
        stdout = io.StringIO()
        McpServer(ctx).run_stdio(stdin=stdin, stdout=stdout)
        for line in stdout.getvalue().splitlines():
            msg = json.loads(line)
            if msg.get("id") == 2:
                names = [t["name"] for t in msg["result"]["tools"]]
                print("tools:", names)
            if msg.get("id") == 3:
                text = msg["result"]["content"][0]["text"]
                print("search result mentions MFMA:",
                      "MFMA" in text or "mfma" in text)
    finally:
        ctx.close()


if __name__ == "__main__":
    main()
