#!/usr/bin/env python3
"""One-command pre-flight: everything that can be verified WITHOUT a
GPU. Mirrors the driver's CPU-side checks plus the examples.

    python scripts/release_check.py
"""
from __future__ import annotations

import subprocess
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent


def run(name: str, cmd: list[str], timeout: int = 900) -> bool:
    t0 = time.perf_counter()
    r = subprocess.run(cmd, cwd=ROOT, capture_output=True, text=True,
                       timeout=timeout)
    dt = time.perf_counter() - t0
    ok = r.returncode == 0
    print(f"[{'OK' if ok else 'FAIL'}] {name} ({dt:.1f}s)")
    if not ok:
        print(r.stdout[-1500:])
        print(r.stderr[-1500:])
    return ok


def main() -> int:
    checks = [
        ("hip extension builds (gfx950 cross-compile)",
         [sys.executable, "-c",
          "import __graft_entry__ as g; g.build()"]),
        ("CPU test suite",
         [sys.executable, "-m", "pytest", "tests", "-q",
          "-m", "not gpu", "-p", "no:cacheprovider"]),
        ("quickstart example",
         [sys.executable, "examples/quickstart.py"]),
        ("serve+SDK example",
         [sys.executable, "examples/serve_and_query.py"]),
        ("MCP stdio example",
         [sys.executable, "examples/mcp_stdio_client.py"]),
        ("adapter RAG example",
         [sys.executable, "examples/rag_with_adapters.py"]),
        ("bench contract (1 rank, CPU sanity)",
         [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
          "--docs-per-gpu", "5000"]),
    ]
    bad = sum(not run(n, c) for n, c in checks)
    print("ALL GREEN" if bad == 0 else f"{bad} check(s) FAILED")
    return 1 if bad else 0


if __name__ == "__main__":
    raise SystemExit(main())
