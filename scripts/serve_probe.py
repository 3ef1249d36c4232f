#!/usr/bin/env python3
"""MCP-HTTP serving throughput probe: concurrent clients → micro-batcher
→ GPU plane. Evidence for VERDICT #1 (the benched batch QPS must be
reachable from the real entry points, not only bench.py).

Server runs in this process (uvicorn, one worker); clients run in
SEPARATE processes (so client-side Python does not steal the server's
GIL), each an asyncio httpx pool. Reports aggregate QPS + latency
percentiles + the server's batcher stats.

Usage (GPU box):
  python scripts/serve_probe.py --docs 50000 --clients 4 --conc 64 --secs 10
"""
from __future__ import annotations

import argparse
import json
import os
import multiprocessing as mp
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

WORDS = ("rocm hip kernel mfma lds wave xcd hbm cache stream graph "
         "matrix tensor shard index query search crawl robot feed "
         "credit trust merkle audit snapshot passage rerank summarize "
         "scheduler python torch rccl xgmi fabric allgather broadcast "
         "segment posting cosine bm25 fusion batch latency throughput").split()


def _client_proc(port: int, conc: int, secs: int, seed: int, out_q):
    import asyncio
    import random

    import httpx

    rng = random.Random(seed)
    queries = [" ".join(rng.sample(WORDS, 3)) for _ in range(512)]

    batch = int(os.environ.get("PROBE_BATCH", "0"))

    async def run():
        lat = []
        n = 0
        errors = 0
        deadline = time.perf_counter() + secs
        async with httpx.AsyncClient(
                base_url=f"http://127.0.0.1:{port}", timeout=30) as client:

            async def worker(wid: int):
                nonlocal n, errors
                i = wid
                while time.perf_counter() < deadline:
                    i += conc
                    if batch > 1:
                        qs = [queries[(i + j) % len(queries)]
                              for j in range(batch)]
                        args = {"queries": qs, "limit": 10}
                        name = "batch_search"
                    else:
                        args = {"query": queries[i % len(queries)],
                                "limit": 10}
                        name = "web_search"
                    body = {"jsonrpc": "2.0", "id": i,
                            "method": "tools/call",
                            "params": {"name": name, "arguments": args}}
                    t0 = time.perf_counter()
                    try:
                        r = await client.post("/mcp", json=body)
                        ok = r.status_code == 200 and \
                            "result" in r.json()
                    except Exception:
                        ok = False
                    lat.append((time.perf_counter() - t0) * 1e3)
                    n += 1
                    if not ok:
                        errors += 1
            await asyncio.gather(*[worker(w) for w in range(conc)])
        return n, errors, lat

    n, errors, lat = asyncio.run(run())
    out_q.put((n, errors, lat))


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--docs", type=int, default=50_000)
    ap.add_argument("--clients", type=int, default=4)
    ap.add_argument("--conc", type=int, default=64)
    ap.add_argument("--secs", type=int, default=10)
    ap.add_argument("--port", type=int, default=8931)
    ap.add_argument("--no-dense", action="store_true")
    ap.add_argument("--batch", type=int, default=0,
                    help="use the batch_search tool with N queries per "
                         "HTTP request (bulk-caller mode)")
    args = ap.parse_args()
    if args.batch:
        os.environ["PROBE_BATCH"] = str(args.batch)

    import random
    import threading

    import torch
    import uvicorn

    from infomesh_amd.config import Config
    from infomesh_amd.engine import HybridEngine
    from infomesh_amd.index.local_store import Document
    from infomesh_amd.mcp.server import McpServer
    from infomesh_amd.services import AppContext

    use_gpu = torch.cuda.is_available()
    if use_gpu:
        from infomesh_amd.ops import _build
        _build.build()

    t0 = time.perf_counter()
    ctx = AppContext.create(config=Config(), with_engine=False,
                            with_worker=False, in_memory=True)
    ctx.engine = HybridEngine(device="cuda" if use_gpu else "cpu",
                              use_encoder=use_gpu and not args.no_dense)
    rng = random.Random(7)
    batch = []
    for i in range(args.docs):
        text = " ".join(rng.choices(WORDS, k=24))
        batch.append(Document(url=f"https://corp.example/d{i}",
                              title=f"doc {i} {text[:18]}", text=text))
    for d in batch:
        ctx.index_document(d, attest=False, credit=False)
    n_flushed = ctx.flush_engine()
    print(f"setup: {args.docs} docs indexed+flushed({n_flushed}) in "
          f"{time.perf_counter() - t0:.1f}s", flush=True)

    server = McpServer(ctx)
    config = uvicorn.Config(server.asgi_app(), host="127.0.0.1",
                            port=args.port, log_level="error",
                            workers=1)
    usrv = uvicorn.Server(config)
    th = threading.Thread(target=usrv.run, daemon=True)
    th.start()
    for _ in range(100):
        time.sleep(0.1)
        if usrv.started:
            break

    # warm the GPU path once
    ctx.search("rocm hip kernel", use_cache=False, deduct=False)

    procs = []
    out_q: mp.Queue = mp.Queue()
    for c in range(args.clients):
        p = mp.Process(target=_client_proc,
                       args=(args.port, args.conc, args.secs, 100 + c,
                             out_q))
        p.start()
        procs.append(p)
    total = errs = 0
    lats: list[float] = []
    for _ in procs:
        n, e, lat = out_q.get()
        total += n
        errs += e
        lats.extend(lat)
    for p in procs:
        p.join()
    lats.sort()
    per_req = max(args.batch, 1)
    qps = total * per_req / args.secs
    res = {
        "http_qps": round(qps, 1),
        "queries_per_request": per_req,
        "requests": total,
        "errors": errs,
        "p50_ms": round(lats[len(lats) // 2], 2) if lats else None,
        "p95_ms": round(lats[int(len(lats) * 0.95)], 2) if lats else None,
        "p99_ms": round(lats[int(len(lats) * 0.99)], 2) if lats else None,
        "clients": args.clients, "conc_per_client": args.conc,
        "docs": args.docs,
        "batcher": ctx.batcher.stats() if ctx.batcher else None,
        "gpu": use_gpu,
    }
    print(json.dumps(res))
    usrv.should_exit = True
    ctx.close()


if __name__ == "__main__":
    main()
