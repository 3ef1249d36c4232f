"""FastAPI admin app on localhost.

Reference parity: infomesh/api/local_api.py (localhost-only middleware,
security headers, rate limiting, API keys; routes /health, /readiness,
/search, /status, /config(+/reload), /index/stats, /credits/balance,
/analytics, /metrics, /dashboard; OpenAPI spec comes with FastAPI).
"""
from __future__ import annotations

import dataclasses
import time
from collections import defaultdict, deque

from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import JSONResponse, PlainTextResponse

from ..search.formatter import result_to_dict
from ..services import AppContext
from ..utils.observability import MetricsRegistry

SECURITY_HEADERS = {
    "X-Content-Type-Options": "nosniff",
    "X-Frame-Options": "DENY",
    "Referrer-Policy": "no-referrer",
    "Cache-Control": "no-store",
}


def create_app(ctx: AppContext, api_key: str = "",
               rate_limit_per_min: int = 120,
               metrics: MetricsRegistry | None = None) -> FastAPI:
    app = FastAPI(title="infomesh-amd admin API", version="0.1.0",
                  docs_url="/docs")
    metrics = metrics or MetricsRegistry()
    buckets: dict[str, deque] = defaultdict(deque)

    @app.middleware("http")
    async def guard(request: Request, call_next):
        client = request.client.host if request.client else ""
        # localhost-only (reference local_api.py:106-183)
        if client not in ("127.0.0.1", "::1", "testclient", ""):
            return JSONResponse({"error": "localhost only"}, status_code=403)
        if api_key and request.headers.get("x-api-key") != api_key \
                and request.url.path not in ("/health", "/readiness"):
            return JSONResponse({"error": "bad api key"}, status_code=401)
        q = buckets[client]
        now = time.time()
        while q and q[0] < now - 60:
            q.popleft()
        if len(q) >= rate_limit_per_min:
            return JSONResponse({"error": "rate limited"}, status_code=429)
        q.append(now)
        t0 = time.perf_counter()
        resp = await call_next(request)
        metrics.observe("api_request_seconds", time.perf_counter() - t0,
                        labels={"path": request.url.path})
        metrics.inc("api_requests_total", labels={
            "path": request.url.path, "status": str(resp.status_code)})
        for k, v in SECURITY_HEADERS.items():
            resp.headers[k] = v
        return resp

    @app.get("/health")
    def health(detail: int = 0):
        out = {"ok": True, "ts": time.time()}
        if detail:
            out["status"] = ctx.status()
        return out

    @app.get("/readiness")
    def readiness():
        ready = ctx.store is not None
        return {"ready": ready, "engine": ctx.engine is not None}

    @app.get("/search")
    def search(q: str, limit: int = 10, mode: str = "auto"):
        if not q.strip():
            raise HTTPException(400, "empty query")
        resp = ctx.search(q, limit=min(limit, 50), mode=mode)
        return {"query": q, "mode": resp.mode,
                "elapsed_ms": round(resp.elapsed_ms, 2),
                "results": [result_to_dict(r) for r in resp.results]}

    @app.get("/status")
    def status():
        return ctx.status()

    @app.get("/config")
    def get_config():
        return dataclasses.asdict(ctx.config)

    @app.post("/config/reload")
    def reload_config():
        from ..config import load_config
        new = load_config()
        object.__setattr__(ctx, "config", new) if dataclasses.is_dataclass(ctx) \
            else setattr(ctx, "config", new)
        return {"reloaded": True}

    @app.get("/index/stats")
    def index_stats():
        out = ctx.store.stats()
        if ctx.engine:
            out["engine"] = ctx.engine.stats()
        return out

    @app.get("/credits/balance")
    def credits_balance():
        return ctx.ledger.stats()

    @app.get("/network/peers")
    def peers():
        """Intra-node fabric view (replaces the libp2p peer list)."""
        eng = ctx.engine
        return {"world_size": eng.fabric.world if eng else 1,
                "backend": eng.fabric.backend if eng else "none",
                "shards": [eng.stats()] if eng else []}

    @app.get("/analytics")
    def analytics():
        return {"cache": ctx.cache.stats(),
                "crawler": ctx.worker.stats if ctx.worker else None}

    @app.get("/analytics/tools")
    def analytics_tools():
        """MCP tool usage breakdown (reference local_api.py:396-413)."""
        stats = getattr(app.state, "tool_counts",
                        {"web_search": 0, "crawl_url": 0, "fetch_page": 0})
        total = sum(stats.values())
        searches = stats.get("web_search", 0)
        return {"tool_usage": {**stats, "total": total},
                "search_fetch_rate": round(
                    stats.get("fetch_page", 0) / searches * 100, 1)
                if searches else 0.0}

    @app.get("/index/compression")
    def index_compression():
        """Index compression statistics (reference local_api.py:417-435)."""
        st = ctx.store.stats()
        docs = int(st.get("documents", 0))
        db_mb = round(float(st.get("db_bytes", 0)) / 1e6, 2)
        return {"documents": docs, "db_size_mb": db_mb,
                "avg_doc_kb": round(db_mb * 1024 / docs, 2) if docs else 0.0,
                "compression_enabled": True,
                "compression_level": 12}

    @app.get("/openapi-spec")
    def openapi_spec():
        """Reference-named alias of FastAPI's /openapi.json."""
        return app.openapi()

    @app.get("/metrics")
    def prom_metrics():
        return PlainTextResponse(metrics.render(),
                                 media_type="text/plain; version=0.0.4")

    @app.get("/dashboard")
    def dashboard():
        from ..utils.text_report import render_report
        return PlainTextResponse(render_report(ctx))

    @app.post("/feedback")
    def feedback(url: str, signal: str, q: str = ""):
        from ..search.feedback import FeedbackStore
        if not hasattr(app.state, "feedback"):
            app.state.feedback = FeedbackStore(
                ctx.config.data_dir / "feedback.db"
                if ctx.store.path != ":memory:" else ":memory:")
        app.state.feedback.record(url, signal, q)
        return {"recorded": True}

    return app


def run_api(ctx: AppContext | None = None, host: str = "127.0.0.1",
            port: int = 8080, api_key: str = "") -> None:
    import uvicorn
    ctx = ctx or AppContext.create()
    uvicorn.run(create_app(ctx, api_key=api_key), host=host, port=port,
                log_level="warning")
