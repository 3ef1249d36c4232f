"""MCP tool handlers over an AppContext.

Reference parity: infomesh/mcp/handlers.py (handle_web_search mode
routing -> explain / RAG / answer / plain search; fetch/crawl/fact_check/
status; crawl rate limit 60/hr; query preprocessing + did-you-mean
post-processing).
"""
from __future__ import annotations

import asyncio
import time
from typing import Any

from ..errors import InfoMeshError, format_error
from ..search.explain import explain_search
from ..search.formatter import result_to_dict
from ..search.nlp import did_you_mean
from ..search.rag import format_rag_output, extract_answer
from ..services import AppContext

CRAWL_RATE_PER_HOUR = 60


class Handlers:
    def __init__(self, ctx: AppContext, reranker=None, summarizer=None):
        self.ctx = ctx
        self.reranker = reranker
        self.summarizer = summarizer
        self._crawl_times: list[float] = []

    # ------------------------------------------------------------ search
    def web_search(self, query: str, limit: int = 10, mode: str = "auto",
                   explain: bool = False, chunk_size: int = 0,
                   answer_mode: bool | str = False, rerank: bool = False,
                   summarize: bool = False, top_k: int | None = None,
                   local_only: bool = False, language: str = "",
                   recency_days: float = 0,
                   domain_allowlist: list[str] | None = None,
                   domain_blocklist: list[str] | None = None,
                   fetch_full_content: bool = False,
                   validate: bool = False,
                   **_) -> dict[str, Any]:
        if not query or not query.strip():
            raise InfoMeshError("SRCH001", "empty query")
        # Reference-style argument forms (infomesh mcp/tools.py:53-136):
        # top_k aliases limit; local_only forces the local mode;
        # answer_mode may be the 'snippets'/'answer'/'summary' enum.
        if top_k:
            limit = int(top_k)
        if local_only:
            mode = "local"
        if isinstance(answer_mode, str):
            summarize = summarize or answer_mode == "summary"
            answer_mode = answer_mode in ("answer", "summary")
        if answer_mode and not chunk_size:
            chunk_size = 800
        if language:
            query = f"{query} lang:{language}"
        limit = max(1, min(int(limit), 50))
        if explain:
            return {"mode": "explain",
                    "results": explain_search(
                        self.ctx.store, query, limit,
                        authority_fn=self.ctx.link_graph.url_authority,
                        trust_fn=self.ctx.trust.trust_fn(),
                        boost_fn=(self.ctx.feedback.url_boost
                                  if self.ctx.feedback else None))}
        over = limit * 2 if (domain_allowlist or domain_blocklist
                             or recency_days) else limit
        resp = self.ctx.search(query, limit=over, mode=mode)
        results = [result_to_dict(r) for r in resp.results]
        if domain_allowlist or domain_blocklist or recency_days:
            from urllib.parse import urlparse
            import time as _t
            cutoff = _t.time() - float(recency_days) * 86400 \
                if recency_days else None

            def _keep(r):
                dom = urlparse(r["url"]).netloc.lower()
                if domain_allowlist and not any(
                        dom == a.lower() or dom.endswith("." + a.lower())
                        for a in domain_allowlist):
                    return False
                if domain_blocklist and any(
                        dom == b.lower() or dom.endswith("." + b.lower())
                        for b in domain_blocklist):
                    return False
                if cutoff is not None:
                    doc = self.ctx.store.get_document_by_url(r["url"])
                    if doc is None or doc.crawled_at < cutoff:
                        return False
                return True
            results = [r for r in results if _keep(r)][:limit]
        # attach full text for RAG modes
        if chunk_size or answer_mode or summarize or fetch_full_content:
            for r in results:
                doc = self.ctx.store.get_document_by_url(r["url"])
                if doc:
                    r["text"] = doc.text[:20_000]
        if rerank and self.reranker is not None:
            from ..search.reranker import rerank_results
            results = rerank_results(
                query, results, self.reranker, keep=limit,
                text_of=lambda r: f"{r.get('title','')} {r.get('snippet','')}")
        out: dict[str, Any] = {
            "query": query,
            "effective_query": resp.effective_query,
            "mode": resp.mode,
            "elapsed_ms": round(resp.elapsed_ms, 2),
            "degraded": resp.degraded,
            "results": results,
        }
        if validate:
            out["validation"] = self._cross_validate(query, results)
        if chunk_size:
            rag = format_rag_output(query, results, chunk_size=chunk_size,
                                    answer_mode=answer_mode,
                                    summarizer=self.summarizer
                                    if summarize else None)
            out["chunks"] = [{"text": c.text, "url": c.url,
                              "title": c.title, "score": round(c.score, 3)}
                             for c in rag.chunks]
            out["entities"] = rag.entities
            if answer_mode:
                out["answer"] = rag.answer
                out["confidence"] = rag.confidence
            if summarize:
                out["summary"] = rag.summary
        elif answer_mode:
            out["answer"], out["confidence"] = extract_answer(query, results)
        elif summarize and self.summarizer is not None:
            out["summary"] = self.summarizer.summarize_results(
                results, query).summary
        if not results:
            vocab = {t.lower(): 1
                     for s in self.ctx.store.suggest(query.split()[0], 20)
                     for t in s.split()}
            suggestion = did_you_mean(query, vocab) if vocab else None
            if suggestion:
                out["did_you_mean"] = suggestion
        return out

    # ------------------------------------------------------------- fetch
    def fetch_page(self, url: str, live: bool = False,
                   max_chars: int = 20_000, **_) -> dict[str, Any]:
        doc = self.ctx.fetch_page(url)
        if doc is None and live and self.ctx.worker is not None:
            res = asyncio.run(self.ctx.crawl_and_index(url))
            if res.get("indexed"):
                doc = self.ctx.fetch_page(url)
        if doc is None:
            return {"url": url, "found": False}
        from ..crawler.parser import is_paywall_content
        return {"url": doc.url, "found": True, "title": doc.title,
                "language": doc.language, "crawled_at": doc.crawled_at,
                "text": doc.text[:max_chars],
                "truncated": len(doc.text) > max_chars,
                "paywalled": is_paywall_content(doc.text)}

    # ------------------------------------------------------------- crawl
    def crawl_url(self, url: str, force: bool = False, depth: int = 0,
                  **_) -> dict[str, Any]:
        now = time.time()
        self._crawl_times = [t for t in self._crawl_times if now - t < 3600]
        if len(self._crawl_times) >= CRAWL_RATE_PER_HOUR:
            raise InfoMeshError("SEC001",
                                f"crawl rate limit {CRAWL_RATE_PER_HOUR}/hr")
        self._crawl_times.append(now)
        # depth flows to the crawl worker, which BFS-schedules the
        # page's links within its own politeness/budget limits
        return asyncio.run(self.ctx.crawl_and_index(url, depth=depth,
                                                    force=force))

    # -------------------------------------------------------- fact check
    def fact_check(self, claim: str, limit: int = 5,
                   top_k: int | None = None, **_) -> dict[str, Any]:
        if top_k:
            limit = int(top_k)
        from ..summarizer.verify import _fact_support  # same scorer
        hits = self.ctx.store.search(claim, limit=limit)
        if not hits:  # recall-first retry: any-term match
            hits = self.ctx.store.search(claim, limit=limit, match_any=True)
        evidence = []
        for r in hits:
            url = getattr(r, "url", "")
            doc = self.ctx.store.get_document_by_url(url)
            if doc is None:
                continue
            score, sent = _fact_support(claim, doc.text[:20_000])
            evidence.append({"url": url, "title": doc.title,
                             "support": round(score, 3), "evidence": sent})
        evidence.sort(key=lambda e: -e["support"])
        supported = bool(evidence) and evidence[0]["support"] >= 0.5
        return {"claim": claim, "supported": supported,
                "confidence": evidence[0]["support"] if evidence else 0.0,
                "evidence": evidence[:limit]}

    # ------------------------------------------------------------ status
    def status(self, **_) -> dict[str, Any]:
        return self.ctx.status()

    # ---------------------------------------------------------- dispatch
    # -------- legacy utility tools (reference mcp/server.py:205-457) ---

    def ping(self, **_) -> dict[str, Any]:
        return {"pong": True, "ts": time.time()}

    def suggest(self, query: str = "", prefix: str = "",
                limit: int = 5, **_) -> dict[str, Any]:
        """Title-prefix completions (reference suggest tool) — related
        searches are appended when the tracker knows the query."""
        p = (prefix or query).strip()
        out = {"suggestions": self.ctx.store.suggest(p, limit=limit)}
        rel = getattr(self.ctx, "related", None)
        if rel is not None and p:
            out["related"] = rel.related(p, limit=limit)
        return out

    def credit_balance(self, **_) -> dict[str, Any]:
        if self.ctx.ledger is None:
            return {"enabled": False}
        return self.ctx.ledger.stats()

    def index_stats(self, **_) -> dict[str, Any]:
        return self.ctx.store.stats()

    def network_stats(self, **_) -> dict[str, Any]:
        """Single-node analogue of the reference's peer stats: the GPU
        shard fabric replaces the libp2p swarm."""
        eng = self.ctx.engine
        out: dict[str, Any] = {"world_size": 1, "shards": []}
        if eng is not None:
            out["world_size"] = eng.fabric.world
            out["shards"] = [{"rank": eng.fabric.rank,
                              "docs": eng.shard.n_docs,
                              "hbm_bytes": eng.shard.hbm_bytes(),
                              "device": str(eng.shard.device)}]
        return out

    def _cross_validate(self, query: str,
                        engine_results: list[dict]) -> dict[str, Any]:
        """Cross-validate the engine plane against the SQLite FTS plane
        (two independent scorers of the same corpus — the intra-node
        analogue of the reference's multi-peer check,
        infomesh/mcp/handlers.py:432-452 + search/cross_validate.py).
        A corrupted GPU shard shows up as wild score deviation or
        snippet disagreement vs the FTS ground truth."""
        from ..search.cross_validate import (SourceResult,
                                             cross_validate_results)
        from ..search.query import search_local
        fts = search_local(self.ctx.store, query, limit=10,
                           enhance_snippets=False)
        sources = {
            "engine": [SourceResult(url=r.get("url", ""),
                                    title=r.get("title", ""),
                                    snippet=r.get("snippet", ""),
                                    score=float(r.get("score", 0.0)))
                       for r in engine_results],
            "fts": [SourceResult(url=h.url, title=h.title,
                                 snippet=h.snippet, score=h.score)
                    for h in fts.results],
        }
        report = cross_validate_results(sources)
        return {
            "n_sources": report.n_sources,
            "n_suspicious": report.n_suspicious,
            "suspicious_urls": report.suspicious_urls,
            "verdicts": {r.url: r.verdict for r in report.results},
        }

    def batch_search(self, queries: list[str] | None = None,
                     limit: int = 10, **_) -> dict[str, Any]:
        from ..search.extended import batch_search as _bs
        queries = list(queries or [])[:20]
        resps = _bs(lambda q: self.web_search(query=q, limit=limit),
                    queries)
        return {"batches": resps}

    def search_history(self, limit: int = 20, query: str = "",
                       **_) -> dict[str, Any]:
        tracker = getattr(self.ctx, "related", None)
        if tracker is None:
            return {"history": []}
        if query:
            return {"related": tracker.related(query, limit=limit)}
        hist = list(getattr(tracker, "_recent", []))[-limit:]
        return {"history": hist}

    def analytics(self, **_) -> dict[str, Any]:
        tracker = getattr(self, "_analytics", None)
        return tracker.report() if tracker is not None else {}

    def register_webhook(self, event: str = "", url: str = "",
                         **_) -> dict[str, Any]:
        reg = getattr(self, "_webhooks", None)
        if reg is None or not event:
            raise InfoMeshError("SRCH001", "webhook registry unavailable "
                                           "or event missing")
        if url:
            def _post(payload, _url=url):  # fired by reg.fire(event,...)
                import json as _json
                import urllib.request
                req = urllib.request.Request(
                    _url, data=_json.dumps(payload).encode(),
                    headers={"Content-Type": "application/json"})
                urllib.request.urlopen(req, timeout=5)
            reg.register(event, _post)
        return {"registered": event}

    def unregister_webhook(self, event: str = "", **_) -> dict[str, Any]:
        reg = getattr(self, "_webhooks", None)
        if reg is None:
            raise InfoMeshError("SRCH001", "webhook registry unavailable")
        n = len(reg._hooks.pop(event, [])) if hasattr(reg, "_hooks") else 0
        return {"unregistered": event, "removed": n}

    def remove_url(self, url: str = "", reason: str = "user-request",
                   **_) -> dict[str, Any]:
        """GDPR-backed removal: records a durable deletion so the URL
        cannot re-enter via crawl/import (reference remove_url)."""
        if not url:
            raise InfoMeshError("SRCH001", "url required")
        self.ctx.deletions.request_deletion(url, reason=reason)
        self.ctx.deletions.enforce()
        removed = not self.ctx.store.get_document_by_url(url)
        return {"url": url, "removed": removed,
                "deletion_recorded": True}

    def call(self, tool: str, args: dict[str, Any]) -> dict[str, Any]:
        from .tools import resolve_tool, validate_args
        resolved = resolve_tool(tool)
        if resolved is None:
            raise InfoMeshError("SRCH001", f"unknown tool {tool!r}")
        violations = validate_args(tool, args)
        if violations:
            raise InfoMeshError("SRCH001",
                                "invalid arguments: " + "; ".join(
                                    violations[:5]))
        # legacy arg adaptation
        if tool == "explain":
            args = {**args, "explain": True}
        elif tool == "search_rag":
            args.setdefault("chunk_size", 512)
        elif tool == "extract_answer":
            args = {**args, "answer_mode": True}
        elif tool in ("search_local", "search_hybrid", "search_distributed"):
            args = {**args, "mode": tool.split("_", 1)[1]}
        elif tool == "verify":
            args.setdefault("claim", args.pop("query", ""))
        fn = getattr(self, resolved)
        try:
            return fn(**args)
        except InfoMeshError as e:
            return {"error": format_error(e), "code": e.code}
