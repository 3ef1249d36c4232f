"""Service orchestration: AppContext wiring + the crawl->index->publish
and search entry paths.

Reference parity: infomesh/services.py (AppContext role-conditional
construction, index_document single source of truth, crawl_and_index,
fetch_page cache-first, create_local_search_fn). The network-publish
path becomes a GPU-engine ingest (SURVEY.md §2.9 last row).
"""
from __future__ import annotations

import logging
import time
from dataclasses import dataclass, field
from typing import Any

from .config import Config, load_config
from .credits.farming import FarmingDetector
from .credits.ledger import Action, CreditLedger
from .crawler.dedup import DeduplicatorDB
from .crawler.robots import RobotsChecker
from .crawler.rss import FeedMonitor
from .crawler.scheduler import Scheduler
from .crawler.worker import CrawlResult, CrawlWorker
from .engine import HybridEngine
from .errors import GpuExtensionMissing, InfoMeshError
from .index.link_graph import LinkGraph
from .index.local_store import Document, LocalStore
from .search.batcher import QueryBatcher
from .search.cache import QueryCache
from .search.nlp import RelatedSearchTracker, parse_query_filters
from .search.passage import fast_snippet
from .search.query import (SearchResponse, preprocess_query,
                           search_hybrid, search_local)
from .trust.attestation import create_attestation
from .trust.dmca import TakedownManager
from .trust.gdpr import DeletionManager
from .trust.keys import KeyPair, ensure_keys
from .trust.scoring import TrustStore

log = logging.getLogger("infomesh.services")


@dataclass
class AppContext:
    """All wired subsystems for one node process.

    Role-conditional (reference services.py:485-567): 'crawler' skips
    the query cache + engine; 'search' skips the crawl worker."""

    config: Config
    store: LocalStore
    link_graph: LinkGraph
    dedup: DeduplicatorDB
    keys: KeyPair
    ledger: CreditLedger
    trust: TrustStore
    takedowns: TakedownManager
    deletions: DeletionManager
    cache: QueryCache
    feeds: FeedMonitor
    related: RelatedSearchTracker
    feedback: object | None = None
    worker: CrawlWorker | None = None
    engine: HybridEngine | None = None
    batcher: QueryBatcher | None = None
    recrawl_queue: Any = None   # crawler.freshness.PriorityRecrawlQueue
    attestations: list = field(default_factory=list)
    farming: FarmingDetector = field(default_factory=FarmingDetector)
    started_at: float = field(default_factory=time.time)
    # resource ladder: crawl loop consults it for backpressure and
    # index_document refuses writes at READ_ONLY (reference
    # governor.py:49-74 semantics)
    governor: Any = None
    otlp: Any = None   # utils.observability.OtlpExporter (off by default)

    # ------------------------------------------------------------ build
    @classmethod
    def create(cls, config: Config | None = None,
               with_engine: bool | None = None,
               with_worker: bool | None = None,
               in_memory: bool = False) -> "AppContext":
        cfg = config or load_config()
        role = cfg.node.role
        data = cfg.data_dir
        if not in_memory:
            data.mkdir(parents=True, exist_ok=True)

        def p(name: str):
            return ":memory:" if in_memory else data / name

        if cfg.node.plugins:
            from .utils.plugins import load_plugins_from_config
            load_plugins_from_config(
                [x.strip() for x in cfg.node.plugins.split(",")
                 if x.strip()])
        store = LocalStore(p("index.db"), tokenizer=cfg.index.fts_tokenizer,
                           max_text_chars=cfg.index.max_text_chars)
        keys = KeyPair.generate() if in_memory else ensure_keys(data)
        ledger = CreditLedger(p("ledger.db"), kp=keys,
                              crawl_reward=cfg.credits.crawl_reward,
                              query_reward=cfg.credits.query_reward,
                              search_cost=cfg.credits.search_cost,
                              grace_hours=cfg.credits.grace_hours)
        trust = TrustStore(p("trust.db"),
                           isolation_failures=cfg.trust.isolation_failures)
        ctx = cls(
            config=cfg,
            store=store,
            link_graph=LinkGraph(p("links.db")),
            dedup=DeduplicatorDB(p("dedup.db")),
            keys=keys,
            ledger=ledger,
            trust=trust,
            takedowns=TakedownManager(store, keys, p("dmca.db")),
            deletions=DeletionManager(store, keys, p("gdpr.db")),
            cache=QueryCache(cfg.search.cache_entries, cfg.search.cache_ttl_s),
            feeds=FeedMonitor(path=None if in_memory else p("feeds.json")),
            related=RelatedSearchTracker(),
        )
        from .search.feedback import FeedbackStore
        ctx.feedback = FeedbackStore(p("feedback.db"))
        from .utils.governor import ResourceGovernor
        ctx.governor = ResourceGovernor()
        if cfg.api.otlp_endpoint:
            from .utils.observability import OtlpExporter
            ctx.otlp = OtlpExporter(endpoint=cfg.api.otlp_endpoint)
        if with_worker if with_worker is not None else role in ("full", "crawler"):
            ctx.worker = CrawlWorker(
                cfg.crawl,
                scheduler=Scheduler(cfg.crawl.politeness_delay_s,
                                    cfg.crawl.max_urls_per_hour,
                                    cfg.crawl.max_depth),
                dedup=ctx.dedup,
                robots=RobotsChecker(cfg.crawl.user_agent))
        want_engine = with_engine if with_engine is not None \
            else role in ("full", "search")
        if want_engine:
            try:
                ctx.engine = HybridEngine(
                    k_per_shard=cfg.gpu.topk_per_shard,
                    emb_dtype="fp8" if cfg.gpu.dtype == "fp8" else "bf16",
                    hbm_budget_gb=cfg.gpu.hbm_budget_gb,
                    embed_max_chars=cfg.index.embed_max_chars,
                    require_extension=cfg.gpu.require_extension)
            except GpuExtensionMissing:
                # gpu.require_extension means FAIL, not degrade: a GPU
                # node silently serving the CPU path is exactly what
                # the knob exists to prevent
                raise
            except Exception as e:
                log.warning("engine unavailable: %s", e)
        if ctx.engine is not None:
            # all entry points share one micro-batching queue so
            # concurrent requests reach the GPU plane as one batch;
            # hydration happens batch-wide inside the executor (one
            # SQLite IN query per batch, fast snippets)
            ctx.batcher = QueryBatcher(
                ctx.engine, max_batch=cfg.search.batch_max,
                max_wait_ms=cfg.search.batch_wait_ms,
                execute=ctx._batch_execute)
        return ctx

    # ------------------------------------------------------------ ingest
    def index_document(self, doc: Document, attest: bool = True,
                       credit: bool = True) -> int | None:
        """THE single crawl->index source of truth
        (reference: services.py:68-110)."""
        from .utils.plugins import GLOBAL_PLUGINS
        if self.governor is not None and not self.governor.writes_allowed():
            raise InfoMeshError(
                "RT002", "node degraded to read-only (resource governor)")
        doc = GLOBAL_PLUGINS.run("pre_index", doc)
        if self.deletions.is_forgotten(doc.url):
            raise InfoMeshError("SEC001", "url under GDPR deletion record")
        if self.takedowns.is_blocked(doc.url):
            raise InfoMeshError("SEC001", "url under DMCA takedown")
        rowid = self.store.add_document(doc)
        if rowid is None:
            return None
        doc.doc_id = rowid
        if self.engine is not None:
            self.engine.add_document(doc)
        if attest:
            # deferred signature: signing eagerly cost ~3 ms/page of
            # pure-python Ed25519 on the hot ingest path for a
            # write-mostly log; signed_attestations() completes them
            # on first serve
            self.attestations.append(create_attestation(
                self.keys, doc.url, doc.raw_hash, doc.text_hash,
                sign=False))
            if len(self.attestations) > 10_000:
                del self.attestations[:5000]
        if credit:
            if self.farming.multiplier() > 0:
                # batched: one signed ledger entry per flush interval
                # (another ~3 ms/page of Ed25519 off the ingest path)
                self.ledger.record_action_async(Action.CRAWL, 1.0)
            self.farming.record("crawl")
        self.cache.invalidate()
        GLOBAL_PLUGINS.run("post_index", doc, rowid=rowid)
        return rowid

    def signed_attestations(self, limit: int = 100) -> list:
        """Serve the newest attestations, completing any deferred
        signatures (reference: attestations published to peers/DHT)."""
        from .trust.attestation import sign_attestation
        out = self.attestations[-limit:]
        for att in out:
            sign_attestation(self.keys, att)
        return out

    async def crawl_and_index(self, url: str, depth: int = 0,
                              force: bool = False) -> dict[str, Any]:
        """Crawl one URL and index it (reference: services.py:354-423)."""
        if self.worker is None:
            raise InfoMeshError("RT001", "no crawl worker in this role")
        from .utils.plugins import GLOBAL_PLUGINS
        url = GLOBAL_PLUGINS.run("pre_crawl", url)
        doc_meta = self.store.get_document_by_url(url)
        res: CrawlResult = await self.worker.crawl_url(
            url, depth=depth, force=force,
            etag=doc_meta.etag if doc_meta else "",
            last_modified=doc_meta.last_modified if doc_meta else "")
        out: dict[str, Any] = {"url": url, "status": res.status,
                               "reason": res.reason}
        if res.not_modified:
            self.store.update_recrawl(url, changed=False)
            return out
        if res.status != "ok" or res.page is None:
            return out
        page = res.page
        self.link_graph.add_links(url, page.links)
        for f in res.feeds:
            self.feeds.add(f)
        doc = Document(url=url, title=page.title, text=page.text,
                       language=page.language, text_hash=page.text_hash,
                       raw_hash=page.raw_html_hash, etag=res.etag,
                       last_modified=res.last_modified)
        rowid = self.index_document(doc)
        if doc_meta is not None:
            self.store.update_recrawl(url, changed=rowid is not None,
                                      etag=res.etag,
                                      last_modified=res.last_modified)
        out.update({"doc_id": rowid, "title": page.title,
                    "links": page.links[:50],
                    "links_scheduled": res.links_scheduled,
                    "indexed": rowid is not None})
        return GLOBAL_PLUGINS.run("post_crawl", out)

    # ------------------------------------------------------------ search
    def search(self, query: str, limit: int | None = None,
               mode: str = "auto", use_cache: bool = True,
               deduct: bool = True) -> SearchResponse:
        """Unified search entry (reference mcp/handlers.py:382 flow):
        cache -> credits -> local/hybrid/distributed -> cache."""
        from .utils.plugins import GLOBAL_PLUGINS
        query = GLOBAL_PLUGINS.run("pre_search", query, mode=mode)
        limit = limit or self.config.search.max_results
        key = QueryCache.make_key(query, limit=limit, mode=mode)
        if use_cache:
            cached = self.cache.get(key)
            if cached is not None:
                return cached
        if deduct and self.config.credits.enabled:
            self.ledger.deduct_search_cost_async()
        self.related.record(query)
        authority = self.link_graph.url_authority
        trust_fn = self.trust.trust_fn()
        engine_ready = (self.engine is not None
                        and self.engine.shard.n_docs > 0)
        # Metadata filters (site:/language/date) live in the SQLite
        # plane only — filtered queries take the FTS path even when the
        # engine is up (reference local_store.py:253-352 filter SQL).
        pq = parse_query_filters(query)
        has_filters = bool(pq.site or pq.after or pq.before or pq.language)
        if mode == "local" or (mode == "auto" and not engine_ready):
            resp = search_local(self.store, query, limit=limit,
                                authority_fn=authority, trust_fn=trust_fn,
                                boost_fn=(self.feedback.url_boost
                                          if self.feedback else None))
        elif mode in ("auto", "hybrid", "distributed"):
            if engine_ready and not has_filters:
                # single-fusion path: the GPU plane RRF-fuses BM25 +
                # dense exactly once; results are hydrated from the
                # LocalStore (round-1 double-fusion fix)
                resp = self._engine_search(query, limit, mode)
            else:
                resp = search_hybrid(
                    self.store, None, query, limit=limit,
                    authority_fn=authority, trust_fn=trust_fn,
                    rrf_k=self.config.search.rrf_k,
                    boost_fn=(self.feedback.url_boost
                              if self.feedback else None))
                if mode == "distributed":
                    resp.mode = "distributed"
        else:
            raise InfoMeshError("SRCH001", f"unknown mode {mode!r}")
        resp = GLOBAL_PLUGINS.run("post_search", resp, query=query)
        if use_cache:
            self.cache.put(key, resp)
        self.ledger.record_action_async(Action.QUERY_SERVED, 1.0)
        return resp

    def ensure_batcher(self) -> QueryBatcher | None:
        """Create the micro-batcher if an engine was attached after
        create() (tests/benches assign ctx.engine directly)."""
        if self.batcher is None and self.engine is not None:
            self.batcher = QueryBatcher(
                self.engine, max_batch=self.config.search.batch_max,
                max_wait_ms=self.config.search.batch_wait_ms,
                execute=self._batch_execute)
        return self.batcher

    def make_auditor(self, fetch_fn):
        """Build the random-audit scheduler with the configured rate
        (trust.audits_per_hour) and quorum size (trust.auditors).
        fetch_fn: async url -> text | None (the node's own fetcher)."""
        from .trust.audit import AuditScheduler
        from .trust.detector import MaliciousNodeDetector
        return AuditScheduler(
            self.store, self.trust, fetch_fn,
            rate_per_hour=self.config.trust.audits_per_hour,
            auditors=self.config.trust.auditors,
            detector=MaliciousNodeDetector())

    def _batch_execute(self, queries: list[str],
                       limit: int) -> list[list]:
        """Batcher executor: ONE plane call + ONE hydration round trip
        for the whole batch (engine hits carry only global doc ids +
        fused scores; url/title/domain/snippet come from LocalStore)."""
        if self.batcher is not None \
                and self.batcher.engine is not self.engine:
            self.batcher.engine = self.engine   # engine was rebound
        per_q = self.engine.search_many(queries, limit=limit)
        gids = [h.doc_id for hits in per_q for h in hits]
        docs = self.store.get_documents(gids)
        out: list[list] = []
        for q, hits in zip(queries, per_q):
            hydrated = []
            for h in hits:
                doc = docs.get(h.doc_id)
                if doc is None:
                    continue
                h.url, h.title = doc.url, doc.title
                h.domain, h.crawled_at = doc.domain, doc.crawled_at
                if doc.text:
                    h.snippet = fast_snippet(q, doc.text)
                hydrated.append(h)
            if self.feedback is not None and hydrated \
                    and self.feedback.has_signals():
                # implicit-feedback boost on the GPU plane too: additive
                # on the fused RRF score, then reorder (cheap: <=limit
                # hits; url_boost is 60 s-cached per url)
                for h in hydrated:
                    h.score += self.feedback.url_boost(h.url)
                hydrated.sort(key=lambda x: x.score, reverse=True)
            out.append(hydrated)
        return out

    def _engine_search(self, query: str, limit: int,
                       mode: str) -> SearchResponse:
        """GPU-plane search via the micro-batcher (fused once on the
        plane, hydrated batch-wide in _batch_execute)."""
        t0 = time.time()
        trace = None
        if self.otlp is not None and self.otlp.enabled:
            from .utils.observability import QueryTrace
            trace = QueryTrace(query)
        eff = preprocess_query(query)
        batcher = self.ensure_batcher()
        if trace is not None:
            with trace.span("engine"):
                hits = (batcher.submit(eff, limit) if batcher is not None
                        else self._batch_execute([eff], limit)[0])
            self.otlp.export(trace)
        elif batcher is not None:
            hits = batcher.submit(eff, limit)
        else:
            hits = self._batch_execute([eff], limit)[0]
        return SearchResponse(
            query=query, effective_query=eff, results=hits,
            elapsed_ms=(time.time() - t0) * 1e3,
            mode="distributed" if mode == "distributed" else "hybrid",
            total_candidates=len(hits),
            degraded=bool(getattr(self.engine.plane.fabric,
                                  "degraded", False)))

    def fetch_page(self, url: str) -> Document | None:
        """Cache-first page fetch (reference: services.py:220-335);
        network fetch is the async crawl path."""
        return self.store.get_document_by_url(url)

    def flush_engine(self) -> int:
        if self.engine is None:
            return 0
        return self.engine.flush()

    # ------------------------------------------------------------- stats
    def status(self) -> dict[str, Any]:
        return {
            "node_id": self.keys.node_id,
            "role": self.config.node.role,
            "uptime_s": round(time.time() - self.started_at, 1),
            "index": self.store.stats(),
            "engine": self.engine.stats() if self.engine else None,
            "credits": self.ledger.stats(),
            "cache": self.cache.stats(),
            "batcher": self.batcher.stats() if self.batcher else None,
            "crawler": self.worker.stats if self.worker else None,
            "link_edges": self.link_graph.edge_count(),
        }

    def close(self) -> None:
        try:
            self.feeds.save()
        except Exception:
            pass
        if self.batcher is not None:
            self.batcher.close()
        for c in (self.store, self.link_graph, self.dedup, self.ledger,
                  self.trust, self.takedowns, self.deletions,
                  self.feedback):
            try:
                c.close()
            except Exception:
                pass


