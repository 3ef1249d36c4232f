"""CLI entry point (click group).

Reference parity: infomesh/cli/ (start/stop/_serve/status, crawl, mcp,
search/feedback, index export/import/stats, config, keys, peer->shard,
doctor, bench). Console entry: `python -m infomesh_amd` or the
`infomesh-amd` script from setup.py.
"""
from __future__ import annotations

import asyncio
import dataclasses
import json
import os
import signal
import subprocess
import sys
import time
from pathlib import Path

import click

from ..config import load_config, save_config
from ..runtime import GracefulShutdown, PidFile, RuntimeStatus, StartupLock


@click.group()
@click.version_option(package_name=None, prog_name="infomesh-amd",
                      version=__import__("infomesh_amd").__version__)
def cli():
    """infomesh-amd — MI355X-native hybrid search/RAG engine."""


def _ctx(with_engine: bool | None = None, with_worker: bool | None = None):
    from ..services import AppContext
    return AppContext.create(with_engine=with_engine,
                             with_worker=with_worker)


# ----------------------------------------------------------------- serve

@cli.command()
@click.option("--foreground", is_flag=True, help="run in this process")
@click.option("--seed-category", default="quickstart")
def start(foreground: bool, seed_category: str):
    """Start the node (crawl loop + heartbeat)."""
    cfg = load_config()
    data = cfg.data_dir
    data.mkdir(parents=True, exist_ok=True)
    if foreground:
        _serve_impl(seed_category)
        return
    pid = PidFile(data).read_running_pid()
    if pid:
        click.echo(f"already running (pid {pid})")
        return
    proc = subprocess.Popen(
        [sys.executable, "-m", "infomesh_amd", "_serve",
         "--seed-category", seed_category],
        stdout=subprocess.DEVNULL, stderr=open(data / "serve.log", "ab"),
        start_new_session=True)
    click.echo(f"started (pid {proc.pid}); logs: {data / 'serve.log'}")


@cli.command(name="_serve", hidden=True)
@click.option("--seed-category", default="quickstart")
def _serve(seed_category: str):
    _serve_impl(seed_category)


def _serve_impl(seed_category: str):
    from ..crawler.crawl_loop import seed_and_crawl_loop
    cfg = load_config()
    data = cfg.data_dir
    lock = StartupLock(data)
    if not lock.acquire():
        click.echo("another instance is starting", err=True)
        sys.exit(1)
    pidfile = PidFile(data)
    pidfile.acquire()
    status = RuntimeStatus(data)
    shutdown = GracefulShutdown()
    shutdown.install()
    ctx = _ctx()
    try:
        async def run():
            async def heartbeat():
                while not shutdown.requested:
                    status.write("running", **{
                        "docs": ctx.store.count(),
                        "engine_docs": ctx.engine.shard.n_docs
                        if ctx.engine else 0})
                    await asyncio.sleep(10)
            hb = asyncio.ensure_future(heartbeat())
            await seed_and_crawl_loop(
                ctx, seed_category=seed_category,
                stop_check=lambda: shutdown.requested)
            hb.cancel()
        asyncio.run(run())
    finally:
        status.write("stopped")
        pidfile.release()
        lock.release()
        ctx.close()


@cli.command()
def stop():
    """Stop a running node (SIGTERM)."""
    cfg = load_config()
    pid = PidFile(cfg.data_dir).read_running_pid()
    if not pid:
        click.echo("not running")
        return
    os.kill(pid, signal.SIGTERM)
    for _ in range(50):
        if PidFile(cfg.data_dir).read_running_pid() is None:
            click.echo("stopped")
            return
        time.sleep(0.2)
    click.echo("still stopping…")


@cli.command()
def status():
    """Show node status."""
    cfg = load_config()
    rs = RuntimeStatus(cfg.data_dir).read()
    click.echo(json.dumps(rs, indent=2))


# ---------------------------------------------------------------- search

@cli.command()
@click.argument("query", nargs=-1, required=True)
@click.option("--limit", default=10)
@click.option("--mode", default="auto")
@click.option("--json", "as_json", is_flag=True)
@click.option("--explain", is_flag=True)
def search(query, limit, mode, as_json, explain):
    """Search the local index."""
    from ..search.explain import explain_search, render_explanation
    from ..search.formatter import format_json, format_text
    q = " ".join(query)
    ctx = _ctx(with_worker=False)
    try:
        if explain:
            click.echo(render_explanation(explain_search(
                ctx.store, q, limit,
                boost_fn=ctx.feedback.url_boost if ctx.feedback else None)))
            return
        resp = ctx.search(q, limit=limit, mode=mode)
        click.echo(format_json(resp) if as_json else format_text(resp))
    finally:
        ctx.close()


@cli.group()
def feedback():
    """Implicit relevance feedback (record / stats / top-urls)."""


@feedback.command(name="record")
@click.argument("url")
@click.option("--signal", "signal_", default="fetch",
              type=click.Choice(["fetch", "cite", "click", "skip"]))
def feedback_record(url, signal_):
    """Record implicit feedback for a URL."""
    from ..search.feedback import FeedbackStore
    cfg = load_config()
    fs = FeedbackStore(cfg.data_dir / "feedback.db")
    fs.record(url, signal_)
    fs.close()
    click.echo("recorded")


@feedback.command(name="stats")
def feedback_stats():
    """Aggregate feedback counters (reference: cli feedback stats)."""
    from ..search.feedback import FeedbackStore
    fs = FeedbackStore(load_config().data_dir / "feedback.db")
    try:
        click.echo(json.dumps(fs.stats(), indent=2))
    finally:
        fs.close()


@feedback.command(name="top-urls")
@click.option("--limit", default=10)
def feedback_top_urls(limit):
    """URLs with the strongest positive feedback boost."""
    from ..search.feedback import FeedbackStore
    fs = FeedbackStore(load_config().data_dir / "feedback.db")
    try:
        for url, boost in fs.top_urls(limit):
            click.echo(f"{boost:+.3f}  {url}")
    finally:
        fs.close()


# ----------------------------------------------------------------- crawl

@cli.command()
@click.argument("url")
@click.option("--force", is_flag=True)
def crawl(url, force):
    """Crawl and index one URL."""
    ctx = _ctx(with_engine=False)
    try:
        out = asyncio.run(ctx.crawl_and_index(url, force=force))
        click.echo(json.dumps(out, indent=2, default=str))
    finally:
        ctx.close()


@cli.command()
@click.option("--http", is_flag=True, help="streamable-HTTP instead of stdio")
@click.option("--port", default=8765)
@click.option("--api-key", default="")
def mcp(http, port, api_key):
    """Run the MCP server (stdio by default)."""
    from ..mcp.server import run_mcp_http_server, run_mcp_server
    if http:
        run_mcp_http_server(host="127.0.0.1", port=port, api_key=api_key)
    else:
        run_mcp_server(api_key=api_key)


@cli.command()
@click.option("--port", default=8080)
@click.option("--api-key", default="")
def api(port, api_key):
    """Run the local admin HTTP API."""
    from ..api.local_api import run_api
    run_api(port=port, api_key=api_key)


@cli.command()
@click.option("--text", is_flag=True, help="one-shot text report")
@click.option("--refresh", default=2.0, help="live refresh seconds")
def dashboard(text, refresh):
    """Live terminal dashboard (--text for a one-shot report)."""
    if text:
        from ..utils.text_report import render_report
        ctx = _ctx(with_worker=False, with_engine=False)
        try:
            click.echo(render_report(ctx))
        finally:
            ctx.close()
        return
    from ..dashboard.tabs import run_tabbed_dashboard
    run_tabbed_dashboard(refresh_s=refresh)


# ----------------------------------------------------------------- index

@cli.group()
def index():
    """Index snapshot/stat operations."""


@index.command()
@click.argument("path", type=click.Path())
@click.option("--max-docs", default=100_000)
def export(path, max_docs):
    """Export the index to a .infomesh-snapshot file."""
    from ..index.snapshot import export_snapshot
    ctx = _ctx(with_worker=False, with_engine=False)
    try:
        header = export_snapshot(
            ctx.store, path, max_docs=max_docs,
            node_name=ctx.keys.node_id[:12],
            level=ctx.config.index.snapshot_compression_level)
        click.echo(json.dumps(header, indent=2))
    finally:
        ctx.close()


@index.command(name="import-urls")
@click.argument("url_file", type=click.Path(exists=True))
@click.option("--max-urls", default=1000)
def index_import_urls(url_file, max_urls):
    """Queue a file of URLs (one per line) for crawling."""
    import asyncio
    urls = [ln.strip() for ln in open(url_file)
            if ln.strip() and not ln.startswith("#")][:max_urls]
    ctx = _ctx(with_engine=False)
    try:
        n = 0
        for u in urls:
            try:
                res = asyncio.run(ctx.crawl_and_index(u))
                n += 1 if res.get("status") == "ok" else 0
            except Exception as e:
                click.echo(f"skip {u}: {e}", err=True)
        click.echo(f"crawled+indexed {n}/{len(urls)}")
    finally:
        ctx.close()


@index.command(name="import-wet")
@click.argument("path", type=click.Path(exists=True))
@click.option("--max-records", default=None, type=int)
def index_import_wet(path, max_records):
    """Import a Common Crawl WET file (reference: index import-wet)."""
    from ..index.commoncrawl import CommonCrawlImporter
    ctx = _ctx(with_engine=False, with_worker=False)
    try:
        imp = CommonCrawlImporter(ctx.store)
        res = imp.import_wet(path, max_records=max_records)
        click.echo(json.dumps(res, indent=2))
    finally:
        ctx.close()


@index.command(name="import")
@click.argument("path", type=click.Path(exists=True))
def import_(path):
    """Import a .infomesh-snapshot file."""
    from ..index.snapshot import import_snapshot
    ctx = _ctx(with_worker=False)
    try:
        hook = (ctx.engine.add_document if ctx.engine is not None else None)
        res = import_snapshot(ctx.store, path, on_document=hook)
        if ctx.engine is not None and ctx.engine.pending_count:
            ctx.flush_engine()
        click.echo(json.dumps({"imported": res["imported"],
                               "skipped": res["skipped"]}, indent=2))
    finally:
        ctx.close()


@index.command()
def stats():
    """Index statistics."""
    ctx = _ctx(with_worker=False, with_engine=False)
    try:
        click.echo(json.dumps(ctx.store.stats(), indent=2))
    finally:
        ctx.close()


# ---------------------------------------------------------------- config

@cli.group()
def config():
    """Configuration operations."""


@config.command(name="show")
def config_show():
    click.echo(json.dumps(dataclasses.asdict(load_config()), indent=2))


@config.command(name="github")
@click.option("--email", default=None,
              help="owner email for cross-node credit aggregation")
def config_github(email):
    """Set/show the GitHub owner identity (reference: config github).
    Falls back to the local git config email; only the SHA-256 hash is
    ever shared with peers."""
    from ..credits.github_identity import ensure_owner_identity
    chosen = ensure_owner_identity(load_config().data_dir, email)
    if chosen is None:
        click.echo("no valid email (pass --email or set git config user.email)")
        raise SystemExit(1)
    click.echo(f"owner identity: {chosen}")


@config.command(name="set")
@click.argument("key")     # section.key
@click.argument("value")
def config_set(key, value):
    cfg = load_config()
    try:
        section_name, field_name = key.split(".", 1)
        section = getattr(cfg, section_name)
        cur = getattr(section, field_name)
    except (ValueError, AttributeError):
        raise click.ClickException(f"unknown key {key!r}")
    from ..config import _coerce
    new_section = dataclasses.replace(section,
                                      **{field_name: _coerce(cur, value)})
    cfg = dataclasses.replace(cfg, **{section_name: new_section})
    path = save_config(cfg)
    click.echo(f"wrote {path}")


# ------------------------------------------------------------------ keys

@cli.group()
def keys():
    """Node identity keys."""


@keys.command(name="show")
def keys_show():
    from ..trust.keys import ensure_keys
    kp = ensure_keys(load_config().data_dir)
    click.echo(json.dumps({"node_id": kp.node_id,
                           "public_key": kp.public.hex()}, indent=2))


@keys.command(name="export")
def keys_export():
    """Export the PUBLIC key material (never the private key)."""
    from ..trust.keys import ensure_keys
    kp = ensure_keys(load_config().data_dir)
    click.echo(json.dumps({"node_id": kp.node_id,
                           "public_key": kp.public.hex(),
                           "format": "ed25519-hex"}, indent=2))


@keys.command(name="rotate")
def keys_rotate():
    from ..trust.keys import ensure_keys, rotate_keys
    data = load_config().data_dir
    old = ensure_keys(data)
    new, record = rotate_keys(data, old)
    click.echo(json.dumps({"new_node_id": new.node_id,
                           "rotation": record}, indent=2))


# ----------------------------------------------------------------- shard

@cli.group()
def shard():
    """GPU shard manifests (warm-start persistence)."""


@shard.command(name="save")
@click.argument("path", type=click.Path())
def shard_save(path):
    """Build the GPU shard from the local index and save its manifest."""
    ctx = _ctx(with_worker=False)
    try:
        if ctx.engine is None:
            raise click.ClickException("engine unavailable on this machine")
        for doc in ctx.store.export_documents():
            ctx.engine.add_document(doc)
        ctx.flush_engine()
        from ..index.manifest import save_shard
        meta = save_shard(ctx.engine.shard, path)
        click.echo(json.dumps(meta, indent=2))
    finally:
        ctx.close()


@shard.command(name="load")
@click.argument("path", type=click.Path(exists=True))
@click.option("--query", default="", help="probe query after loading")
def shard_load(path, query):
    """Load a shard manifest and optionally probe it."""
    from ..index.manifest import load_shard
    import torch
    device = "cuda" if torch.cuda.is_available() else "cpu"
    s = load_shard(path, device=device)
    out = {"n_docs": s.n_docs, "hbm_bytes": s.hbm_bytes(),
           "device": device}
    if query:
        from ..index.gpu_index import bm25_term_ids
        hits = s.search([bm25_term_ids(query)], None, k=5)
        out["probe"] = [int(i) for i in hits.bm25_ids[0] if int(i) >= 0]
    click.echo(json.dumps(out, indent=2))


@shard.command(name="info")
@click.argument("path", type=click.Path(exists=True))
def shard_info(path):
    from ..index.manifest import manifest_info
    info = manifest_info(path)
    if info is None:
        raise click.ClickException("no manifest sidecar found")
    click.echo(json.dumps(info, indent=2))


@cli.command(name="update-check")
def update_check():
    """Check for a newer release (offline-first)."""
    from ..utils.version_check import check_for_update
    cfg = load_config()
    click.echo(json.dumps(check_for_update(
        state_path=cfg.data_dir / "version_check.json"), indent=2))


# ---------------------------------------------------------------- doctor

@cli.command()
def doctor():
    """Run diagnostics checks."""
    from ..utils.diagnostics import run_doctor
    report = run_doctor()
    for check in report["checks"]:
        mark = "✓" if check["ok"] else "✗"
        click.echo(f"{mark} {check['name']}: {check['detail']}")
    sys.exit(0 if report["ok"] else 1)


@cli.command()
@click.option("--iterations", default=200)
def bench(iterations):
    """Run the micro-benchmark suite (CPU pipeline pieces)."""
    from ..utils.benchmarks import run_micro_suite
    click.echo(json.dumps(run_micro_suite(iterations), indent=2))


def main():
    cli()


if __name__ == "__main__":
    main()


# ----------------------------------------------------------------- feeds

@cli.group()
def feeds():
    """Manage RSS/Atom feed monitoring (persistent across restarts)."""


@feeds.command(name="list")
def feeds_list():
    ctx = _ctx(with_worker=False, with_engine=False)
    try:
        rows = [{"url": f.url, "tier": f.tier, "failures": f.failures}
                for f in ctx.feeds.feeds.values()]
        click.echo(json.dumps(rows, indent=2))
    finally:
        ctx.close()


@feeds.command(name="add")
@click.argument("url")
@click.option("--tier", default=3, help="poll tier 0 (1 min) .. 3 (60 min)")
def feeds_add(url, tier):
    ctx = _ctx(with_worker=False, with_engine=False)
    try:
        ctx.feeds.add(url, tier=tier)
        click.echo(f"added {url} (tier {tier})")
    finally:
        ctx.close()


@feeds.command(name="remove")
@click.argument("url")
def feeds_remove(url):
    ctx = _ctx(with_worker=False, with_engine=False)
    try:
        ctx.feeds.remove(url)
        click.echo(f"removed {url}")
    finally:
        ctx.close()


@feeds.command(name="import")
@click.argument("opml_file", type=click.Path(exists=True))
def feeds_import(opml_file):
    """Import RSS/Atom feeds from an OPML file."""
    ctx = _ctx(with_worker=False, with_engine=False)
    try:
        n = ctx.feeds.import_opml(Path(opml_file).read_text())
        ctx.feeds.save()
        click.echo(f"imported {n} feeds")
    finally:
        ctx.close()
