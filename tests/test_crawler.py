"""Crawler-layer tests without network (reference parity:
tests/test_crawler.py — parser/dedup/scheduler/simhash/robots/rss)."""
from __future__ import annotations

import asyncio

import httpx

from infomesh_amd.config import CrawlConfig
from infomesh_amd.crawler.dedup import DeduplicatorDB, normalize_url
from infomesh_amd.crawler.lang_detect import detect_language
from infomesh_amd.crawler.parser import extract_content, looks_like_js_app
from infomesh_amd.crawler.robots import RobotsChecker
from infomesh_amd.crawler.rss import Feed, FeedMonitor, parse_feed
from infomesh_amd.crawler.scheduler import Scheduler
from infomesh_amd.crawler.seeds import load_seeds, load_all_seeds
from infomesh_amd.crawler.simhash import (SimHashIndex, hamming_distance,
                                          simhash)
from infomesh_amd.crawler.worker import CrawlWorker

HTML = """
<html><head><title>Test Page — GPU Kernels</title>
<link rel="canonical" href="https://example.com/canonical">
<link rel="alternate" type="application/rss+xml" href="/feed.xml">
<meta name="description" content="A page about GPU kernels.">
<script>var ignored = "not content";</script>
<style>.x{color:red}</style>
</head><body>
<nav>Home | About | Ignored nav text here</nav>
<h1>GPU Kernels on CDNA4</h1>
<p>Writing HIP kernels for the MI355X requires understanding wavefronts,
matrix cores and the LDS. This paragraph is the main content and easily
long enough to be kept by the extractor.</p>
<p>Another paragraph with a <a href="/relative">relative link</a> and an
<a href="https://other.com/page#frag">absolute link</a>.</p>
</body></html>
"""


# ---------------------------------------------------------------- parser

def test_extract_content():
    page = extract_content("https://example.com/a", HTML)
    assert "GPU Kernels" in page.title
    assert "matrix cores" in page.text
    assert "not content" not in page.text
    assert "Ignored nav text" not in page.text
    assert page.canonical == "https://example.com/canonical"
    assert "https://example.com/relative" in page.links
    assert "https://other.com/page" in page.links  # fragment stripped
    assert page.language == "en"
    assert page.feeds == ["https://example.com/feed.xml"]
    assert page.description.startswith("A page about")


def test_js_app_detection():
    spa = "<html><div id=root></div><script>React.render(window.__DATA__)" \
          "</script><script src=webpack.js></script></html>"
    page = extract_content("https://x.com", spa)
    assert looks_like_js_app(spa, page.text)
    assert not looks_like_js_app(HTML, "long text " * 100)


def test_lang_detect():
    assert detect_language("the quick brown fox is in the barn and it was") == "en"
    assert detect_language("der Hund ist nicht mit der Katze und ein Haus") == "de"
    assert detect_language("これは日本語のテキストですかな") == "ja"
    assert detect_language("Это русский текст для проверки") == "ru"
    assert detect_language("8231 9912") == ""


# ----------------------------------------------------------------- dedup

def test_normalize_url():
    assert normalize_url("HTTPS://Example.COM/Path/?b=2&a=1&utm_source=x") == \
        "https://example.com/Path?a=1&b=2"
    assert normalize_url("http://a.com/x/") == "http://a.com/x"
    assert normalize_url("http://a.com/") == "http://a.com/"
    assert normalize_url("http://a.com:8080/x") == "http://a.com:8080/x"
    assert normalize_url("http://a.com:80/x") == "http://a.com/x"


def test_dedup_db_layers(tmp_path):
    db = DeduplicatorDB(tmp_path / "dedup.db")
    assert not db.seen_url("https://a.com/1")
    db.record("https://a.com/1", "some content text here")
    assert db.seen_url("https://a.com/1")
    assert db.seen_url("https://a.com/1/")       # normalization
    assert db.seen_content("some content text here")
    near, fp = db.near_duplicate("some content text here")
    assert near and fp != 0
    db.close()
    # fingerprints reload on boot (reference: dedup.py:130-149)
    db2 = DeduplicatorDB(tmp_path / "dedup.db")
    assert len(db2.simhash_index) == 1
    db2.close()


# --------------------------------------------------------------- simhash

def test_simhash_properties():
    a = simhash("the quick brown fox jumps over the lazy dog " * 5)
    b = simhash("the quick brown fox jumps over the lazy dog " * 5 + "extra")
    c = simhash("completely different content about gpu kernels and lds")
    assert hamming_distance(a, b) <= 10
    assert hamming_distance(a, c) > 10
    idx = SimHashIndex()
    idx.add("k1", a)
    assert idx.find_near(b if hamming_distance(a, b) <= 3 else a)


# ------------------------------------------------------------- scheduler

def test_scheduler_depth_and_domain_caps():
    s = Scheduler(politeness_delay_s=0, max_urls_per_hour=1000, max_depth=2)
    assert s.add_url("https://a.com/1", depth=0)
    assert not s.add_url("https://a.com/x", depth=3)  # too deep
    assert not s.add_url("not a url")
    assert s.qsize() == 1


def test_scheduler_get_respects_politeness():
    async def run():
        s = Scheduler(politeness_delay_s=0.05, max_urls_per_hour=1000)
        s.add_url("https://a.com/1")
        s.add_url("https://a.com/2")
        import time
        t0 = time.perf_counter()
        r1 = await s.get_url(timeout=5)
        r2 = await s.get_url(timeout=5)
        elapsed = time.perf_counter() - t0
        assert r1 and r2
        assert elapsed >= 0.04  # second fetch waited for the delay
    asyncio.run(run())


def test_scheduler_timeout_returns_none():
    async def run():
        s = Scheduler()
        assert await s.get_url(timeout=0.05) is None
    asyncio.run(run())


# ---------------------------------------------------------------- robots

def test_robots_parse_offline():
    rc = RobotsChecker(user_agent="infomesh-amd")
    rc.load_parsed("https://example.com/", """
User-agent: *
Disallow: /private/
Crawl-delay: 2
Sitemap: https://example.com/sitemap.xml
""")
    async def run():
        assert await rc.is_allowed("https://example.com/public")
        assert not await rc.is_allowed("https://example.com/private/x")
        assert await rc.get_crawl_delay("https://example.com/") == 2.0
        assert await rc.get_sitemaps("https://example.com/") == \
            ["https://example.com/sitemap.xml"]
    asyncio.run(run())


# ------------------------------------------------------------------- rss

RSS = """<?xml version="1.0"?><rss version="2.0"><channel>
<title>Blog</title>
<item><title>Post 1</title><link>https://b.com/1</link>
<pubDate>Mon, 01 Jan 2024 00:00:00 GMT</pubDate></item>
<item><title>Post 2</title><link>https://b.com/2</link></item>
</channel></rss>"""

ATOM = """<?xml version="1.0"?><feed xmlns="http://www.w3.org/2005/Atom">
<title>AtomBlog</title>
<entry><title>E1</title><link href="https://c.com/e1"/>
<updated>2024-01-01</updated></entry></feed>"""


def test_parse_rss_and_atom():
    f = parse_feed("https://b.com/feed", RSS)
    assert f.title == "Blog" and len(f.items) == 2
    a = parse_feed("https://c.com/feed", ATOM)
    assert a.title == "AtomBlog" and a.items[0].url == "https://c.com/e1"
    assert parse_feed("https://x.com", "not xml") is None


def test_feed_monitor_tiers():
    m = FeedMonitor()
    m.add("https://b.com/feed", tier=2)
    assert len(m.due()) == 1
    new = m.record_poll("https://b.com/feed",
                        parse_feed("https://b.com/feed", RSS))
    assert len(new) == 2
    assert m.feeds["https://b.com/feed"].tier == 1  # promoted
    new2 = m.record_poll("https://b.com/feed",
                         parse_feed("https://b.com/feed", RSS))
    assert new2 == []
    assert m.feeds["https://b.com/feed"].tier == 2  # demoted


def test_opml_import():
    m = FeedMonitor()
    n = m.import_opml("""<opml><body>
        <outline text="a" xmlUrl="https://a.com/f"/>
        <outline text="b" xmlUrl="https://b.com/f"/></body></opml>""")
    assert n == 2 and len(m.feeds) == 2


# ----------------------------------------------------------------- seeds

def test_load_seeds():
    urls = load_seeds("quickstart")
    assert len(urls) >= 5
    assert all(u.startswith("http") for u in urls)
    assert set(load_all_seeds()) == {"quickstart", "tech-docs", "academic",
                                     "encyclopedia", "search-strategy"}


# ---------------------------------------------------------------- worker

def _mock_transport(pages: dict[str, str | tuple[int, str]]):
    def handler(request: httpx.Request) -> httpx.Response:
        url = str(request.url)
        entry = pages.get(url)
        if entry is None:
            return httpx.Response(404, text="not found")
        if isinstance(entry, tuple):
            status, body = entry
            if 300 <= status < 400:
                return httpx.Response(status, headers={"location": body})
            return httpx.Response(status, text=body)
        return httpx.Response(200, text=entry,
                              headers={"content-type": "text/html"})
    return httpx.MockTransport(handler)


def _worker(pages, **cfg_kw):
    cfg = CrawlConfig(politeness_delay_s=0, respect_robots=False, **cfg_kw)
    client = httpx.AsyncClient(transport=_mock_transport(pages))
    return CrawlWorker(cfg, client=client, resolve_dns=False)


def test_worker_crawl_ok():
    pages = {"https://example.com/a": HTML}
    w = _worker(pages)
    async def run():
        res = await w.crawl_url("https://example.com/a")
        assert res.status == "ok", res.reason
        assert res.page.title.startswith("Test Page")
        assert res.links_scheduled >= 1
        assert res.feeds
        # second crawl skips by url dedup
        res2 = await w.crawl_url("https://example.com/a")
        assert res2.status == "skipped"
        await w.close()
    asyncio.run(run())


def test_worker_ssrf_blocked():
    w = _worker({})
    async def run():
        res = await w.crawl_url("http://127.0.0.1/internal")
        assert res.status == "skipped" and "ssrf" in res.reason
        await w.close()
    asyncio.run(run())


def test_worker_redirect_followed():
    pages = {
        "https://example.com/r": (301, "https://example.com/final"),
        "https://example.com/final": HTML,
    }
    w = _worker(pages)
    async def run():
        res = await w.crawl_url("https://example.com/r")
        assert res.status == "ok", res.reason
        await w.close()
    asyncio.run(run())


def test_worker_content_dedup():
    pages = {
        "https://example.com/a": HTML,
        "https://example.com/b": HTML.replace("/canonical", "/canonical2"),
    }
    w = _worker(pages)
    async def run():
        r1 = await w.crawl_url("https://example.com/a")
        assert r1.status == "ok"
        r2 = await w.crawl_url("https://example.com/b")
        assert r2.status == "skipped"
        assert "duplicate" in r2.reason
        await w.close()
    asyncio.run(run())


def test_worker_http_error_and_thin_page():
    pages = {
        "https://example.com/404": (404, "x"),
        "https://example.com/thin": "<html><p>hi</p></html>",
    }
    w = _worker(pages)
    async def run():
        r = await w.crawl_url("https://example.com/404")
        assert r.status == "error"
        r2 = await w.crawl_url("https://example.com/thin")
        assert r2.status == "skipped" and "text" in r2.reason
        await w.close()
    asyncio.run(run())


def test_worker_5xx_retries():
    calls = {"n": 0}
    def handler(request):
        calls["n"] += 1
        if calls["n"] < 3:
            return httpx.Response(503, text="busy")
        return httpx.Response(200, text=HTML,
                              headers={"content-type": "text/html"})
    cfg = CrawlConfig(politeness_delay_s=0, respect_robots=False, retries=2)
    w = CrawlWorker(cfg, client=httpx.AsyncClient(
        transport=httpx.MockTransport(handler)), retry_backoff_s=(0.0,),
        resolve_dns=False)
    async def run():
        res = await w.crawl_url("https://example.com/flaky")
        assert res.status == "ok"
        assert calls["n"] == 3
        await w.close()
    asyncio.run(run())


def test_fetch_streams_capped(tmp_path):
    """The streaming fetch never holds more than max_response_bytes:
    a 10x-oversized body is truncated at the cap, not fully buffered."""
    import asyncio

    import httpx

    from infomesh_amd.config import CrawlConfig
    from infomesh_amd.crawler.worker import CrawlWorker

    big = "<html><body>" + ("spam " * 40_000) + "</body></html>"

    def handler(request):
        return httpx.Response(200, text=big,
                              headers={"content-type": "text/html"})

    cfg = CrawlConfig(max_response_bytes=16_384, respect_robots=False)
    w = CrawlWorker(cfg, client=httpx.AsyncClient(
        transport=httpx.MockTransport(handler)), resolve_dns=False)
    resp = asyncio.run(w._fetch("https://big.example/x"))
    assert resp.status_code == 200
    assert len(resp.content) == 16_384
    asyncio.run(w.close())
