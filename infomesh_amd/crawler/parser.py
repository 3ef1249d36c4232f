"""HTML content extraction (no trafilatura in this image — an
html.parser-based extractor with the same contract).
Reference parity: infomesh/crawler/parser.py (ParsedPage, extract_content,
extract_links, extract_canonical).
"""
from __future__ import annotations

import re
from dataclasses import dataclass, field
from html.parser import HTMLParser
from urllib.parse import urljoin, urlparse

from ..hashing import content_hash
from .lang_detect import detect_language

_BLOCK_TAGS = frozenset(
    "p div article section main li h1 h2 h3 h4 h5 h6 td th blockquote pre "
    "figcaption summary dd dt".split())
_SKIP_TAGS = frozenset("script style noscript svg template iframe "
                       "nav footer aside form button".split())
_JS_SIGNALS = ("react", "angular", "vue", "__next_data__", "window.__",
               "require.js", "webpack")


@dataclass
class ParsedPage:
    url: str
    title: str = ""
    text: str = ""
    language: str = ""
    raw_html_hash: str = ""
    text_hash: str = ""
    links: list[str] = field(default_factory=list)
    canonical: str | None = None
    description: str = ""
    feeds: list[str] = field(default_factory=list)


class _Extractor(HTMLParser):
    def __init__(self, base_url: str):
        super().__init__(convert_charrefs=True)
        self.base = base_url
        self.title_parts: list[str] = []
        self.chunks: list[str] = []
        self.links: list[str] = []
        self.canonical: str | None = None
        self.description = ""
        self.feeds: list[str] = []
        self._skip_depth = 0
        self._in_title = False
        self._buf: list[str] = []

    def handle_starttag(self, tag, attrs):
        if tag in _SKIP_TAGS:
            self._skip_depth += 1
            return
        ad = dict(attrs)
        if tag == "title":
            self._in_title = True
        elif tag == "a" and ad.get("href"):
            try:
                link = urljoin(self.base, ad["href"].strip())
            except ValueError:
                return
            if link.startswith(("http://", "https://")):
                self.links.append(link.split("#", 1)[0])
        elif tag == "link":
            rel = (ad.get("rel") or "").lower()
            if "canonical" in rel and ad.get("href"):
                try:
                    self.canonical = urljoin(self.base, ad["href"])
                except ValueError:
                    pass
            if "alternate" in rel and "rss" in (ad.get("type") or "") \
                    or "alternate" in rel and "atom" in (ad.get("type") or ""):
                if ad.get("href"):
                    try:
                        self.feeds.append(urljoin(self.base, ad["href"]))
                    except ValueError:
                        pass
        elif tag == "meta":
            if (ad.get("name") or "").lower() == "description":
                self.description = ad.get("content", "")[:500]
        if tag in _BLOCK_TAGS:
            self._flush()

    def handle_endtag(self, tag):
        if tag in _SKIP_TAGS and self._skip_depth > 0:
            self._skip_depth -= 1
            return
        if tag == "title":
            self._in_title = False
        if tag in _BLOCK_TAGS:
            self._flush()

    def handle_data(self, data):
        if self._skip_depth:
            return
        if self._in_title:
            self.title_parts.append(data)
        else:
            self._buf.append(data)

    def _flush(self):
        text = " ".join("".join(self._buf).split())
        self._buf.clear()
        if len(text) >= 2:
            self.chunks.append(text)


_PAYWALL_SIGNALS = (
    "subscribe to continue", "subscription required", "sign in to read",
    "create a free account", "this content is for subscribers",
    "to continue reading", "register to continue", "metered paywall",
    "already a subscriber", "unlock this article", "premium content",
    "paywall",
)


def is_paywall_content(text: str) -> bool:
    """Heuristic paywall detection for fetched pages (reference:
    services.py is_paywall_content): a short extraction carrying
    subscription-wall phrasing. Total on any input."""
    if not text:
        return False
    t = text.lower()
    hits = sum(1 for s in _PAYWALL_SIGNALS if s in t)
    if hits == 0:
        return False
    # a long article that merely MENTIONS subscriptions is fine; a
    # short stub with wall phrasing is the signal
    return len(text) < 2500 or hits >= 2


def looks_like_js_app(html: str, text: str) -> bool:
    """SPA detection: tiny extracted text + JS framework signals
    (reference: crawler/js_detect.py:88-148, simplified)."""
    if len(text) > 500:
        return False
    low = html[:20000].lower()
    return sum(1 for s in _JS_SIGNALS if s in low) >= 2


def extract_content(url: str, html: str, max_chars: int = 500_000
                    ) -> ParsedPage:
    page = ParsedPage(url=url, raw_html_hash=content_hash(html))
    ex = _Extractor(url)
    try:
        ex.feed(html)
        ex.close()
    except Exception:
        pass
    ex._flush()
    page.title = " ".join("".join(ex.title_parts).split())[:300]
    # Keep substantial chunks; drop boilerplate-ish tiny fragments.
    body = [c for c in ex.chunks if len(c) >= 30 or len(c.split()) >= 5]
    page.text = "\n".join(body)[:max_chars]
    if not page.text and ex.chunks:
        page.text = "\n".join(ex.chunks)[:max_chars]
    page.text_hash = content_hash(page.text)
    page.language = detect_language(page.text or page.title)
    seen: set[str] = set()
    links = []
    for l in ex.links:
        if l not in seen and l != url:
            seen.add(l)
            links.append(l)
    page.links = links[:500]
    page.canonical = ex.canonical
    page.description = ex.description
    page.feeds = ex.feeds[:10]
    return page


def extract_links(url: str, html: str) -> list[str]:
    return extract_content(url, html).links


_DOMAIN_RE = re.compile(r"^[a-z0-9.-]+$")


def same_domain(a: str, b: str) -> bool:
    try:
        return (urlparse(a).hostname or "") == (urlparse(b).hostname or "")
    except ValueError:
        return False
