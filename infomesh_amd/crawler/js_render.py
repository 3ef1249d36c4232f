"""JS rendering hook (reference parity: infomesh/crawler/js_render.py —
Playwright headless renderer with tab/memory limits).

Playwright is not installed in the MI355X image; detection
(parser.looks_like_js_app) still runs, and this module exposes the same
surface: available() gates the feature, render() uses Playwright when
present and raises a clear error otherwise."""
from __future__ import annotations

RENDER_TIMEOUT_S = 20.0
MAX_TABS = 2


def available() -> bool:
    try:
        import playwright.sync_api  # noqa: F401
        return True
    except ImportError:
        return False


def render(url: str, timeout_s: float = RENDER_TIMEOUT_S) -> str:
    """Rendered HTML of a JS app page (requires playwright)."""
    if not available():
        raise RuntimeError(
            "JS rendering requires playwright (`pip install playwright && "
            "playwright install chromium`); this deployment indexes the "
            "static HTML of JS apps instead")
    from playwright.sync_api import sync_playwright
    with sync_playwright() as p:
        browser = p.chromium.launch(headless=True)
        try:
            page = browser.new_page()
            page.goto(url, timeout=timeout_s * 1000)
            page.wait_for_load_state("networkidle",
                                     timeout=timeout_s * 1000)
            return page.content()
        finally:
            browser.close()
