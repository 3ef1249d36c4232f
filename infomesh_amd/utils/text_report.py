"""Text dashboard report (reference parity: infomesh/dashboard/
text_report.py — the `--text` rich report; the full TUI is the
interactive layer over the same data)."""
from __future__ import annotations

import time


def _bar(frac: float, width: int = 24) -> str:
    frac = max(0.0, min(1.0, frac))
    n = int(frac * width)
    return "█" * n + "░" * (width - n)


def render_report(ctx) -> str:
    st = ctx.status()
    lines = [
        "╔══════════════════════════════════════════════════╗",
        "║            infomesh-amd node report              ║",
        "╚══════════════════════════════════════════════════╝",
        f"node        {st['node_id'][:16]}  role={st['role']}",
        f"uptime      {st['uptime_s']:.0f} s",
        "",
        "— index —",
        f"documents   {st['index']['documents']}",
        f"domains     {st['index']['domains']}",
        f"db size     {st['index']['db_bytes'] / 1e6:.1f} MB",
        f"link edges  {st['link_edges']}",
    ]
    eng = st.get("engine")
    if eng:
        lines += [
            "",
            "— GPU engine —",
            f"device      {eng['device']}  world={eng['world_size']}",
            f"indexed     {eng['docs_indexed']}  pending={eng['docs_pending']}",
            f"HBM         {eng['hbm_bytes'] / 1e9:.2f} GB",
            f"encoder     {'yes' if eng['encoder'] else 'no'}",
        ]
    cr = st["credits"]
    lines += [
        "",
        "— credits —",
        f"balance     {cr['balance']:.2f}  tier={cr['tier']}"
        f"  cost/search={cr['search_cost']:.3f}"
        + ("  [DEBT MODE]" if cr.get("debt_mode") else ""),
    ]
    ca = st["cache"]
    total = max(1, ca["hits"] + ca["misses"])
    lines += [
        "",
        "— query cache —",
        f"entries     {ca['entries']}/{ca['max_entries']}",
        f"hit rate    {_bar(ca['hits'] / total)} {ca['hits'] / total:.0%}",
    ]
    if st.get("crawler"):
        c = st["crawler"]
        lines += [
            "",
            "— crawler —",
            f"crawled     {c['crawled']}  skipped={c['skipped']}"
            f"  errors={c['errors']}",
        ]
    lines.append("")
    lines.append(time.strftime("generated %Y-%m-%d %H:%M:%S"))
    return "\n".join(lines)
