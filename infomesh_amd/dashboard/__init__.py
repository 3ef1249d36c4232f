"""Dashboard: rich-based live TUI + text report.
Reference parity: infomesh/dashboard/ (Textual app with Overview/Crawl/
Search/Network/Credits/Settings tabs; here rich.Live panels over the
same data sources — runtime_status.json heartbeat + SQLite WAL reads —
plus the plain-text report in utils/text_report.py)."""
