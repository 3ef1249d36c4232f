"""MCP (Model Context Protocol) entry point: 5 consolidated tools +
legacy aliases over stdio and streamable-HTTP transports.
Reference parity: infomesh/mcp/ (SURVEY.md §2.8, §3.2)."""
