"""MCP server: JSON-RPC 2.0 over stdio + streamable-HTTP transport.

Reference parity: infomesh/mcp/server.py (stdio + Streamable-HTTP
transports, tool dispatch with legacy aliases, query cache wiring,
API-key check, analytics). The protocol surface implements MCP
`initialize`, `tools/list`, `tools/call`, `resources/list`,
`resources/read`, `prompts/list`, `prompts/get`, `ping`.
"""
from __future__ import annotations

import hmac
import json
import sys
import time
from typing import Any, TextIO

from ..services import AppContext
from .handlers import Handlers
from .session import AnalyticsTracker, SessionStore, WebhookRegistry
from .tools import PROMPTS, RESOURCES, TOOLS

# hard cap on one HTTP request body — a JSON-RPC tool call is KBs;
# anything larger is abuse (matches the reference's 10 MB wire cap,
# infomesh/p2p/protocol.py:358-369 spirit)
MAX_BODY_BYTES = 8 * 1024 * 1024

PROTOCOL_VERSION = "2024-11-05"
SERVER_INFO = {"name": "infomesh-amd", "version": "0.1.0"}


class McpServer:
    def __init__(self, ctx: AppContext, api_key: str = "",
                 reranker=None, summarizer=None):
        self.ctx = ctx
        self.api_key = api_key
        self.handlers = Handlers(ctx, reranker=reranker,
                                 summarizer=summarizer)
        self.sessions = SessionStore()
        self.analytics = AnalyticsTracker()
        self.webhooks = WebhookRegistry()
        # legacy analytics/webhook tools reach these through the handlers
        self.handlers._analytics = self.analytics
        self.handlers._webhooks = self.webhooks

    # ------------------------------------------------------- JSON-RPC
    def handle_message(self, msg: dict[str, Any]) -> dict[str, Any] | None:
        mid = msg.get("id")
        method = msg.get("method", "")
        params = msg.get("params") or {}
        try:
            if method == "initialize":
                result = {
                    "protocolVersion": PROTOCOL_VERSION,
                    "serverInfo": SERVER_INFO,
                    "capabilities": {"tools": {}, "resources": {},
                                     "prompts": {}},
                }
            elif method == "notifications/initialized":
                return None
            elif method == "ping":
                result = {}
            elif method == "tools/list":
                result = {"tools": TOOLS}
            elif method == "tools/call":
                result = self._call_tool(params)
            elif method == "resources/list":
                result = {"resources": RESOURCES}
            elif method == "resources/templates/list":
                result = {"resourceTemplates": [
                    {"uriTemplate": "infomesh://doc/{url}",
                     "name": "document",
                     "description": "An indexed document by URL",
                     "mimeType": "application/json"},
                ]}
            elif method == "resources/read":
                result = self._read_resource(params)
            elif method == "prompts/list":
                result = {"prompts": [
                    {k: v for k, v in p.items() if k != "template"}
                    for p in PROMPTS]}
            elif method == "prompts/get":
                result = self._get_prompt(params)
            else:
                return self._error(mid, -32601, f"method {method!r} not found")
        except Exception as e:  # tool errors -> JSON-RPC error
            return self._error(mid, -32000, str(e))
        if mid is None:
            return None
        return {"jsonrpc": "2.0", "id": mid, "result": result}

    def _read_resource(self, params: dict[str, Any]) -> dict[str, Any]:
        import json as _json
        uri = params.get("uri", "")
        if uri == "infomesh://index/stats":
            body = _json.dumps(self.ctx.store.stats(), default=str)
        elif uri == "infomesh://node/status":
            from ..runtime import RuntimeStatus
            body = _json.dumps(
                RuntimeStatus(self.ctx.config.data_dir).read(),
                default=str)
        elif uri == "infomesh://credits/balance":
            body = _json.dumps(
                self.ctx.ledger.summary() if self.ctx.ledger else {},
                default=str)
        elif uri.startswith("infomesh://doc/"):
            doc = self.ctx.store.get_document_by_url(
                uri[len("infomesh://doc/"):])
            if doc is None:
                raise ValueError(f"no document for {uri!r}")
            body = _json.dumps({"url": doc.url, "title": doc.title,
                                "text": doc.text[:20000]}, default=str)
        else:
            raise ValueError(f"unknown resource {uri!r}")
        return {"contents": [{"uri": uri, "mimeType": "application/json",
                              "text": body}]}

    def _get_prompt(self, params: dict[str, Any]) -> dict[str, Any]:
        name = params.get("name", "")
        args = params.get("arguments") or {}
        for p in PROMPTS:
            if p["name"] == name:
                text = p["template"].format(**{
                    a["name"]: args.get(a["name"], "")
                    for a in p.get("arguments", [])})
                return {"description": p["description"],
                        "messages": [{"role": "user", "content": {
                            "type": "text", "text": text}}]}
        raise ValueError(f"unknown prompt {name!r}")

    def _call_tool(self, params: dict[str, Any]) -> dict[str, Any]:
        name = params.get("name", "")
        args = params.get("arguments") or {}
        t0 = time.perf_counter()
        error = False
        try:
            out = self.handlers.call(name, args)
            error = "error" in out
        except Exception:
            error = True
            raise
        finally:
            self.analytics.record(name, (time.perf_counter() - t0) * 1e3,
                                  error)
        return {
            "content": [{"type": "text",
                         "text": json.dumps(out, ensure_ascii=False,
                                            default=str)}],
            "isError": error,
        }

    @staticmethod
    def _error(mid, code: int, message: str) -> dict[str, Any]:
        return {"jsonrpc": "2.0", "id": mid,
                "error": {"code": code, "message": message}}

    # ---------------------------------------------------------- stdio
    def run_stdio(self, stdin: TextIO | None = None,
                  stdout: TextIO | None = None) -> None:
        stdin = stdin or sys.stdin
        stdout = stdout or sys.stdout
        for line in stdin:
            line = line.strip()
            if not line:
                continue
            try:
                msg = json.loads(line)
            except json.JSONDecodeError:
                continue
            resp = self.handle_message(msg)
            if resp is not None:
                stdout.write(json.dumps(resp) + "\n")
                stdout.flush()

    # ----------------------------------------------------------- HTTP
    def asgi_app(self):
        """Streamable-HTTP transport: POST /mcp with a JSON-RPC body."""
        server = self

        async def app(scope, receive, send):
            if scope["type"] != "http":
                return
            headers = {k.decode(): v.decode()
                       for k, v in scope.get("headers", [])}
            if server.api_key and not hmac.compare_digest(
                    headers.get("x-api-key", ""), server.api_key):
                await _respond(send, 401, {"error": "bad api key"})
                return
            if scope["method"] == "POST" and scope["path"] in ("/mcp", "/"):
                body = b""
                too_big = False
                while True:
                    ev = await receive()
                    body += ev.get("body", b"")
                    if len(body) > MAX_BODY_BYTES:
                        too_big = True   # keep draining to not stall ASGI
                        body = body[:MAX_BODY_BYTES]
                    if not ev.get("more_body"):
                        break
                if too_big:
                    await _respond(send, 413, {"error": "body too large"})
                    return
                try:
                    msg = json.loads(body)
                except json.JSONDecodeError:
                    await _respond(send, 400, {"error": "bad json"})
                    return
                # thread-dispatch so concurrent requests overlap and
                # the query batcher can group them into one GPU batch
                # (dedicated wide pool: asyncio's default executor is
                # cpu_count+4 workers, which caps in-flight requests
                # far below the GPU batch size)
                import asyncio
                resp = await asyncio.get_running_loop().run_in_executor(
                    _handler_pool(), server.handle_message, msg)
                await _respond(send, 200, resp or {})
            elif scope["method"] == "GET" and scope["path"] == "/health":
                await _respond(send, 200, {"ok": True})
            else:
                await _respond(send, 404, {"error": "not found"})
        return app


_POOL = None


def _handler_pool():
    global _POOL
    if _POOL is None:
        from concurrent.futures import ThreadPoolExecutor
        _POOL = ThreadPoolExecutor(max_workers=256,
                                   thread_name_prefix="mcp-handler")
    return _POOL


async def _respond(send, status: int, payload: dict) -> None:
    body = json.dumps(payload, default=str).encode()
    await send({"type": "http.response.start", "status": status,
                "headers": [(b"content-type", b"application/json"),
                            (b"content-length",
                             str(len(body)).encode()),
                            (b"access-control-allow-origin", b"*")]})
    await send({"type": "http.response.body", "body": body})


def run_mcp_server(ctx: AppContext | None = None, api_key: str = "") -> None:
    """stdio entry (blocking)."""
    ctx = ctx or AppContext.create()
    McpServer(ctx, api_key=api_key).run_stdio()


def run_mcp_http_server(ctx: AppContext | None = None, host: str = "127.0.0.1",
                        port: int = 8765, api_key: str = "") -> None:
    import uvicorn
    ctx = ctx or AppContext.create()
    server = McpServer(ctx, api_key=api_key)
    uvicorn.run(server.asgi_app(), host=host, port=port, log_level="warning")
