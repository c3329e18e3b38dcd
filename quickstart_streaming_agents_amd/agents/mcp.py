"""MCP streamable-HTTP JSON-RPC client + a local stub MCP server.

The reference wires agents to a remote MCP server over streamable HTTP
(terraform/lab1-tool-calling/main.tf:66-72: CREATE CONNECTION type
MCP_SERVER, transport STREAMABLE_HTTP; tools http_get / http_post /
send_email per LAB1-Walkthrough.md:141-148 and LAB3:385-392).  This module
provides:

- ``McpClient`` — minimal JSON-RPC 2.0 over HTTP POST (initialize,
  tools/list, tools/call) with a per-request timeout (the reference's
  CREATE TOOL request_timeout=30).
- ``StubMcpServer`` — an in-process HTTP server exposing the same three
  tools plus the lab fixtures the tools fetch: the lab1 competitor price
  site (assets/lab1/competitor-site analog: prices undercut the catalog)
  and the lab3 vessel_catalog / dispatch API.  Everything runs air-gapped.
"""

from __future__ import annotations

import json
import threading
import time
import urllib.error
import urllib.request
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Any


class McpError(RuntimeError):
    pass


class McpClient:
    """JSON-RPC 2.0 over streamable HTTP (single POST endpoint)."""

    def __init__(self, endpoint: str, timeout_s: float = 30.0):
        self.endpoint = endpoint
        self.timeout_s = timeout_s
        self._id = 0
        self._lock = threading.Lock()

    def _call(self, method: str, params: dict | None = None) -> Any:
        with self._lock:
            self._id += 1
            rid = self._id
        payload = {"jsonrpc": "2.0", "id": rid, "method": method,
                   "params": params or {}}
        req = urllib.request.Request(
            self.endpoint, data=json.dumps(payload).encode(),
            headers={"Content-Type": "application/json",
                     "Accept": "application/json, text/event-stream"})
        # transport-level retry with backoff (the reference wraps flaky
        # externals the same way, deploy.py:83-91): under hundreds of
        # concurrent episodes a connection can be refused/reset; only
        # transport errors retry — JSON-RPC errors surface immediately
        body = None
        delay = 0.05
        for attempt in range(4):
            try:
                with urllib.request.urlopen(req,
                                            timeout=self.timeout_s) as resp:
                    body = json.loads(resp.read().decode())
                break
            except (urllib.error.URLError, ConnectionError, OSError):
                if attempt == 3:
                    raise
                time.sleep(delay)
                delay *= 2
        if "error" in body:
            raise McpError(str(body["error"]))
        return body.get("result")

    def initialize(self) -> Any:
        return self._call("initialize", {
            "protocolVersion": "2025-03-26",
            "capabilities": {}, "clientInfo": {"name": "qsa-amd", "version": "0.1"}})

    def tools_list(self) -> list[dict]:
        return self._call("tools/list").get("tools", [])

    def tools_call(self, name: str, arguments: dict) -> str:
        result = self._call("tools/call", {"name": name, "arguments": arguments})
        content = result.get("content", [])
        texts = [c.get("text", "") for c in content if c.get("type") == "text"]
        if result.get("isError"):
            raise McpError("; ".join(texts) or "tool error")
        return "\n".join(texts)


# ---------------------------------------------------------------------------
# Stub server (air-gapped lab fixtures)
# ---------------------------------------------------------------------------

# Competitor site: undercuts the lab1 catalog (reference
# assets/lab1/competitor-site/index.html, e.g. AirPods $249 -> $209.99).
COMPETITOR_PRICES = {
    "AirPods Pro": 209.99,
    "Mechanical Keyboard": 104.99,
    "4K Monitor": 339.00,
    "Espresso Machine": 469.00,
    "Chef Knife": 74.50,
    "Cast Iron Skillet": 36.99,
    "Running Shoes": 118.00,
    "Yoga Mat": 27.99,
    "Carbon Road Bike": 1999.00,
    "Noise-Cancel Headphones": 279.00,
    "Smart Thermostat": 159.00,
    "Robot Vacuum": 419.00,
    "Standing Desk": 519.00,
    "Ergonomic Chair": 659.00,
    "USB-C Dock": 134.00,
    "E-Reader": 119.00,
    "Action Camera": 259.00,
}

VESSELS = [
    {"vessel_id": f"BOAT-{i:02d}", "name": name, "capacity": cap,
     "status": "available"}
    for i, (name, cap) in enumerate([
        ("River Queen", 12), ("Bayou Runner", 8), ("Crescent Star", 10),
        ("Delta Dawn", 6), ("Gulf Breeze", 8), ("Pelican", 4),
        ("Magnolia", 10), ("Cypress", 6), ("Jazz Line", 12), ("Steamboat W", 20),
    ], start=1)
]


def competitor_html() -> str:
    rows = "\n".join(
        f'<tr><td class="product">{name}</td><td class="price">${price:.2f}</td></tr>'
        for name, price in COMPETITOR_PRICES.items())
    return ("<html><head><title>MegaDeals Outlet</title></head><body>"
            "<h1>MegaDeals Outlet — Today's Prices</h1><table>"
            f"{rows}</table></body></html>")


class _Handler(BaseHTTPRequestHandler):
    server_version = "QsaStubMcp/0.1"

    def log_message(self, fmt, *args):  # silence
        pass

    def _send(self, code: int, body: bytes, ctype: str = "application/json"):
        self.send_response(code)
        self.send_header("Content-Type", ctype)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def do_GET(self):
        if self.path.startswith("/competitor"):
            self._send(200, competitor_html().encode(), "text/html")
        elif self.path.startswith("/api/vessel_catalog"):
            self._send(200, json.dumps({"vessels": VESSELS}).encode())
        else:
            self._send(404, b'{"error":"not found"}')

    def do_POST(self):
        n = int(self.headers.get("Content-Length", 0))
        raw = self.rfile.read(n)
        if self.path.startswith("/api/dispatch"):
            try:
                req = json.loads(raw or b"{}")
            except json.JSONDecodeError:
                self._send(400, b'{"error":"bad json"}')
                return
            boats = req.get("boats") or req.get("vessels") or []
            self.server.ctx["dispatches"].append(req)
            self._send(200, json.dumps({
                "status": "dispatched", "count": len(boats),
                "confirmation_id": f"DSP-{len(self.server.ctx['dispatches']):04d}",
            }).encode())
            return
        if self.path.startswith("/mcp"):
            self._handle_mcp(raw)
            return
        self._send(404, b'{"error":"not found"}')

    # ---- MCP JSON-RPC ----
    def _handle_mcp(self, raw: bytes):
        try:
            req = json.loads(raw)
        except json.JSONDecodeError:
            self._send(400, b'{"error":"bad json"}')
            return
        rid = req.get("id")
        method = req.get("method", "")
        params = req.get("params", {}) or {}
        try:
            if method == "initialize":
                result = {"protocolVersion": "2025-03-26",
                          "serverInfo": {"name": "qsa-stub-mcp", "version": "0.1"},
                          "capabilities": {"tools": {}}}
            elif method == "tools/list":
                result = {"tools": [
                    {"name": "http_get",
                     "description": "Fetch a URL and return its body as text",
                     "inputSchema": {"type": "object",
                                     "properties": {"url": {"type": "string"}},
                                     "required": ["url"]}},
                    {"name": "http_post",
                     "description": "POST a JSON body to a URL",
                     "inputSchema": {"type": "object",
                                     "properties": {"url": {"type": "string"},
                                                    "body": {"type": "string"}},
                                     "required": ["url"]}},
                    {"name": "send_email",
                     "description": "Send an email",
                     "inputSchema": {"type": "object",
                                     "properties": {"to": {"type": "string"},
                                                    "subject": {"type": "string"},
                                                    "body": {"type": "string"}},
                                     "required": ["to", "subject", "body"]}},
                ]}
            elif method == "tools/call":
                # tool-execution failures map to an isError *result* per
                # MCP; protocol-level failures below are JSON-RPC errors
                try:
                    result = self._tool_call(params.get("name", ""),
                                             params.get("arguments", {}) or {})
                except Exception as e:
                    result = {"isError": True,
                              "content": [{"type": "text", "text": str(e)}]}
            else:
                raise McpError(f"unknown method {method}")
            body = {"jsonrpc": "2.0", "id": rid, "result": result}
        except McpError as e:  # method not found -> JSON-RPC error object
            body = {"jsonrpc": "2.0", "id": rid,
                    "error": {"code": -32601, "message": str(e)}}
        self._send(200, json.dumps(body).encode())

    def _tool_call(self, name: str, args: dict) -> dict:
        ctx = self.server.ctx
        if name == "http_get":
            url = args["url"]
            with urllib.request.urlopen(url, timeout=10) as resp:
                text = resp.read().decode(errors="replace")
            return {"content": [{"type": "text", "text": text}]}
        if name == "http_post":
            url = args["url"]
            data = args.get("body", "")
            if isinstance(data, (dict, list)):
                data = json.dumps(data)
            req = urllib.request.Request(
                url, data=data.encode(),
                headers={"Content-Type": "application/json"})
            with urllib.request.urlopen(req, timeout=10) as resp:
                text = resp.read().decode(errors="replace")
            return {"content": [{"type": "text", "text": text}]}
        if name == "send_email":
            ctx["emails"].append({"to": args.get("to", ""),
                                  "subject": args.get("subject", ""),
                                  "body": args.get("body", "")})
            return {"content": [{"type": "text",
                                 "text": f"Email sent to {args.get('to','')}"}]}
        raise McpError(f"unknown tool {name}")


class _StubHttpServer(ThreadingHTTPServer):
    request_queue_size = 256   # default backlog of 5 drops connections
                               # under hundreds of concurrent episodes


class StubMcpServer:
    """In-process stub: MCP endpoint + competitor site + vessel API."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self.httpd = _StubHttpServer((host, port), _Handler)
        self.httpd.ctx = {"emails": [], "dispatches": []}
        self._thread: threading.Thread | None = None

    @property
    def port(self) -> int:
        return self.httpd.server_address[1]

    @property
    def base_url(self) -> str:
        return f"http://127.0.0.1:{self.port}"

    @property
    def mcp_endpoint(self) -> str:
        return f"{self.base_url}/mcp"

    @property
    def emails(self) -> list[dict]:
        return self.httpd.ctx["emails"]

    @property
    def dispatches(self) -> list[dict]:
        return self.httpd.ctx["dispatches"]

    def start(self) -> "StubMcpServer":
        self._thread = threading.Thread(target=self.httpd.serve_forever,
                                        daemon=True)
        self._thread.start()
        return self

    def stop(self) -> None:
        self.httpd.shutdown()
        self.httpd.server_close()

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()
