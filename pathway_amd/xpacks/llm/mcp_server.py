"""MCP server (reference xpacks/llm/mcp_server.py: PathwayMcp over
fastmcp).

Implements the Model Context Protocol directly — JSON-RPC 2.0 with
``initialize`` / ``tools/list`` / ``tools/call`` — over two transports:

  * stdio (newline-delimited JSON-RPC, the MCP reference transport)
  * streamable-http (POST JSON-RPC to a local HTTP endpoint)

Tools are registered by `McpServable` components (e.g. DocumentStore
query surfaces) via register_mcp(server).
"""

from __future__ import annotations

import json
import sys
import threading
from typing import Any, Callable


class McpServable:
    def register_mcp(self, server: "McpServer") -> None:
        raise NotImplementedError


class McpServer:
    """Tool registry + JSON-RPC dispatcher."""

    PROTOCOL_VERSION = "2024-11-05"

    def __init__(self, name: str = "pathway-amd"):
        self.name = name
        self.tools: dict[str, tuple[Callable | None, Any]] = {}

    def tool(self, name: str, *, request_handler: Callable | None = None,
             schema: Any = None) -> None:
        self.tools[name] = (request_handler, schema)

    # -- JSON-RPC dispatch --

    def handle(self, msg: dict) -> dict | None:
        rid = msg.get("id")
        method = msg.get("method")

        def ok(result):
            return {"jsonrpc": "2.0", "id": rid, "result": result}

        def err(code, text):
            return {"jsonrpc": "2.0", "id": rid,
                    "error": {"code": code, "message": text}}

        if method == "initialize":
            return ok({
                "protocolVersion": self.PROTOCOL_VERSION,
                "capabilities": {"tools": {}},
                "serverInfo": {"name": self.name, "version": "1.0"},
            })
        if method == "notifications/initialized":
            return None  # notification: no response
        if method == "tools/list":
            tools = []
            for name, (_h, schema) in self.tools.items():
                tools.append({
                    "name": name,
                    "description": f"pathway tool {name}",
                    "inputSchema": schema or {"type": "object"},
                })
            return ok({"tools": tools})
        if method == "tools/call":
            params = msg.get("params") or {}
            name = params.get("name")
            entry = self.tools.get(name)
            if entry is None or entry[0] is None:
                return err(-32601, f"unknown tool {name!r}")
            handler = entry[0]
            try:
                result = handler(params.get("arguments") or {})
                if not isinstance(result, str):
                    result = json.dumps(result, default=str)
                return ok({"content": [{"type": "text", "text": result}]})
            except Exception as e:  # tool errors surface as MCP errors
                return ok({
                    "content": [{"type": "text", "text": f"error: {e}"}],
                    "isError": True,
                })
        if method == "ping":
            return ok({})
        return err(-32601, f"unknown method {method!r}")

    # -- transports --

    def serve_stdio(self, infile=None, outfile=None) -> None:
        """Newline-delimited JSON-RPC over stdio (the MCP reference
        transport); blocks until EOF."""
        infile = infile or sys.stdin
        outfile = outfile or sys.stdout
        for line in infile:
            line = line.strip()
            if not line:
                continue
            try:
                msg = json.loads(line)
            except json.JSONDecodeError:
                continue
            resp = self.handle(msg)
            if resp is not None:
                outfile.write(json.dumps(resp) + "\n")
                outfile.flush()

    def serve_http(self, host: str = "127.0.0.1", port: int = 0):
        """Streamable-HTTP transport: POST one JSON-RPC message per
        request.  Returns the running server (``.server_address``)."""
        from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

        mcp = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def do_POST(self):
                n = int(self.headers.get("Content-Length", 0))
                try:
                    msg = json.loads(self.rfile.read(n))
                    resp = mcp.handle(msg)
                except Exception as e:
                    resp = {"jsonrpc": "2.0", "id": None,
                            "error": {"code": -32700, "message": str(e)}}
                body = json.dumps(resp or {}).encode()
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

        server = ThreadingHTTPServer((host, port), Handler)
        th = threading.Thread(target=server.serve_forever, daemon=True)
        th.start()
        return server


class McpConfig:
    def __init__(self, name: str = "pathway-amd",
                 transport: str = "streamable-http",
                 host: str | None = None, port: int | None = None):
        self.name = name
        self.transport = transport
        self.host = host or "127.0.0.1"
        self.port = port or 0


class PathwayMcp:
    """Top-level MCP app (reference PathwayMcp): collects servables,
    serves over the configured transport when pw.run() starts."""

    def __init__(self, serve: list | None = None, name: str = "pathway-amd",
                 transport: str = "streamable-http", host: str | None = None,
                 port: int | None = None):
        self.config = McpConfig(name, transport, host, port)
        self.server = McpServer(name)
        for s in serve or []:
            s.register_mcp(self.server)
        self._http = None

    def start(self):
        if self.config.transport == "stdio":
            self.server.serve_stdio()
        else:
            self._http = self.server.serve_http(
                self.config.host, self.config.port or 0
            )
            return self._http
