"""MCP server surface (reference xpacks/llm/mcp_server.py) — tool registry
over the DocumentStore query tables; network transport lands later."""
from __future__ import annotations


class McpServable:
    def register_mcp(self, server):
        raise NotImplementedError


class McpServer:
    def __init__(self, name: str = "pathway-amd"):
        self.name = name
        self.tools = {}

    def tool(self, name, *, request_handler=None, schema=None):
        self.tools[name] = (request_handler, schema)


class McpConfig:
    def __init__(self, name="pathway-amd", transport="streamable-http", host=None, port=None):
        self.name = name
        self.transport = transport
