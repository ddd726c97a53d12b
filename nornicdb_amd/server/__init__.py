"""Protocol servers: HTTP (FastAPI), MCP JSON-RPC, Prometheus metrics."""

from .http import create_app
from .mcp import MCPServer
from .metrics import MetricsRegistry

__all__ = ["create_app", "MCPServer", "MetricsRegistry"]
