"""MCP (Model Context Protocol) server — JSON-RPC 2.0.

Parity: reference pkg/mcp — 6 tools store/recall/discover/link/task/tasks
(tools.go:8-345), initialize/tools.list/tools.call dispatch (server.go:68).
"""

from __future__ import annotations

import json
import time
from typing import Any, Dict

PROTOCOL_VERSION = "2024-11-05"

TOOLS = [
    {
        "name": "store",
        "description": "Store a memory (text content with optional title/tags).",
        "inputSchema": {
            "type": "object",
            "properties": {
                "content": {"type": "string"},
                "title": {"type": "string"},
                "memory_type": {"type": "string",
                                "enum": ["episodic", "semantic", "procedural"]},
                "importance": {"type": "number"},
                "tags": {"type": "array", "items": {"type": "string"}},
            },
            "required": ["content"],
        },
    },
    {
        "name": "recall",
        "description": "Hybrid-search stored memories by natural language query.",
        "inputSchema": {
            "type": "object",
            "properties": {"query": {"type": "string"},
                           "limit": {"type": "integer"}},
            "required": ["query"],
        },
    },
    {
        "name": "discover",
        "description": "Explore the neighborhood of a memory (linked memories).",
        "inputSchema": {
            "type": "object",
            "properties": {"id": {"type": "string"},
                           "depth": {"type": "integer"}},
            "required": ["id"],
        },
    },
    {
        "name": "link",
        "description": "Create a relationship between two memories.",
        "inputSchema": {
            "type": "object",
            "properties": {"from": {"type": "string"}, "to": {"type": "string"},
                           "type": {"type": "string"}},
            "required": ["from", "to"],
        },
    },
    {
        "name": "task",
        "description": "Create or update a task memory.",
        "inputSchema": {
            "type": "object",
            "properties": {"title": {"type": "string"},
                           "status": {"type": "string"},
                           "id": {"type": "string"}},
            "required": ["title"],
        },
    },
    {
        "name": "tasks",
        "description": "List open tasks.",
        "inputSchema": {"type": "object", "properties": {
            "status": {"type": "string"}}},
    },
]


class MCPServer:
    def __init__(self, mgr, db_name: str = None):
        self.mgr = mgr
        self.db_name = db_name

    @property
    def db(self):
        return self.mgr.get(self.db_name)

    def handle(self, req: Dict[str, Any]) -> Dict[str, Any]:
        rid = req.get("id")
        method = req.get("method", "")
        params = req.get("params") or {}
        try:
            if method == "initialize":
                result = {
                    "protocolVersion": PROTOCOL_VERSION,
                    "capabilities": {"tools": {}},
                    "serverInfo": {"name": "nornicdb-amd", "version": "0.1.0"},
                }
            elif method == "notifications/initialized":
                return {"jsonrpc": "2.0", "id": rid, "result": {}}
            elif method == "tools/list":
                result = {"tools": TOOLS}
            elif method == "tools/call":
                result = self._call_tool(params.get("name"),
                                         params.get("arguments") or {})
            elif method == "ping":
                result = {}
            else:
                return self._err(rid, -32601, f"method not found: {method}")
            return {"jsonrpc": "2.0", "id": rid, "result": result}
        except Exception as e:
            return self._err(rid, -32000, str(e))

    @staticmethod
    def _err(rid, code, msg):
        return {"jsonrpc": "2.0", "id": rid,
                "error": {"code": code, "message": msg}}

    @staticmethod
    def _text(obj) -> Dict[str, Any]:
        return {"content": [{"type": "text",
                             "text": json.dumps(obj, default=str)}]}

    def _call_tool(self, name: str, args: Dict[str, Any]):
        db = self.db
        if name == "store":
            m = db.store(args["content"], title=args.get("title", ""),
                         memory_type=args.get("memory_type", "episodic"),
                         importance=args.get("importance", 0.5),
                         tags=args.get("tags", []))
            return self._text({"id": m.id, "stored": True})
        if name == "recall":
            res = db.recall(args["query"], limit=args.get("limit", 10))
            return self._text([{"id": m.id, "title": m.title,
                                "content": m.content[:500],
                                "type": m.memory_type} for m in res])
        if name == "discover":
            res = db.neighbors(args["id"], depth=args.get("depth", 1))
            return self._text([{"id": m.id, "title": m.title,
                                "content": m.content[:200]} for m in res])
        if name == "link":
            e = db.link(args["from"], args["to"],
                        args.get("type", "RELATES_TO"))
            return self._text({"id": e.id, "linked": True})
        if name == "task":
            if args.get("id"):
                node = db.engine.get_node(args["id"])
                node.properties["status"] = args.get("status", "open")
                node.properties["title"] = args.get("title",
                                                    node.properties.get("title"))
                db.engine.update_node(node)
                return self._text({"id": node.id, "updated": True})
            from ..storage import Node, new_id
            node = Node(id=new_id("t"), labels=["Task"],
                        properties={"title": args["title"],
                                    "status": args.get("status", "open"),
                                    "created_at": time.time()})
            db.engine.create_node(node)
            return self._text({"id": node.id, "created": True})
        if name == "tasks":
            status = args.get("status")
            tasks = [
                {"id": n.id, "title": n.properties.get("title"),
                 "status": n.properties.get("status")}
                for n in db.engine.get_nodes_by_label("Task")
                if status is None or n.properties.get("status") == status
            ]
            return self._text(tasks)
        raise ValueError(f"unknown tool {name}")
