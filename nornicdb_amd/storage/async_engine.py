"""Async write-behind engine wrapper.

Parity: reference pkg/storage/async_engine.go:22 (write buffering, 50 ms
default flush loop :197). Reads are served write-through (buffer first,
then inner) so read-your-writes holds before the flush.
"""

from __future__ import annotations

import threading
from typing import Dict, List, Optional

from .types import Edge, Engine, Node, NotFoundError


class AsyncEngine(Engine):
    def __init__(self, inner: Engine, flush_interval: float = 0.05,
                 max_buffer: int = 10000):
        self.inner = inner
        self._lock = threading.Lock()
        self._buf_nodes: Dict[str, Optional[Node]] = {}   # None = deleted
        self._buf_edges: Dict[str, Optional[Edge]] = {}
        self._order: List[tuple] = []
        self._flush_interval = flush_interval
        self._max_buffer = max_buffer
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def _loop(self):
        while not self._stop.wait(self._flush_interval):
            try:
                self.flush()
            except Exception:
                pass

    def flush(self):
        with self._lock:
            order = self._order
            self._order = []
            self._buf_nodes = {}
            self._buf_edges = {}
        for kind, op, obj in order:
            try:
                if kind == "node":
                    if op == "create":
                        self.inner.create_node(obj)
                    elif op == "update":
                        self.inner.update_node(obj)
                    elif op == "delete":
                        self.inner.delete_node(obj)
                    elif op == "detach":
                        self.inner.detach_delete_node(obj)
                else:
                    if op == "create":
                        self.inner.create_edge(obj)
                    elif op == "update":
                        self.inner.update_edge(obj)
                    elif op == "delete":
                        self.inner.delete_edge(obj)
            except Exception:
                pass

    def _maybe_flush(self):
        if len(self._order) >= self._max_buffer:
            self.flush()

    # --- nodes ---
    def create_node(self, node: Node) -> Node:
        with self._lock:
            self._buf_nodes[node.id] = node.copy()
            self._order.append(("node", "create", node.copy()))
        self._maybe_flush()
        return node

    def get_node(self, node_id: str) -> Node:
        with self._lock:
            if node_id in self._buf_nodes:
                n = self._buf_nodes[node_id]
                if n is None:
                    raise NotFoundError(f"node {node_id} not found")
                return n.copy()
        return self.inner.get_node(node_id)

    def update_node(self, node: Node) -> Node:
        with self._lock:
            self._buf_nodes[node.id] = node.copy()
            self._order.append(("node", "update", node.copy()))
        self._maybe_flush()
        return node

    def delete_node(self, node_id: str) -> None:
        with self._lock:
            self._buf_nodes[node_id] = None
            self._order.append(("node", "delete", node_id))
        self._maybe_flush()

    def detach_delete_node(self, node_id: str) -> None:
        with self._lock:
            self._buf_nodes[node_id] = None
            self._order.append(("node", "detach", node_id))
        self._maybe_flush()

    def get_nodes_by_label(self, label: str):
        self.flush()
        return self.inner.get_nodes_by_label(label)

    def all_nodes(self):
        self.flush()
        return self.inner.all_nodes()

    def node_count(self) -> int:
        self.flush()
        return self.inner.node_count()

    # --- edges ---
    def create_edge(self, edge: Edge) -> Edge:
        with self._lock:
            self._buf_edges[edge.id] = edge.copy()
            self._order.append(("edge", "create", edge.copy()))
        self._maybe_flush()
        return edge

    def get_edge(self, edge_id: str) -> Edge:
        with self._lock:
            if edge_id in self._buf_edges:
                e = self._buf_edges[edge_id]
                if e is None:
                    raise NotFoundError(f"edge {edge_id} not found")
                return e.copy()
        return self.inner.get_edge(edge_id)

    def update_edge(self, edge: Edge) -> Edge:
        with self._lock:
            self._buf_edges[edge.id] = edge.copy()
            self._order.append(("edge", "update", edge.copy()))
        self._maybe_flush()
        return edge

    def delete_edge(self, edge_id: str) -> None:
        with self._lock:
            self._buf_edges[edge_id] = None
            self._order.append(("edge", "delete", edge_id))
        self._maybe_flush()

    def get_edges_by_type(self, edge_type: str):
        self.flush()
        return self.inner.get_edges_by_type(edge_type)

    def all_edges(self):
        self.flush()
        return self.inner.all_edges()

    def edge_count(self) -> int:
        self.flush()
        return self.inner.edge_count()

    # --- adjacency ---
    def get_out_edges(self, node_id: str):
        self.flush()
        return self.inner.get_out_edges(node_id)

    def get_in_edges(self, node_id: str):
        self.flush()
        return self.inner.get_in_edges(node_id)

    def neighbors(self, node_id: str):
        self.flush()
        return self.inner.neighbors(node_id)

    # --- passthrough ---
    def mark_pending_embedding(self, node_id: str):
        self.inner.mark_pending_embedding(node_id)

    def pending_embeddings(self, limit: int = 0):
        return self.inner.pending_embeddings(limit)

    def clear_pending_embedding(self, node_id: str):
        self.inner.clear_pending_embedding(node_id)

    def register_callback(self, cb):
        self.inner.register_callback(cb)

    def add_validator(self, fn):
        # same forwarding rule as events: enforcement lives in the
        # physical engine's write path
        if hasattr(self.inner, "add_validator"):
            self.inner.add_validator(fn)

    def iter_nodes_by_label(self, label):
        self.flush()
        return self.inner.iter_nodes_by_label(label)

    def create_property_index(self, label, prop):
        self.flush()
        return self.inner.create_property_index(label, prop)

    def drop_property_index(self, label, prop):
        return self.inner.drop_property_index(label, prop)

    def lookup_property_index(self, label, prop, value):
        self.flush()
        return self.inner.lookup_property_index(label, prop, value)

    def close(self):
        self._stop.set()
        self._thread.join(timeout=1)
        self.flush()
        self.inner.close()
