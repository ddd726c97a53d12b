"""In-memory storage engine — the universal test fixture and the in-RAM
store behind the WAL/persistent engines.

Parity: reference pkg/storage/memory.go:37 (MemoryEngine) with the label
index and out/in adjacency indexes of the Badger engine's key prefixes
(reference pkg/storage/badger.go:16-26: node/edge/label-index/out/in/
edgetype/pending-embed).
"""

from __future__ import annotations

import threading
import time
from typing import Any, Callable, Dict, Iterator, List, Optional, Set

from .types import (ConstraintViolation, Edge, Engine, EventType, Node,
                    NotFoundError)


class MemoryEngine(Engine):
    def __init__(self):
        self._lock = threading.RLock()
        self._nodes: Dict[str, Node] = {}
        self._edges: Dict[str, Edge] = {}
        self._label_index: Dict[str, Set[str]] = {}
        self._type_index: Dict[str, Set[str]] = {}
        self._out: Dict[str, Set[str]] = {}  # node -> edge ids
        self._in: Dict[str, Set[str]] = {}
        self._pending_embed: Dict[str, float] = {}
        # O(1) per-(label, namespace-prefix) counts; prefix = id up to ':'
        self._label_ns_counts: Dict[tuple, int] = {}
        self._callbacks: List[Callable[[str, Any], None]] = []
        self._validators: List[Callable] = []
        # property indexes: (label, prop) -> value -> set[node_id]
        self._prop_indexes: Dict[tuple, Dict[Any, Set[str]]] = {}

    # ---- events ----
    def register_callback(self, cb):
        with self._lock:
            self._callbacks.append(cb)

    def add_validator(self, fn):
        """fn(node, is_update) raises ConstraintViolation to veto a write."""
        with self._lock:
            self._validators.append(fn)

    def _check(self, node, is_update):
        for v in list(self._validators):
            v(node, is_update)

    @staticmethod
    def _ns_of(node_id: str) -> str:
        i = node_id.find(":")
        return node_id[:i] if i > 0 else ""

    def _count_adjust(self, node: Node, delta: int):
        ns = self._ns_of(node.id)
        for lb in node.labels:
            k = (lb, ns)
            v = self._label_ns_counts.get(k, 0) + delta
            if v <= 0:
                self._label_ns_counts.pop(k, None)
            else:
                self._label_ns_counts[k] = v

    def _emit(self, ev: str, obj):
        for cb in list(self._callbacks):
            try:
                cb(ev, obj)
            except Exception:
                pass  # callbacks must not break storage

    # ---- nodes ----
    def create_node(self, node: Node) -> Node:
        self._check(node, False)
        with self._lock:
            if node.id in self._nodes:
                raise ConstraintViolation(f"node {node.id} already exists")
            n = node.copy()
            n.created_at = n.created_at or time.time()
            n.updated_at = n.created_at
            self._nodes[n.id] = n
            for lb in n.labels:
                self._label_index.setdefault(lb, set()).add(n.id)
            self._count_adjust(n, +1)
            self._index_node_props(n, add=True)
        self._emit(EventType.NODE_CREATED, n.copy())
        return n.copy()

    def get_node(self, node_id: str) -> Node:
        with self._lock:
            n = self._nodes.get(node_id)
            if n is None:
                raise NotFoundError(f"node {node_id} not found")
            return n.copy()

    def update_node(self, node: Node) -> Node:
        self._check(node, True)
        with self._lock:
            old = self._nodes.get(node.id)
            if old is None:
                raise NotFoundError(f"node {node.id} not found")
            for lb in old.labels:
                self._label_index.get(lb, set()).discard(node.id)
            self._count_adjust(old, -1)
            self._index_node_props(old, add=False)
            n = node.copy()
            n.created_at = old.created_at
            n.updated_at = time.time()
            self._nodes[n.id] = n
            for lb in n.labels:
                self._label_index.setdefault(lb, set()).add(n.id)
            self._count_adjust(n, +1)
            self._index_node_props(n, add=True)
        self._emit(EventType.NODE_UPDATED, n.copy())
        return n.copy()

    def delete_node(self, node_id: str) -> None:
        with self._lock:
            n = self._nodes.get(node_id)
            if n is None:
                raise NotFoundError(f"node {node_id} not found")
            if self._out.get(node_id) or self._in.get(node_id):
                raise ConstraintViolation(
                    f"node {node_id} still has relationships (use DETACH DELETE)")
            del self._nodes[node_id]
            for lb in n.labels:
                self._label_index.get(lb, set()).discard(node_id)
            self._count_adjust(n, -1)
            self._index_node_props(n, add=False)
            self._pending_embed.pop(node_id, None)
        self._emit(EventType.NODE_DELETED, n)

    def get_nodes_by_label(self, label: str) -> List[Node]:
        with self._lock:
            ids = self._label_index.get(label, set())
            return [self._nodes[i].copy() for i in ids if i in self._nodes]

    def all_nodes(self) -> Iterator[Node]:
        with self._lock:
            snap = [n.copy() for n in self._nodes.values()]
        return iter(snap)

    def node_count(self) -> int:
        with self._lock:
            return len(self._nodes)

    def node_count_by_label(self, label: str, ns: str = None) -> int:
        with self._lock:
            if ns is None:
                return len(self._label_index.get(label, ()))
            return self._label_ns_counts.get((label, ns), 0)

    def iter_nodes_by_label(self, label: str):
        """Lazy copying label scan: snapshot the id set (cheap), then copy
        nodes one at a time — pagination (GraphQL nodes(limit:)) stops
        after `limit` copies instead of copying the whole label set."""
        with self._lock:
            ids = list(self._label_index.get(label, ()))
        for i in ids:
            with self._lock:
                n = self._nodes.get(i)
                c = n.copy() if n is not None else None
            if c is not None:
                yield c

    def iter_nodes_raw(self, label: str = None):
        """Yield LIVE node objects without copying — read-only fast paths
        (reference storage_fastpaths.go). Callers must not mutate."""
        with self._lock:
            if label is None:
                snap = list(self._nodes.values())
            else:
                snap = [self._nodes[i]
                        for i in self._label_index.get(label, ())
                        if i in self._nodes]
        return iter(snap)

    # ---- edges ----
    def create_edge(self, edge: Edge) -> Edge:
        with self._lock:
            if edge.id in self._edges:
                raise ConstraintViolation(f"edge {edge.id} already exists")
            if edge.start_node not in self._nodes:
                raise NotFoundError(f"start node {edge.start_node} not found")
            if edge.end_node not in self._nodes:
                raise NotFoundError(f"end node {edge.end_node} not found")
            e = edge.copy()
            e.created_at = e.created_at or time.time()
            e.updated_at = e.created_at
            self._edges[e.id] = e
            self._type_index.setdefault(e.type, set()).add(e.id)
            self._out.setdefault(e.start_node, set()).add(e.id)
            self._in.setdefault(e.end_node, set()).add(e.id)
        self._emit(EventType.EDGE_CREATED, e.copy())
        return e.copy()

    def get_edge(self, edge_id: str) -> Edge:
        with self._lock:
            e = self._edges.get(edge_id)
            if e is None:
                raise NotFoundError(f"edge {edge_id} not found")
            return e.copy()

    def update_edge(self, edge: Edge) -> Edge:
        with self._lock:
            old = self._edges.get(edge.id)
            if old is None:
                raise NotFoundError(f"edge {edge.id} not found")
            e = edge.copy()
            e.created_at = old.created_at
            e.updated_at = time.time()
            if old.type != e.type:
                self._type_index.get(old.type, set()).discard(e.id)
                self._type_index.setdefault(e.type, set()).add(e.id)
            self._edges[e.id] = e
        self._emit(EventType.EDGE_UPDATED, e.copy())
        return e.copy()

    def delete_edge(self, edge_id: str) -> None:
        with self._lock:
            e = self._edges.get(edge_id)
            if e is None:
                raise NotFoundError(f"edge {edge_id} not found")
            del self._edges[edge_id]
            self._type_index.get(e.type, set()).discard(edge_id)
            self._out.get(e.start_node, set()).discard(edge_id)
            self._in.get(e.end_node, set()).discard(edge_id)
        self._emit(EventType.EDGE_DELETED, e)

    def get_edges_by_type(self, edge_type: str) -> List[Edge]:
        with self._lock:
            ids = self._type_index.get(edge_type, set())
            return [self._edges[i].copy() for i in ids if i in self._edges]

    def all_edges(self) -> Iterator[Edge]:
        with self._lock:
            snap = [e.copy() for e in self._edges.values()]
        return iter(snap)

    def edge_count(self) -> int:
        with self._lock:
            return len(self._edges)

    # ---- adjacency ----
    def get_out_edges(self, node_id: str) -> List[Edge]:
        with self._lock:
            return [self._edges[i].copy() for i in self._out.get(node_id, set())
                    if i in self._edges]

    def get_in_edges(self, node_id: str) -> List[Edge]:
        with self._lock:
            return [self._edges[i].copy() for i in self._in.get(node_id, set())
                    if i in self._edges]

    def neighbors(self, node_id: str) -> List[str]:
        with self._lock:
            out = {self._edges[i].end_node for i in self._out.get(node_id, set())
                   if i in self._edges}
            inn = {self._edges[i].start_node for i in self._in.get(node_id, set())
                   if i in self._edges}
            return sorted(out | inn)

    def detach_delete_node(self, node_id: str) -> None:
        with self._lock:
            for eid in list(self._out.get(node_id, set()) | self._in.get(node_id, set())):
                if eid in self._edges:
                    self.delete_edge(eid)
        self.delete_node(node_id)

    # ---- pending embeddings ----
    def mark_pending_embedding(self, node_id: str) -> None:
        with self._lock:
            self._pending_embed[node_id] = time.time()

    def pending_embeddings(self, limit: int = 0) -> List[str]:
        with self._lock:
            ids = sorted(self._pending_embed, key=self._pending_embed.get)
            return ids[:limit] if limit else ids

    def clear_pending_embedding(self, node_id: str) -> None:
        with self._lock:
            self._pending_embed.pop(node_id, None)

    # ---- property indexes (reference pkg/storage/schema.go) ----
    def create_property_index(self, label: str, prop: str) -> None:
        with self._lock:
            key = (label, prop)
            if key in self._prop_indexes:
                return
            idx: Dict[Any, Set[str]] = {}
            for nid in self._label_index.get(label, set()):
                v = self._nodes[nid].properties.get(prop)
                if v is not None and isinstance(v, (str, int, float, bool)):
                    idx.setdefault(v, set()).add(nid)
            self._prop_indexes[key] = idx

    def drop_property_index(self, label: str, prop: str) -> None:
        with self._lock:
            self._prop_indexes.pop((label, prop), None)

    def lookup_property_index(self, label: str, prop: str, value) -> Optional[List[Node]]:
        """None if no such index; else exact-match node list."""
        with self._lock:
            idx = self._prop_indexes.get((label, prop))
            if idx is None:
                return None
            return [self._nodes[i].copy() for i in idx.get(value, set())
                    if i in self._nodes]

    def _index_node_props(self, node: Node, add: bool):
        for (label, prop), idx in self._prop_indexes.items():
            if label in node.labels:
                v = node.properties.get(prop)
                if v is not None and isinstance(v, (str, int, float, bool)):
                    if add:
                        idx.setdefault(v, set()).add(node.id)
                    else:
                        idx.get(v, set()).discard(node.id)

    # ---- bulk state (snapshot support) ----
    def dump_state(self):
        with self._lock:
            return {
                "nodes": [
                    (n.id, n.labels, n.properties, n.embedding, n.created_at, n.updated_at)
                    for n in self._nodes.values()
                ],
                "edges": [
                    (e.id, e.type, e.start_node, e.end_node, e.properties,
                     e.created_at, e.updated_at)
                    for e in self._edges.values()
                ],
                "pending": dict(self._pending_embed),
            }

    def load_state(self, state):
        with self._lock:
            self._nodes.clear(); self._edges.clear()
            self._label_index.clear(); self._type_index.clear()
            self._out.clear(); self._in.clear()
            self._pending_embed = dict(state.get("pending", {}))
            self._label_ns_counts.clear()
            for (nid, labels, props, emb, ca, ua) in state["nodes"]:
                n = Node(nid, list(labels), dict(props), emb, ca, ua)
                self._nodes[nid] = n
                for lb in n.labels:
                    self._label_index.setdefault(lb, set()).add(nid)
                self._count_adjust(n, +1)
            for (eid, et, s, t, props, ca, ua) in state["edges"]:
                e = Edge(eid, et, s, t, dict(props), ca, ua)
                self._edges[eid] = e
                self._type_index.setdefault(et, set()).add(eid)
                self._out.setdefault(s, set()).add(eid)
                self._in.setdefault(t, set()).add(eid)
            for key in list(self._prop_indexes):
                self._prop_indexes[key] = {}
                lbl, prop = key
                for nid in self._label_index.get(lbl, set()):
                    v = self._nodes[nid].properties.get(prop)
                    if v is not None and isinstance(v, (str, int, float, bool)):
                        self._prop_indexes[key].setdefault(v, set()).add(nid)
