"""Namespaced engine: multi-tenant ID prefixing over a shared base engine.

Parity: reference pkg/storage/namespaced.go:1-40 — every node/edge id is
transparently prefixed with "<ns>:" so multiple logical databases share one
physical engine; label/type scans filter to the namespace.
"""

from __future__ import annotations

from typing import List

from .types import Edge, Engine, Node


class NamespacedEngine(Engine):
    def __init__(self, inner: Engine, namespace: str):
        self.inner = inner
        self.ns = namespace
        self._p = namespace + ":"

    # --- id mapping ---
    def _wrap(self, i: str) -> str:
        return i if i.startswith(self._p) else self._p + i

    def _unwrap(self, i: str) -> str:
        return i[len(self._p):] if i.startswith(self._p) else i

    def _mine(self, i: str) -> bool:
        return i.startswith(self._p)

    def _wrap_node(self, n: Node) -> Node:
        c = n.copy()
        c.id = self._wrap(c.id)
        return c

    def _unwrap_node(self, n: Node) -> Node:
        n.id = self._unwrap(n.id)
        return n

    def _wrap_edge(self, e: Edge) -> Edge:
        c = e.copy()
        c.id = self._wrap(c.id)
        c.start_node = self._wrap(c.start_node)
        c.end_node = self._wrap(c.end_node)
        return c

    def _unwrap_edge(self, e: Edge) -> Edge:
        e.id = self._unwrap(e.id)
        e.start_node = self._unwrap(e.start_node)
        e.end_node = self._unwrap(e.end_node)
        return e

    # --- validators (constraint enforcement) ---
    def add_validator(self, fn):
        """Register on the PHYSICAL engine (the wrapper is never consulted
        by inner write paths), scoped to this namespace and with the node
        translated back to namespace-local ids so the validator's
        duplicate lookups compare like with like."""
        def wrapped(node: Node, is_update: bool):
            if not self._mine(node.id):
                return
            c = node.copy()
            c.id = self._unwrap(c.id)
            fn(c, is_update)
        self.inner.add_validator(wrapped)

    # --- nodes ---
    def create_node(self, node: Node) -> Node:
        return self._unwrap_node(self.inner.create_node(self._wrap_node(node)))

    def get_node(self, node_id: str) -> Node:
        return self._unwrap_node(self.inner.get_node(self._wrap(node_id)))

    def update_node(self, node: Node) -> Node:
        return self._unwrap_node(self.inner.update_node(self._wrap_node(node)))

    def delete_node(self, node_id: str) -> None:
        self.inner.delete_node(self._wrap(node_id))

    def detach_delete_node(self, node_id: str) -> None:
        self.inner.detach_delete_node(self._wrap(node_id))

    def get_nodes_by_label(self, label: str) -> List[Node]:
        return [self._unwrap_node(n) for n in self.inner.get_nodes_by_label(label)
                if self._mine(n.id)]

    def iter_nodes_by_label(self, label: str):
        # inner yields fresh copies (Engine read contract), so unwrapping
        # in place is safe
        for n in self.inner.iter_nodes_by_label(label):
            if self._mine(n.id):
                yield self._unwrap_node(n)

    def all_nodes(self):
        for n in self.inner.all_nodes():
            if self._mine(n.id):
                yield self._unwrap_node(n)

    def node_count(self) -> int:
        return sum(1 for _ in self.all_nodes())

    def node_count_by_label(self, label: str) -> int:
        fast = getattr(self.inner, "node_count_by_label", None)
        if fast is not None:
            try:
                return fast(label, ns=self.ns)
            except TypeError:
                pass
        return len(self.get_nodes_by_label(label))

    def iter_nodes_raw(self, label: str = None):
        inner = getattr(self.inner, "iter_nodes_raw", None)
        if inner is None:
            return iter(())
        # raw nodes keep their PREFIXED ids (read-only count/filter paths
        # only look at labels/properties)
        return (n for n in inner(label) if self._mine(n.id))

    # --- edges ---
    def create_edge(self, edge: Edge) -> Edge:
        return self._unwrap_edge(self.inner.create_edge(self._wrap_edge(edge)))

    def get_edge(self, edge_id: str) -> Edge:
        return self._unwrap_edge(self.inner.get_edge(self._wrap(edge_id)))

    def update_edge(self, edge: Edge) -> Edge:
        return self._unwrap_edge(self.inner.update_edge(self._wrap_edge(edge)))

    def delete_edge(self, edge_id: str) -> None:
        self.inner.delete_edge(self._wrap(edge_id))

    def get_edges_by_type(self, edge_type: str) -> List[Edge]:
        return [self._unwrap_edge(e) for e in self.inner.get_edges_by_type(edge_type)
                if self._mine(e.id)]

    def all_edges(self):
        for e in self.inner.all_edges():
            if self._mine(e.id):
                yield self._unwrap_edge(e)

    def edge_count(self) -> int:
        return sum(1 for _ in self.all_edges())

    # --- adjacency ---
    def get_out_edges(self, node_id: str) -> List[Edge]:
        return [self._unwrap_edge(e) for e in self.inner.get_out_edges(self._wrap(node_id))]

    def get_in_edges(self, node_id: str) -> List[Edge]:
        return [self._unwrap_edge(e) for e in self.inner.get_in_edges(self._wrap(node_id))]

    def neighbors(self, node_id: str) -> List[str]:
        return [self._unwrap(i) for i in self.inner.neighbors(self._wrap(node_id))]

    # --- pending / events / index passthrough ---
    def mark_pending_embedding(self, node_id: str):
        self.inner.mark_pending_embedding(self._wrap(node_id))

    def pending_embeddings(self, limit: int = 0):
        ids = [i for i in self.inner.pending_embeddings(0) if self._mine(i)]
        ids = ids[:limit] if limit else ids
        return [self._unwrap(i) for i in ids]

    def clear_pending_embedding(self, node_id: str):
        self.inner.clear_pending_embedding(self._wrap(node_id))

    def register_callback(self, cb):
        def filtered(ev, obj):
            oid = getattr(obj, "id", None)
            if oid is not None and self._mine(oid):
                o = obj.copy()
                if isinstance(o, Edge):
                    self._unwrap_edge(o)
                else:
                    self._unwrap_node(o)
                cb(ev, o)
        self.inner.register_callback(filtered)

    def create_property_index(self, label, prop):
        return self.inner.create_property_index(label, prop)

    def drop_property_index(self, label, prop):
        return self.inner.drop_property_index(label, prop)

    def lookup_property_index(self, label, prop, value):
        r = self.inner.lookup_property_index(label, prop, value)
        if r is None:
            return None
        return [self._unwrap_node(n) for n in r if self._mine(n.id)]

    def flush(self):
        self.inner.flush()

    def close(self):
        pass  # shared base engine is closed by its owner
