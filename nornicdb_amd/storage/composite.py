"""Composite engine: Neo4j-Fabric-style routing across constituent
databases.

Parity: reference pkg/storage/composite_engine.go (:14-40) — reads fan
out over all constituents; writes route by label -> constituent mapping
(default constituent otherwise). IDs are globally prefixed with the
constituent name so cross-constituent reads stay unambiguous.
"""

from __future__ import annotations

from typing import Dict, Iterator, List, Optional

from .types import Edge, Engine, Node, NotFoundError


class CompositeEngine(Engine):
    def __init__(self, constituents: Dict[str, Engine], default: str,
                 label_routes: Dict[str, str] = None):
        assert default in constituents
        self.parts = constituents
        self.default = default
        self.routes = dict(label_routes or {})

    # ---- routing ----
    def _route_for_labels(self, labels) -> str:
        for lb in labels or []:
            if lb in self.routes:
                return self.routes[lb]
        return self.default

    def _part_of(self, gid: str):
        if ":" in gid:
            name, local = gid.split(":", 1)
            if name in self.parts:
                return name, local
        return self.default, gid

    @staticmethod
    def _gid(name, local):
        return f"{name}:{local}"

    def _wrap_node(self, name: str, n: Node) -> Node:
        c = n.copy()
        c.id = self._gid(name, c.id)
        return c

    def _wrap_edge(self, name: str, e: Edge) -> Edge:
        c = e.copy()
        c.id = self._gid(name, c.id)
        c.start_node = self._gid(name, c.start_node)
        c.end_node = self._gid(name, c.end_node)
        return c

    # ---- nodes ----
    def create_node(self, node: Node) -> Node:
        name = self._route_for_labels(node.labels)
        local = node.copy()
        if ":" in local.id:
            name, local.id = self._part_of(local.id)
        created = self.parts[name].create_node(local)
        return self._wrap_node(name, created)

    def get_node(self, node_id: str) -> Node:
        name, local = self._part_of(node_id)
        return self._wrap_node(name, self.parts[name].get_node(local))

    def update_node(self, node: Node) -> Node:
        name, local_id = self._part_of(node.id)
        local = node.copy()
        local.id = local_id
        return self._wrap_node(name, self.parts[name].update_node(local))

    def delete_node(self, node_id: str) -> None:
        name, local = self._part_of(node_id)
        self.parts[name].delete_node(local)

    def detach_delete_node(self, node_id: str) -> None:
        name, local = self._part_of(node_id)
        self.parts[name].detach_delete_node(local)

    def get_nodes_by_label(self, label: str) -> List[Node]:
        out = []
        for name, eng in self.parts.items():
            out.extend(self._wrap_node(name, n)
                       for n in eng.get_nodes_by_label(label))
        return out

    def all_nodes(self) -> Iterator[Node]:
        for name, eng in self.parts.items():
            for n in eng.all_nodes():
                yield self._wrap_node(name, n)

    def node_count(self) -> int:
        return sum(e.node_count() for e in self.parts.values())

    # ---- edges (within one constituent only, like Fabric) ----
    def create_edge(self, edge: Edge) -> Edge:
        sname, slocal = self._part_of(edge.start_node)
        tname, tlocal = self._part_of(edge.end_node)
        if sname != tname:
            raise NotFoundError(
                "composite: relationships cannot span constituents "
                f"({sname} -> {tname})")
        local = edge.copy()
        local.id = self._part_of(local.id)[1]
        local.start_node, local.end_node = slocal, tlocal
        return self._wrap_edge(sname, self.parts[sname].create_edge(local))

    def get_edge(self, edge_id: str) -> Edge:
        name, local = self._part_of(edge_id)
        return self._wrap_edge(name, self.parts[name].get_edge(local))

    def update_edge(self, edge: Edge) -> Edge:
        name, _ = self._part_of(edge.id)
        local = edge.copy()
        local.id = self._part_of(local.id)[1]
        local.start_node = self._part_of(local.start_node)[1]
        local.end_node = self._part_of(local.end_node)[1]
        return self._wrap_edge(name, self.parts[name].update_edge(local))

    def delete_edge(self, edge_id: str) -> None:
        name, local = self._part_of(edge_id)
        self.parts[name].delete_edge(local)

    def get_edges_by_type(self, edge_type: str) -> List[Edge]:
        out = []
        for name, eng in self.parts.items():
            out.extend(self._wrap_edge(name, e)
                       for e in eng.get_edges_by_type(edge_type))
        return out

    def all_edges(self) -> Iterator[Edge]:
        for name, eng in self.parts.items():
            for e in eng.all_edges():
                yield self._wrap_edge(name, e)

    def edge_count(self) -> int:
        return sum(e.edge_count() for e in self.parts.values())

    # ---- adjacency ----
    def get_out_edges(self, node_id: str) -> List[Edge]:
        name, local = self._part_of(node_id)
        return [self._wrap_edge(name, e)
                for e in self.parts[name].get_out_edges(local)]

    def get_in_edges(self, node_id: str) -> List[Edge]:
        name, local = self._part_of(node_id)
        return [self._wrap_edge(name, e)
                for e in self.parts[name].get_in_edges(local)]

    def neighbors(self, node_id: str) -> List[str]:
        name, local = self._part_of(node_id)
        return [self._gid(name, i) for i in self.parts[name].neighbors(local)]

    # ---- misc passthrough ----
    def mark_pending_embedding(self, node_id: str) -> None:
        name, local = self._part_of(node_id)
        self.parts[name].mark_pending_embedding(local)

    def pending_embeddings(self, limit: int = 0) -> List[str]:
        out = []
        for name, eng in self.parts.items():
            out.extend(self._gid(name, i) for i in eng.pending_embeddings(0))
        return out[:limit] if limit else out

    def clear_pending_embedding(self, node_id: str) -> None:
        name, local = self._part_of(node_id)
        self.parts[name].clear_pending_embedding(local)

    def register_callback(self, cb):
        for name, eng in self.parts.items():
            def wrapped(ev, obj, _name=name):
                o = obj.copy()
                if isinstance(o, Edge):
                    o = self._wrap_edge(_name, obj)
                else:
                    o = self._wrap_node(_name, obj)
                cb(ev, o)
            eng.register_callback(wrapped)

    def flush(self):
        for e in self.parts.values():
            e.flush()
