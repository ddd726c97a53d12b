"""Core storage types and the Engine interface.

Parity target: reference pkg/storage/types.go (Node/Edge at :185,:298,
Engine interface at :355-415, event callbacks at :431-461). The rebuild
keeps the same layering rule: storage is the single source of truth and
search indexes stay in sync via event callbacks registered on the innermost
engine.
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, Iterator, List, Optional


@dataclass
class Node:
    id: str
    labels: List[str] = field(default_factory=list)
    properties: Dict[str, Any] = field(default_factory=dict)
    embedding: Optional[List[float]] = None
    created_at: float = 0.0
    updated_at: float = 0.0

    def copy(self) -> "Node":
        return Node(
            id=self.id,
            labels=list(self.labels),
            properties=dict(self.properties),
            embedding=list(self.embedding) if self.embedding is not None else None,
            created_at=self.created_at,
            updated_at=self.updated_at,
        )


@dataclass
class Edge:
    id: str
    type: str
    start_node: str
    end_node: str
    properties: Dict[str, Any] = field(default_factory=dict)
    created_at: float = 0.0
    updated_at: float = 0.0

    def copy(self) -> "Edge":
        return Edge(
            id=self.id,
            type=self.type,
            start_node=self.start_node,
            end_node=self.end_node,
            properties=dict(self.properties),
            created_at=self.created_at,
            updated_at=self.updated_at,
        )


class StorageError(Exception):
    pass


class NotFoundError(StorageError):
    pass


class ConstraintViolation(StorageError):
    pass


class EventType:
    NODE_CREATED = "node_created"
    NODE_UPDATED = "node_updated"
    NODE_DELETED = "node_deleted"
    EDGE_CREATED = "edge_created"
    EDGE_UPDATED = "edge_updated"
    EDGE_DELETED = "edge_deleted"


class Engine:
    """Abstract storage engine (reference pkg/storage/types.go:355-415).

    All mutating methods raise on error; reads return copies so callers can
    mutate freely.
    """

    # --- nodes ---
    def create_node(self, node: Node) -> Node: raise NotImplementedError
    def get_node(self, node_id: str) -> Node: raise NotImplementedError
    def update_node(self, node: Node) -> Node: raise NotImplementedError
    def delete_node(self, node_id: str) -> None: raise NotImplementedError
    def get_nodes_by_label(self, label: str) -> List[Node]: raise NotImplementedError
    def all_nodes(self) -> Iterator[Node]: raise NotImplementedError
    def node_count(self) -> int: raise NotImplementedError

    # lazy label-scan primitives: engines override to avoid copying the
    # whole label set when the caller wants a page or a count (GraphQL
    # nodes(limit:)/nodeCount resolvers). Defaults fall back to the
    # eager list so every Engine keeps working unchanged.
    def iter_nodes_by_label(self, label: str) -> Iterator[Node]:
        return iter(self.get_nodes_by_label(label))

    def node_ids_by_label(self, label: str) -> List[str]:
        return [n.id for n in self.get_nodes_by_label(label)]

    def node_count_by_label(self, label: str) -> int:
        return len(self.node_ids_by_label(label))
    def has_node(self, node_id: str) -> bool:
        try:
            self.get_node(node_id)
            return True
        except NotFoundError:
            return False

    # --- edges ---
    def create_edge(self, edge: Edge) -> Edge: raise NotImplementedError
    def get_edge(self, edge_id: str) -> Edge: raise NotImplementedError
    def update_edge(self, edge: Edge) -> Edge: raise NotImplementedError
    def delete_edge(self, edge_id: str) -> None: raise NotImplementedError
    def get_edges_by_type(self, edge_type: str) -> List[Edge]: raise NotImplementedError
    def all_edges(self) -> Iterator[Edge]: raise NotImplementedError
    def edge_count(self) -> int: raise NotImplementedError

    # --- adjacency ---
    def get_out_edges(self, node_id: str) -> List[Edge]: raise NotImplementedError
    def get_in_edges(self, node_id: str) -> List[Edge]: raise NotImplementedError
    def neighbors(self, node_id: str) -> List[str]: raise NotImplementedError

    # --- embedding queue (reference badger.go:24 pending_embed prefix) ---
    def mark_pending_embedding(self, node_id: str) -> None: raise NotImplementedError
    def pending_embeddings(self, limit: int = 0) -> List[str]: raise NotImplementedError
    def clear_pending_embedding(self, node_id: str) -> None: raise NotImplementedError

    # --- events ---
    def register_callback(self, cb: Callable[[str, Any], None]) -> None:
        raise NotImplementedError

    # --- lifecycle ---
    def flush(self) -> None: pass
    def close(self) -> None: pass


_id_lock = threading.Lock()
_id_counter = [0]


def new_id(prefix: str = "n") -> str:
    """Monotonic unique id: <prefix><epoch_ms>-<counter>."""
    with _id_lock:
        _id_counter[0] += 1
        c = _id_counter[0]
    return f"{prefix}{int(time.time() * 1000):x}-{c:x}"
