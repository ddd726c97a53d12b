"""Explicit-transaction support: an undo-recording engine wrapper.

The reference backs BEGIN/COMMIT/ROLLBACK with BadgerTransaction
(pkg/cypher/transaction.go:119 handleRollback, pkg/storage badger tx).
Here the same contract is provided by compensation: mutations apply to
the live engine immediately (so statements inside the transaction read
their own writes with zero machinery), while an undo log records the
inverse of every mutation; ROLLBACK replays the log in reverse.

Isolation note (documented behavior): other sessions observe writes
before commit (read-uncommitted) — rollback correctness is what the
Neo4j drivers' `tx.rollback()` / context-manager error paths rely on.
CALL procedures that mutate through side channels are not recorded.
"""

from __future__ import annotations

from typing import Any, Callable, List

from .types import Edge, Node, NotFoundError


class TxRecorder:
    """Engine facade recording inverse operations for rollback.

    Duck-typed like ReplicatedEngine (NOT an Engine subclass — the base
    class's NotImplementedError stubs would shadow __getattr__)."""

    def __init__(self, engine):
        self.engine = engine
        self._undo: List[Callable[[], None]] = []
        self.closed = False

    # ---- mutations (record inverse, then delegate) ----
    def create_node(self, node: Node) -> Node:
        out = self.engine.create_node(node)
        nid = out.id
        self._undo.append(lambda: self.engine.delete_node(nid))
        return out

    def update_node(self, node: Node) -> Node:
        old = self.engine.get_node(node.id)
        out = self.engine.update_node(node)
        self._undo.append(lambda: self.engine.update_node(old))
        return out

    def delete_node(self, node_id: str) -> None:
        old = self.engine.get_node(node_id)
        self.engine.delete_node(node_id)
        self._undo.append(lambda: self.engine.create_node(old))

    def detach_delete_node(self, node_id: str) -> None:
        old = self.engine.get_node(node_id)
        edges = list(self.engine.get_out_edges(node_id)) + \
            list(self.engine.get_in_edges(node_id))
        self.engine.detach_delete_node(node_id)

        def undo():
            self.engine.create_node(old)
            seen = set()
            for e in edges:
                if e.id in seen:
                    continue
                seen.add(e.id)
                try:
                    self.engine.create_edge(e)
                except Exception:
                    pass   # other endpoint also gone (deleted later in tx)
        self._undo.append(undo)

    def create_edge(self, edge: Edge) -> Edge:
        out = self.engine.create_edge(edge)
        eid = out.id
        self._undo.append(lambda: self.engine.delete_edge(eid))
        return out

    def update_edge(self, edge: Edge) -> Edge:
        old = self.engine.get_edge(edge.id)
        out = self.engine.update_edge(edge)
        self._undo.append(lambda: self.engine.update_edge(old))
        return out

    def delete_edge(self, edge_id: str) -> None:
        old = self.engine.get_edge(edge_id)
        self.engine.delete_edge(edge_id)
        self._undo.append(lambda: self.engine.create_edge(old))

    # ---- lifecycle ----
    def commit(self) -> None:
        self._undo.clear()
        self.closed = True

    def rollback(self) -> int:
        """Apply inverses in reverse order; returns ops undone."""
        n = 0
        for undo in reversed(self._undo):
            try:
                undo()
                n += 1
            except NotFoundError:
                pass   # state already reverted by a later inverse
            except Exception:
                pass
        self._undo.clear()
        self.closed = True
        return n

    # ---- reads / passthrough ----
    def __getattr__(self, item: str) -> Any:
        return getattr(self.engine, item)
