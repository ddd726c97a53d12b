"""Storage adapter: applies replicated commands to an Engine.

Parity: reference pkg/replication/storage_adapter.go. Commands use the
same op vocabulary as the WAL so a replica's log replay is identical to
crash recovery.
"""

from __future__ import annotations

from typing import Any, Dict

from ..storage import wal as W
from ..storage.persistent import _edge_from_wire, _node_from_wire
from ..storage.types import Engine, StorageError


def command_for(op: int, payload: Dict[str, Any]) -> Dict[str, Any]:
    return {"op": op, "payload": payload}


class StorageAdapter:
    def __init__(self, engine: Engine):
        self.engine = engine

    def apply(self, command: Dict[str, Any]) -> None:
        op = command["op"]
        p = command["payload"]
        try:
            if op == W.OP_CREATE_NODE:
                self.engine.create_node(_node_from_wire(p))
            elif op == W.OP_UPDATE_NODE:
                self.engine.update_node(_node_from_wire(p))
            elif op == W.OP_DELETE_NODE:
                self.engine.delete_node(p["id"])
            elif op == W.OP_DETACH_DELETE:
                self.engine.detach_delete_node(p["id"])
            elif op == W.OP_CREATE_EDGE:
                self.engine.create_edge(_edge_from_wire(p))
            elif op == W.OP_UPDATE_EDGE:
                self.engine.update_edge(_edge_from_wire(p))
            elif op == W.OP_DELETE_EDGE:
                self.engine.delete_edge(p["id"])
        except StorageError:
            pass  # replicated replay is idempotence-tolerant
