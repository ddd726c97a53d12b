"""Persistent engine: MemoryEngine state + WAL + snapshot compaction.

This is the rebuild's replacement for the reference's BadgerDB + WALEngine
stack (reference pkg/storage/badger.go + wal_engine.go:16-40; snapshot
compaction every 5 min per wal_engine.go:120-149). Data lives in RAM (the
working set the reference also keeps hot), durability comes from the WAL,
and restart = load snapshot + replay WAL with per-record CRC checks.

Transactions: begin() returns a Transaction buffering mutations; commit
writes TX_BEGIN .. ops .. TX_COMMIT to the WAL and applies atomically.
Replay ignores uncommitted transaction bodies (reference
wal.go RecoverWithTransactions :1845).
"""

from __future__ import annotations

import os
import threading
import time
from typing import Optional

import msgpack

from . import codec as _codec

from .memory import MemoryEngine
from .types import Edge, Node, StorageError
from . import wal as W


def _node_to_wire(n: Node):
    return {"id": n.id, "labels": n.labels, "props": n.properties,
            "emb": n.embedding, "ca": n.created_at, "ua": n.updated_at}


def _node_from_wire(d) -> Node:
    return Node(d["id"], d["labels"], d["props"], d.get("emb"),
                d.get("ca", 0.0), d.get("ua", 0.0))


def _edge_to_wire(e: Edge):
    return {"id": e.id, "type": e.type, "s": e.start_node, "t": e.end_node,
            "props": e.properties, "ca": e.created_at, "ua": e.updated_at}


def _edge_from_wire(d) -> Edge:
    return Edge(d["id"], d["type"], d["s"], d["t"], d["props"],
                d.get("ca", 0.0), d.get("ua", 0.0))


class Transaction:
    """Buffered transaction; applied atomically on commit."""

    def __init__(self, eng: "PersistentEngine"):
        self._eng = eng
        self._ops = []
        self._done = False

    def create_node(self, node: Node):
        self._ops.append((W.OP_CREATE_NODE, _node_to_wire(node)))
        return node

    def update_node(self, node: Node):
        self._ops.append((W.OP_UPDATE_NODE, _node_to_wire(node)))
        return node

    def delete_node(self, node_id: str):
        self._ops.append((W.OP_DELETE_NODE, {"id": node_id}))

    def detach_delete_node(self, node_id: str):
        self._ops.append((W.OP_DETACH_DELETE, {"id": node_id}))

    def create_edge(self, edge: Edge):
        self._ops.append((W.OP_CREATE_EDGE, _edge_to_wire(edge)))
        return edge

    def update_edge(self, edge: Edge):
        self._ops.append((W.OP_UPDATE_EDGE, _edge_to_wire(edge)))
        return edge

    def delete_edge(self, edge_id: str):
        self._ops.append((W.OP_DELETE_EDGE, {"id": edge_id}))

    def commit(self):
        if self._done:
            raise StorageError("transaction already finished")
        self._done = True
        self._eng._commit_tx(self._ops)

    def rollback(self):
        self._done = True
        self._ops = []


class PersistentEngine(MemoryEngine):
    SNAPSHOT = "snapshot.bin"
    WAL_FILE = "wal.log"

    def __init__(self, data_dir: str, sync_on_write: bool = False,
                 snapshot_interval: float = 300.0,
                 snapshot_wal_bytes: int = 64 << 20,
                 replay_embeddings: bool = True):
        super().__init__()
        self.data_dir = data_dir
        os.makedirs(data_dir, exist_ok=True)
        self._tx_lock = threading.Lock()
        self._replay(replay_embeddings)
        self._wal = W.WAL(os.path.join(data_dir, self.WAL_FILE),
                          sync_on_write=sync_on_write)
        self._snapshot_interval = snapshot_interval
        self._snapshot_wal_bytes = snapshot_wal_bytes
        self._stop = threading.Event()
        self._compactor = threading.Thread(target=self._compact_loop, daemon=True)
        self._compactor.start()

    # ---- recovery ----
    def _replay(self, replay_embeddings: bool):
        snap_path = os.path.join(self.data_dir, self.SNAPSHOT)
        if os.path.exists(snap_path):
            with open(snap_path, "rb") as f:
                state = msgpack.unpackb(f.read(), raw=False, strict_map_key=False,
                                        object_hook=_codec.object_hook)
            self.load_state(state)
        wal_path = os.path.join(self.data_dir, self.WAL_FILE)
        # two-phase: collect committed tx ids, then apply
        records = list(W.WAL.replay(wal_path))
        committed = {p["tx"] for op, p in records if op == W.OP_TX_COMMIT}
        cur_tx = None
        for op, p in records:
            if op == W.OP_TX_BEGIN:
                cur_tx = p["tx"]
                continue
            if op in (W.OP_TX_COMMIT, W.OP_TX_ABORT):
                cur_tx = None
                continue
            if cur_tx is not None and cur_tx not in committed:
                continue  # uncommitted tx body
            self._apply(op, p, replay_embeddings)

    def _apply(self, op, p, replay_embeddings=True, strict=False):
        try:
            self._apply_inner(op, p, replay_embeddings)
        except StorageError:
            # replay is idempotence-tolerant (reference recovery behavior)
            if strict:
                raise

    def _apply_inner(self, op, p, replay_embeddings=True):
        if True:
            if op == W.OP_CREATE_NODE:
                MemoryEngine.create_node(self, _node_from_wire(p))
            elif op == W.OP_UPDATE_NODE:
                MemoryEngine.update_node(self, _node_from_wire(p))
            elif op == W.OP_DELETE_NODE:
                MemoryEngine.delete_node(self, p["id"])
            elif op == W.OP_DETACH_DELETE:
                MemoryEngine.detach_delete_node(self, p["id"])
            elif op == W.OP_CREATE_EDGE:
                MemoryEngine.create_edge(self, _edge_from_wire(p))
            elif op == W.OP_UPDATE_EDGE:
                MemoryEngine.update_edge(self, _edge_from_wire(p))
            elif op == W.OP_DELETE_EDGE:
                MemoryEngine.delete_edge(self, p["id"])
            elif op == W.OP_MARK_PENDING:
                MemoryEngine.mark_pending_embedding(self, p["id"])
            elif op == W.OP_CLEAR_PENDING:
                MemoryEngine.clear_pending_embedding(self, p["id"])
            elif op == W.OP_UPDATE_EMBEDDING:
                if replay_embeddings:
                    try:
                        n = MemoryEngine.get_node(self, p["id"])
                        n.embedding = p["emb"]
                        MemoryEngine.update_node(self, n)
                    except StorageError:
                        pass
                else:
                    MemoryEngine.mark_pending_embedding(self, p["id"])

    # ---- snapshot ----
    def snapshot(self):
        """Write a consistent snapshot and truncate the WAL."""
        with self._tx_lock:
            state = self.dump_state()
            tmp = os.path.join(self.data_dir, self.SNAPSHOT + ".tmp")
            with open(tmp, "wb") as f:
                f.write(msgpack.packb(state, use_bin_type=True,
                                      default=_codec.default))
                f.flush()
                os.fsync(f.fileno())
            os.replace(tmp, os.path.join(self.data_dir, self.SNAPSHOT))
            self._wal.truncate()

    def _compact_loop(self):
        last = time.time()
        while not self._stop.wait(1.0):
            try:
                if (time.time() - last >= self._snapshot_interval
                        or self._wal.size() >= self._snapshot_wal_bytes):
                    self.snapshot()
                    last = time.time()
            except Exception:
                pass

    # ---- logged mutations ----
    def create_node(self, node: Node) -> Node:
        with self._tx_lock:
            self._wal.append(W.OP_CREATE_NODE, _node_to_wire(node))
            return MemoryEngine.create_node(self, node)

    def update_node(self, node: Node) -> Node:
        with self._tx_lock:
            self._wal.append(W.OP_UPDATE_NODE, _node_to_wire(node))
            return MemoryEngine.update_node(self, node)

    def update_embedding(self, node_id: str, embedding) -> None:
        """Separately-logged embedding write (skippable on replay)."""
        with self._tx_lock:
            self._wal.append(W.OP_UPDATE_EMBEDDING, {"id": node_id, "emb": embedding})
            n = MemoryEngine.get_node(self, node_id)
            n.embedding = embedding
            MemoryEngine.update_node(self, n)

    def delete_node(self, node_id: str) -> None:
        with self._tx_lock:
            self._wal.append(W.OP_DELETE_NODE, {"id": node_id})
            MemoryEngine.delete_node(self, node_id)

    def detach_delete_node(self, node_id: str) -> None:
        with self._tx_lock:
            self._wal.append(W.OP_DETACH_DELETE, {"id": node_id})
            MemoryEngine.detach_delete_node(self, node_id)

    def create_edge(self, edge: Edge) -> Edge:
        with self._tx_lock:
            self._wal.append(W.OP_CREATE_EDGE, _edge_to_wire(edge))
            return MemoryEngine.create_edge(self, edge)

    def update_edge(self, edge: Edge) -> Edge:
        with self._tx_lock:
            self._wal.append(W.OP_UPDATE_EDGE, _edge_to_wire(edge))
            return MemoryEngine.update_edge(self, edge)

    def delete_edge(self, edge_id: str) -> None:
        with self._tx_lock:
            self._wal.append(W.OP_DELETE_EDGE, {"id": edge_id})
            MemoryEngine.delete_edge(self, edge_id)

    def mark_pending_embedding(self, node_id: str) -> None:
        with self._tx_lock:
            self._wal.append(W.OP_MARK_PENDING, {"id": node_id})
            MemoryEngine.mark_pending_embedding(self, node_id)

    def clear_pending_embedding(self, node_id: str) -> None:
        with self._tx_lock:
            self._wal.append(W.OP_CLEAR_PENDING, {"id": node_id})
            MemoryEngine.clear_pending_embedding(self, node_id)

    # ---- transactions ----
    _tx_counter = 0

    def begin(self) -> Transaction:
        return Transaction(self)

    def _commit_tx(self, ops):
        with self._tx_lock:
            PersistentEngine._tx_counter += 1
            tx = f"tx{PersistentEngine._tx_counter}-{os.getpid()}-{time.time_ns()}"
            self._wal.append(W.OP_TX_BEGIN, {"tx": tx})
            for op, payload in ops:
                self._wal.append(op, payload)
            # apply after logging body; commit marker only if all applied
            applied = []
            try:
                for op, payload in ops:
                    self._apply_commit(op, payload)
                    applied.append((op, payload))
            except StorageError:
                self._wal.append(W.OP_TX_ABORT, {"tx": tx})
                # roll back applied ops in reverse
                for op, payload in reversed(applied):
                    self._undo(op, payload)
                raise
            self._wal.append(W.OP_TX_COMMIT, {"tx": tx})
            self._wal.sync()

    def _apply_commit(self, op, payload):
        # live commits must surface conflicts (strict), unlike replay
        self._apply(op, payload, replay_embeddings=True, strict=True)

    def _undo(self, op, p):
        try:
            if op == W.OP_CREATE_NODE:
                MemoryEngine.delete_node(self, p["id"])
            elif op == W.OP_CREATE_EDGE:
                MemoryEngine.delete_edge(self, p["id"])
        except StorageError:
            pass

    def flush(self):
        self._wal.sync()

    def close(self):
        self._stop.set()
        self._compactor.join(timeout=2)
        try:
            self.snapshot()
        except Exception:
            pass
        self._wal.close()
