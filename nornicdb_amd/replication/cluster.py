"""ClusterNode: a replicated NornicDB instance — Raft + TCP + engine.

Composes the pieces into the reference's deployment story
(reference pkg/replication/replicator.go + transport.go + cmd serve
--cluster): every node runs RaftNode over TcpTransport; writes are
proposed by the leader and applied on every member through
StorageAdapter (same op vocabulary as the WAL); a follower that
receives a write FORWARDS it to the current leader over the cluster
transport and waits for the committed apply (the reference forwards
via Bolt to the leader — same outcome, one fewer protocol hop).
Reads are always local (leader writes / follower reads).
"""

from __future__ import annotations

import threading
import time
import uuid
from collections import OrderedDict
from typing import Any, Dict, List, Optional, Tuple

from ..storage import wal as W
from ..storage.persistent import _edge_to_wire, _node_to_wire
from ..storage.types import Edge, Engine, Node, StorageError
from .adapter import StorageAdapter
from .raft import RaftNode
from .tcp import TcpTransport


class NotLeader(StorageError):
    def __init__(self, leader: Optional[str]):
        super().__init__(f"not the leader (leader: {leader})")
        self.leader = leader


class ClusterNode:
    """One member. `peers` maps node id -> (host, port) for the cluster
    port (reference :7688, separate from Bolt)."""

    WRITE_TIMEOUT = 5.0

    def __init__(self, node_id: str, peers: Dict[str, Tuple[str, int]],
                 engine: Engine, transport=None):
        self.id = node_id
        self.engine = engine
        self.adapter = StorageAdapter(engine)
        self.transport = transport or TcpTransport(node_id, peers)
        self._applied: Dict[str, threading.Event] = {}
        self._seen_ids: "OrderedDict[str, bool]" = OrderedDict()
        self._lock = threading.Lock()
        self.raft = RaftNode(node_id, sorted(peers), _RaftTap(self),
                             apply_fn=self._apply)
        self.replicated = ReplicatedEngine(engine, self)

    # ---- raft plumbing ----
    def start(self):
        self.raft.start()
        return self

    def stop(self):
        self.raft.stop()
        self.transport.close()

    def _on_message(self, msg: Dict[str, Any]):
        t = msg.get("type")
        if t == "fwd_write":
            # handle OFF the transport reader thread: propose-and-wait
            # needs the reader to keep delivering the follower's
            # append-acks, or the commit (and this handler) deadlock
            threading.Thread(target=self._handle_forward, args=(msg,),
                             daemon=True).start()
        elif t == "fwd_ack":
            with self._lock:
                ev = self._applied.get("fwd:" + msg["cmd_id"])
            if ev is not None:
                ev.ok = msg.get("ok", False)  # type: ignore[attr-defined]
                ev.set()
        else:
            self.raft.on_message(msg)

    def _handle_forward(self, msg: Dict[str, Any]):
        ok = False
        if self.raft.is_leader:
            ok = self._propose_and_wait(msg["command"], cmd_id=msg["cmd_id"])
        self.transport.send(msg["from"], {
            "type": "fwd_ack", "from": self.id,
            "cmd_id": msg["cmd_id"], "ok": ok,
            "leader": self.raft.leader_id})

    def _apply(self, command: Dict[str, Any]):
        self.adapter.apply(command)
        cid = command.get("cmd_id")
        if cid:
            with self._lock:
                self._seen_ids[cid] = True
                while len(self._seen_ids) > 4096:
                    self._seen_ids.popitem(last=False)
                ev = self._applied.get(cid)
            if ev is not None:
                ev.set()

    # ---- write path ----
    def _propose_and_wait(self, command: Dict[str, Any],
                          cmd_id: str = None) -> bool:
        cid = cmd_id or uuid.uuid4().hex
        command = dict(command, cmd_id=cid)
        ev = threading.Event()
        with self._lock:
            self._applied[cid] = ev
        try:
            if not self.raft.propose(command):
                return False
            return ev.wait(self.WRITE_TIMEOUT)
        finally:
            with self._lock:
                self._applied.pop(cid, None)

    def write(self, op: int, payload: Dict[str, Any]) -> None:
        """Replicated write from ANY member: leader proposes; follower
        forwards to the leader and waits for the committed apply."""
        command = {"op": op, "payload": payload}
        if self.raft.is_leader:
            if not self._propose_and_wait(command):
                raise StorageError("replication commit timeout")
            return
        leader = self.raft.leader_id
        if leader is None:
            # give an in-progress election a moment
            deadline = time.monotonic() + 2.0
            while leader is None and time.monotonic() < deadline:
                time.sleep(0.05)
                leader = self.raft.leader_id
            if leader is None:
                raise NotLeader(None)
        cid = uuid.uuid4().hex
        ev = threading.Event()
        ev.ok = False  # type: ignore[attr-defined]
        with self._lock:
            self._applied["fwd:" + cid] = ev
        try:
            self.transport.send(leader, {"type": "fwd_write", "from": self.id,
                                         "cmd_id": cid, "command": command})
            if not ev.wait(self.WRITE_TIMEOUT) or not ev.ok:  # type: ignore
                raise StorageError(f"write forward to {leader} failed")
            # leader committed; wait until OUR raft applied it locally so
            # the caller reads its own write
            deadline = time.monotonic() + self.WRITE_TIMEOUT
            while time.monotonic() < deadline:
                if self._has_applied(cid):
                    return
                time.sleep(0.01)
            raise StorageError("forwarded write not yet applied locally")
        finally:
            with self._lock:
                self._applied.pop("fwd:" + cid, None)

    def _has_applied(self, cid: str) -> bool:
        with self._lock:
            return cid in self._seen_ids

    def health(self) -> Dict[str, Any]:
        return self.raft.health()


class _RaftTap:
    """Transport facade handed to RaftNode: registers the cluster node's
    dispatcher (so fwd_* messages are intercepted) and sends through the
    real transport."""

    def __init__(self, node: ClusterNode):
        self.node = node

    def register(self, node_id, handler):
        self.node.transport.register(node_id, self.node._on_message)

    def send(self, dst, msg):
        self.node.transport.send(dst, msg)


class ReplicatedEngine:
    """Engine facade (duck-typed, NOT an Engine subclass — the base
    class's NotImplementedError stubs would shadow __getattr__
    delegation): mutations replicate through the cluster; reads pass
    through to the local engine (leader writes / follower reads)."""

    def __init__(self, local: Engine, node: ClusterNode):
        self.local = local
        self.node = node

    # ---- replicated writes ----
    def create_node(self, n: Node) -> Node:
        self.node.write(W.OP_CREATE_NODE, _node_to_wire(n))
        return self.local.get_node(n.id)

    def update_node(self, n: Node) -> Node:
        self.node.write(W.OP_UPDATE_NODE, _node_to_wire(n))
        return self.local.get_node(n.id)

    def delete_node(self, nid: str) -> None:
        self.node.write(W.OP_DELETE_NODE, {"id": nid})

    def detach_delete_node(self, nid: str) -> None:
        self.node.write(W.OP_DETACH_DELETE, {"id": nid})

    def create_edge(self, e: Edge) -> Edge:
        self.node.write(W.OP_CREATE_EDGE, _edge_to_wire(e))
        return self.local.get_edge(e.id)

    def update_edge(self, e: Edge) -> Edge:
        self.node.write(W.OP_UPDATE_EDGE, _edge_to_wire(e))
        return self.local.get_edge(e.id)

    def delete_edge(self, eid: str) -> None:
        self.node.write(W.OP_DELETE_EDGE, {"id": eid})

    # ---- local reads / passthrough ----
    def __getattr__(self, item):
        return getattr(self.local, item)
