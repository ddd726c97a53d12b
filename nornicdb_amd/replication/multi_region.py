"""Multi-region replication: per-region Raft cluster + asynchronous
cross-region WAL streaming.

Parity: reference pkg/replication/multi_region.go:11-30 — each region
runs its own consensus group; a designated regional primary streams
committed commands to peer regions asynchronously (eventual consistency
across regions, strong within a region).
"""

from __future__ import annotations

import threading
from typing import Callable, Dict, List, Optional

from .ha import HAPrimary, HAStandby
from .raft import RaftNode


class Region:
    """One region: a Raft cluster whose leader also forwards committed
    commands to remote-region receivers."""

    def __init__(self, name: str, node_ids: List[str], transport,
                 apply_fn: Callable, remote_regions: List[str] = None,
                 region_transport=None):
        self.name = name
        self.apply_fn = apply_fn
        self.remote_regions = remote_regions or []
        self.region_transport = region_transport or transport
        self._seq = 0
        self._lock = threading.Lock()

        def apply_and_stream(cmd):
            apply_fn(cmd)
            self._stream(cmd)

        self.nodes = [RaftNode(nid, node_ids, transport,
                               apply_fn=apply_and_stream if i == 0 else apply_fn,
                               seed=i)
                      for i, nid in enumerate(node_ids)]

    def _stream(self, cmd):
        """Async cross-region push (only the applying leader's region
        streams; receivers dedupe by seq)."""
        leader = self.leader()
        if leader is None or leader is not self.nodes[0]:
            # stream from whichever node applied it — guarded by seq dedupe
            pass
        with self._lock:
            seq = self._seq
            self._seq += 1
        for r in self.remote_regions:
            self.region_transport.send(f"region:{r}", {
                "type": "xregion_entry", "from": f"region:{self.name}",
                "region": self.name, "seq": seq, "command": cmd})

    def leader(self) -> Optional[RaftNode]:
        for n in self.nodes:
            if n.is_leader:
                return n
        return None

    def tick_all(self):
        for n in self.nodes:
            n.tick()

    def propose(self, cmd) -> bool:
        l = self.leader()
        if l is None:
            return False
        return l.propose(cmd)


class RegionReceiver:
    """Applies cross-region entries exactly once per (region, seq)."""

    def __init__(self, region_name: str, transport, apply_fn: Callable):
        self.name = region_name
        self.apply_fn = apply_fn
        self._applied: Dict[str, int] = {}
        self._buffer: Dict[str, Dict[int, dict]] = {}
        self._lock = threading.Lock()
        transport.register(f"region:{region_name}", self._on_message)

    def _on_message(self, msg):
        if msg.get("type") != "xregion_entry":
            return
        src = msg["region"]
        seq = msg["seq"]
        with self._lock:
            nxt = self._applied.get(src, -1) + 1
            buf = self._buffer.setdefault(src, {})
            buf[seq] = msg["command"]
            while nxt in buf:
                try:
                    self.apply_fn(buf.pop(nxt))
                except Exception:
                    pass
                self._applied[src] = nxt
                nxt += 1

    def lag(self, src_region: str) -> int:
        with self._lock:
            return len(self._buffer.get(src_region, {}))
