"""HA primary/standby replication via WAL-op streaming + failover.

Parity: reference pkg/replication/ha_standby.go:12-30 (primary WALStreamer
pushes entries to the standby; standby promotes on primary failure) and
multi_region.go (per-region cluster + async cross-region streaming).
"""

from __future__ import annotations

import threading
import time
from typing import Any, Callable, Dict, List, Optional


class HAPrimary:
    """Streams storage commands to standbys; tracks acks."""

    def __init__(self, node_id: str, standbys: List[str], transport,
                 now_fn=time.monotonic):
        self.id = node_id
        self.standbys = standbys
        self.transport = transport
        self.now = now_fn
        self.seq = 0
        self._lock = threading.Lock()
        self.acks: Dict[str, int] = {s: -1 for s in standbys}
        transport.register(node_id, self._on_message)
        self._heartbeat_stop = threading.Event()

    def replicate(self, command: Dict[str, Any]) -> int:
        with self._lock:
            seq = self.seq
            self.seq += 1
        for s in self.standbys:
            self.transport.send(s, {"type": "wal_entry", "from": self.id,
                                    "seq": seq, "command": command})
        return seq

    def heartbeat(self):
        for s in self.standbys:
            self.transport.send(s, {"type": "ha_heartbeat", "from": self.id,
                                    "seq": self.seq})

    def _on_message(self, msg):
        if msg.get("type") == "wal_ack":
            with self._lock:
                self.acks[msg["from"]] = max(self.acks.get(msg["from"], -1),
                                             msg["seq"])

    def lag(self, standby: str) -> int:
        with self._lock:
            return self.seq - 1 - self.acks.get(standby, -1)


class HAStandby:
    """Applies streamed commands; promotes itself if the primary goes quiet."""

    PROMOTE_AFTER = 0.5  # seconds without heartbeat/entries

    def __init__(self, node_id: str, primary_id: str, transport,
                 apply_fn: Callable[[Dict[str, Any]], None],
                 now_fn=time.monotonic):
        self.id = node_id
        self.primary_id = primary_id
        self.transport = transport
        self.apply_fn = apply_fn
        self.now = now_fn
        self.applied_seq = -1
        self.promoted = False
        self._buffer: Dict[int, Dict] = {}
        self._last_heard = self.now()
        self._lock = threading.Lock()
        transport.register(node_id, self._on_message)

    def _on_message(self, msg):
        t = msg.get("type")
        with self._lock:
            if t == "wal_entry":
                self._last_heard = self.now()
                self._buffer[msg["seq"]] = msg["command"]
                # apply in order
                while self.applied_seq + 1 in self._buffer:
                    self.applied_seq += 1
                    cmd = self._buffer.pop(self.applied_seq)
                    try:
                        self.apply_fn(cmd)
                    except Exception:
                        pass
                self.transport.send(self.primary_id, {
                    "type": "wal_ack", "from": self.id,
                    "seq": self.applied_seq})
            elif t == "ha_heartbeat":
                self._last_heard = self.now()

    def check_failover(self) -> bool:
        """Promote if the primary has been silent too long."""
        with self._lock:
            if not self.promoted and \
                    self.now() - self._last_heard > self.PROMOTE_AFTER:
                self.promoted = True
        return self.promoted

    def health(self):
        with self._lock:
            return {"id": self.id, "role": "primary" if self.promoted else "standby",
                    "applied_seq": self.applied_seq,
                    "buffered": len(self._buffer)}
