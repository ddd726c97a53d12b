"""Replication transports: in-process message bus + chaos wrapper.

Parity: reference pkg/replication/transport.go (cluster TCP :7688) and
chaos_test.go:22-54 (ChaosConfig: packet loss/duplicate/reorder, latency
jitter, connection drops) — the chaos wrapper here is a first-class
library feature so multi-node logic is testable in-process, exactly like
the reference's chaos transport tests.
"""

from __future__ import annotations

import queue
import random
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional


class Transport:
    """Point-to-point message transport: send(dst, msg); register handler."""

    def send(self, dst: str, msg: Dict[str, Any]) -> None:
        raise NotImplementedError

    def register(self, node_id: str, handler: Callable[[Dict[str, Any]], None]):
        raise NotImplementedError


class InProcTransport(Transport):
    """Shared bus for in-process clusters; delivery on a pump thread."""

    def __init__(self):
        self._handlers: Dict[str, Callable] = {}
        self._q: "queue.Queue" = queue.Queue()
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._pump, daemon=True)
        self._thread.start()

    def register(self, node_id, handler):
        self._handlers[node_id] = handler

    def unregister(self, node_id):
        self._handlers.pop(node_id, None)

    def send(self, dst, msg):
        self._q.put((dst, msg))

    def _pump(self):
        while not self._stop.is_set():
            try:
                dst, msg = self._q.get(timeout=0.05)
            except queue.Empty:
                continue
            h = self._handlers.get(dst)
            if h:
                try:
                    h(msg)
                except Exception:
                    pass

    def close(self):
        self._stop.set()
        self._thread.join(timeout=1)


@dataclass
class ChaosConfig:
    drop_rate: float = 0.0
    duplicate_rate: float = 0.0
    reorder_rate: float = 0.0
    max_delay_s: float = 0.0
    partition: frozenset = frozenset()   # node ids cut off from the rest
    seed: int = 0


class ChaosTransport(Transport):
    """Wraps a transport with fault injection (reference chaos_test.go)."""

    def __init__(self, inner: Transport, config: ChaosConfig = None):
        self.inner = inner
        self.cfg = config or ChaosConfig()
        self._rng = random.Random(self.cfg.seed)
        self._delayed: List = []
        self._lock = threading.Lock()
        self._src_of: Dict[int, str] = {}

    def register(self, node_id, handler):
        self.inner.register(node_id, handler)

    def set_partition(self, nodes):
        self.cfg.partition = frozenset(nodes)

    def send(self, dst, msg):
        src = msg.get("from", "")
        part = self.cfg.partition
        if part and ((src in part) != (dst in part)):
            return  # across the partition: dropped
        if self._rng.random() < self.cfg.drop_rate:
            return
        sends = 1
        if self._rng.random() < self.cfg.duplicate_rate:
            sends = 2
        for _ in range(sends):
            if self.cfg.max_delay_s > 0 and self._rng.random() < self.cfg.reorder_rate:
                delay = self._rng.uniform(0, self.cfg.max_delay_s)
                t = threading.Timer(delay, self.inner.send, args=(dst, msg))
                t.daemon = True
                t.start()
            else:
                self.inner.send(dst, msg)
