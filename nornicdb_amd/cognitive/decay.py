"""Memory decay: tiered half-lives, composite scoring, reinforcement,
archival.

Parity: reference pkg/decay/decay.go — half-lives episodic 7d /
semantic 69d / procedural 693d, score = recency + frequency + importance,
reinforcement on access, archival below threshold; optional
Kalman-filtered variant (kalman_adapter.go).
"""

from __future__ import annotations

import math
import time
from dataclasses import dataclass
from typing import Dict, Optional

from ..storage.types import Engine, Node
from .kalman import Kalman1D

DAY = 86400.0
HALF_LIVES = {
    "episodic": 7 * DAY,
    "semantic": 69 * DAY,
    "procedural": 693 * DAY,
}


@dataclass
class DecayConfig:
    archive_threshold: float = 0.1
    delete_threshold: float = 0.02
    recency_weight: float = 0.5
    frequency_weight: float = 0.3
    importance_weight: float = 0.2
    reinforcement_boost: float = 0.1
    use_kalman: bool = False


class DecayManager:
    def __init__(self, engine: Engine, config: DecayConfig = None,
                 now_fn=time.time):
        self.engine = engine
        self.cfg = config or DecayConfig()
        self.now = now_fn
        self._kalman: Dict[str, Kalman1D] = {}

    def score(self, node: Node, now: Optional[float] = None) -> float:
        """Composite decay score in [0, ~1]."""
        now = now or self.now()
        p = node.properties
        tier = p.get("memory_type", "episodic")
        half_life = HALF_LIVES.get(tier, HALF_LIVES["episodic"])
        last = p.get("last_accessed") or p.get("created_at") or now
        age = max(now - last, 0.0)
        recency = math.pow(0.5, age / half_life)
        freq = 1.0 - math.exp(-0.2 * p.get("access_count", 0))
        importance = float(p.get("importance", 0.5))
        s = (self.cfg.recency_weight * recency
             + self.cfg.frequency_weight * freq
             + self.cfg.importance_weight * importance)
        if self.cfg.use_kalman:
            kf = self._kalman.setdefault(node.id, Kalman1D(initial=s))
            s = kf.update(s)
        return s

    def reinforce(self, node_id: str) -> float:
        """Boost importance on access (reference reinforcement)."""
        node = self.engine.get_node(node_id)
        imp = min(1.0, float(node.properties.get("importance", 0.5))
                  + self.cfg.reinforcement_boost)
        node.properties["importance"] = imp
        node.properties["access_count"] = node.properties.get("access_count", 0) + 1
        node.properties["last_accessed"] = self.now()
        self.engine.update_node(node)
        return imp

    def start(self, interval_s: float = 3600.0) -> None:
        """Background recalculation ticker (reference pkg/decay
        decay.go:643 Manager.Start, RecalculateInterval default 1h).
        Errors are swallowed per cycle; call stop() before close."""
        import threading
        if getattr(self, "_bg", None) is not None:
            return
        self._stop = threading.Event()

        def loop():
            while not self._stop.wait(interval_s):
                try:
                    self.run_cycle()
                except Exception:
                    pass   # keep ticking (reference: log and continue)

        self._bg = threading.Thread(target=loop, daemon=True,
                                    name="decay-recalc")
        self._bg.start()

    def stop(self) -> None:
        if getattr(self, "_bg", None) is not None:
            self._stop.set()
            self._bg.join(timeout=5)
            self._bg = None

    def run_cycle(self) -> Dict[str, int]:
        """Score all Memory nodes; archive / delete below thresholds."""
        now = self.now()
        stats = {"scored": 0, "archived": 0, "deleted": 0}
        for node in self.engine.get_nodes_by_label("Memory"):
            if "Pinned" in node.labels:
                continue
            s = self.score(node, now)
            stats["scored"] += 1
            node.properties["decay_score"] = s
            if s < self.cfg.delete_threshold:
                self.engine.detach_delete_node(node.id)
                stats["deleted"] += 1
                continue
            if s < self.cfg.archive_threshold and "Archived" not in node.labels:
                node.labels.append("Archived")
                stats["archived"] += 1
            self.engine.update_node(node)
        return stats
