"""Per-node configuration: edge limits, pin/deny lists, trust levels.

Parity: reference pkg/storage/node_config.go — per-node overrides that
gate automatic edge materialization (max in/out edges, per-label caps,
pinned targets that always link, denied targets that never link, and
trust levels that shift the confidence bar for inferred links).
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict, Optional, Set, Tuple

TRUST_LOW = "low"
TRUST_DEFAULT = "default"
TRUST_HIGH = "high"
TRUST_VERIFIED = "verified"

# confidence-threshold adjustment per trust level (reference
# TrustLevel.ConfidenceAdjustment: low-trust nodes need MORE confidence
# before an automatic edge is allowed, verified ones less)
_TRUST_ADJ = {TRUST_LOW: +0.20, TRUST_DEFAULT: 0.0,
              TRUST_HIGH: -0.10, TRUST_VERIFIED: -0.20}


@dataclass
class LabelConfig:
    max_edges: int = 0          # 0 = unlimited
    disabled: bool = False


@dataclass
class NodeConfig:
    node_id: str
    max_out_edges: int = 0      # 0 = unlimited
    max_in_edges: int = 0
    trust_level: str = TRUST_DEFAULT
    pinned: Set[str] = field(default_factory=set)   # always allowed
    denied: Set[str] = field(default_factory=set)   # never allowed
    label_configs: Dict[str, LabelConfig] = field(default_factory=dict)
    auto_link_disabled: bool = False

    # ---- queries ----
    def is_pinned(self, target: str) -> bool:
        return target in self.pinned

    def is_denied(self, target: str) -> bool:
        return target in self.denied

    def confidence_adjustment(self) -> float:
        return _TRUST_ADJ.get(self.trust_level, 0.0)

    def add_pin(self, target: str):
        self.pinned.add(target)
        self.denied.discard(target)

    def add_deny(self, target: str):
        self.denied.add(target)
        self.pinned.discard(target)


class NodeConfigStore:
    """In-memory registry consulted by the inference engine before
    materializing an automatic edge (reference IsEdgeAllowedWithReason)."""

    def __init__(self, engine=None):
        self._lock = threading.Lock()
        self._configs: Dict[str, NodeConfig] = {}
        self.engine = engine  # optional: for live edge counts

    def get(self, node_id: str) -> Optional[NodeConfig]:
        with self._lock:
            return self._configs.get(node_id)

    def get_or_create(self, node_id: str) -> NodeConfig:
        with self._lock:
            c = self._configs.get(node_id)
            if c is None:
                c = NodeConfig(node_id)
                self._configs[node_id] = c
            return c

    def set(self, config: NodeConfig):
        with self._lock:
            self._configs[config.node_id] = config

    def remove(self, node_id: str):
        with self._lock:
            self._configs.pop(node_id, None)

    def __len__(self):
        with self._lock:
            return len(self._configs)

    # ---- the gate ----
    def is_edge_allowed(self, src: str, dst: str, label: str = "",
                        confidence: float = 1.0,
                        base_threshold: float = 0.0) -> Tuple[bool, str]:
        """(allowed, reason). Pin wins over everything except deny on the
        other side; deny wins over pin on the same side; limits and
        trust apply otherwise (reference node_config.go semantics)."""
        cs = self.get(src)
        cd = self.get(dst)
        for c, other in ((cs, dst), (cd, src)):
            if c and c.is_denied(other):
                return False, f"target {other} denied by {c.node_id}"
        pinned = (cs and cs.is_pinned(dst)) or (cd and cd.is_pinned(src))
        if pinned:
            return True, "pinned"
        if (cs and cs.auto_link_disabled) or (cd and cd.auto_link_disabled):
            return False, "auto-link disabled"
        # trust: shift the confidence bar
        adj = (cs.confidence_adjustment() if cs else 0.0) \
            + (cd.confidence_adjustment() if cd else 0.0)
        if confidence < base_threshold + adj:
            return False, (f"confidence {confidence:.2f} below trust-adjusted "
                           f"threshold {base_threshold + adj:.2f}")
        # per-label caps + edge-count limits (live counts via engine)
        for c, outgoing in ((cs, True), (cd, False)):
            if c is None:
                continue
            lc = c.label_configs.get(label)
            if lc and lc.disabled:
                return False, f"label {label!r} disabled on {c.node_id}"
            if self.engine is not None:
                try:
                    edges = (self.engine.get_out_edges(c.node_id) if outgoing
                             else self.engine.get_in_edges(c.node_id))
                except Exception:
                    edges = []
                cap = c.max_out_edges if outgoing else c.max_in_edges
                if cap and len(edges) >= cap:
                    return False, (f"{'out' if outgoing else 'in'}-edge cap "
                                   f"{cap} reached on {c.node_id}")
                if lc and lc.max_edges:
                    n = sum(1 for e in edges if e.type == label)
                    if n >= lc.max_edges:
                        return False, (f"label {label!r} at max capacity "
                                       f"({n}/{lc.max_edges}) on {c.node_id}")
        return True, "allowed"
