"""Automatic relationship inference ("cognitive" auto-linking).

Parity: reference pkg/inference/inference.go — Engine.OnStore derives
suggested edges from (a) embedding similarity, (b) co-access, (c) temporal
proximity, (d) transitive closure; with per-pair cooldown (cooldown.go),
evidence accumulation (evidence.go) and edge decay (edge_decay.go).
"""

from __future__ import annotations

import time
from collections import defaultdict
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from ..storage.types import Edge, Engine, Node, NotFoundError, new_id
from .evidence import (CooldownTable, EvidenceBuffer, EvidenceThreshold)
from .linkpredict import adamic_adar
from .temporal import AccessTracker

# suggestion reason -> evidence label (reference signal taxonomy)
_REASON_LABEL = {"similarity": "similar_to", "co-access": "coaccess",
                 "temporal": "relates_to", "transitive": "topology"}


@dataclass
class InferenceConfig:
    similarity_threshold: float = 0.82
    max_suggestions_per_store: int = 5
    min_confidence: float = 0.5
    cooldown_s: float = 300.0
    temporal_window_s: float = 600.0
    evidence_required: int = 1
    edge_type: str = "RELATES_TO"
    edge_decay_per_day: float = 0.02
    prune_below: float = 0.1


class InferenceEngine:
    def __init__(self, engine: Engine, search_service=None,
                 tracker: AccessTracker = None,
                 config: InferenceConfig = None, now_fn=time.time):
        self.engine = engine
        self.search = search_service
        self.tracker = tracker or AccessTracker(now_fn=now_fn)
        self.cfg = config or InferenceConfig()
        self.now = now_fn
        # evidence buffering + per-label cooldown (reference evidence.go /
        # cooldown.go). evidence_required <= 1 collapses every threshold
        # to a single signal (the facade's immediate-link mode).
        if self.cfg.evidence_required <= 1:
            thr = EvidenceThreshold(1, self.cfg.min_confidence, 0, 86400.0)
            self.evidence = EvidenceBuffer(
                {lb: thr for lb in _REASON_LABEL.values()}, now_fn=now_fn)
        else:
            base = EvidenceThreshold(self.cfg.evidence_required,
                                     self.cfg.min_confidence, 1, 86400.0)
            self.evidence = EvidenceBuffer(
                {lb: base for lb in _REASON_LABEL.values()}, now_fn=now_fn)
        self.cooldowns = CooldownTable(
            {lb: self.cfg.cooldown_s for lb in _REASON_LABEL.values()},
            now_fn=now_fn)
        from ..storage.node_config import NodeConfigStore
        self.node_configs = NodeConfigStore(engine)

    # ---- main hook ----
    def on_store(self, node: Node, session_id: str = "") -> List[Edge]:
        """Run inference for a freshly stored/embedded node; returns created edges."""
        suggestions = self.suggest(node)
        created = []
        for target, confidence, reason in suggestions:
            e = self._maybe_link(node.id, target, confidence, reason,
                                 session_id)
            if e is not None:
                created.append(e)
        return created

    def suggest(self, node: Node) -> List[Tuple[str, float, str]]:
        out: List[Tuple[str, float, str]] = []
        seen = set()
        # (a) embedding similarity
        if self.search is not None and node.embedding is not None:
            for r in self.search.vector_search(node.embedding,
                                               self.cfg.max_suggestions_per_store + 1):
                if r.id == node.id or r.id in seen:
                    continue
                if r.score >= self.cfg.similarity_threshold:
                    seen.add(r.id)
                    out.append((r.id, float(r.score), "similarity"))
        # (b) co-access
        for a, b, count in self.tracker.co_accessed(self.cfg.temporal_window_s)[:10]:
            other = b if a == node.id else a if b == node.id else None
            if other and other not in seen:
                seen.add(other)
                conf = min(0.5 + 0.1 * count, 0.9)
                out.append((other, conf, "co-access"))
        # (c) temporal proximity: nodes created within the window
        t0 = node.properties.get("created_at", self.now())
        for other in self.engine.get_nodes_by_label("Memory"):
            if other.id == node.id or other.id in seen:
                continue
            dt = abs(other.properties.get("created_at", 0) - t0)
            if 0 < dt <= self.cfg.temporal_window_s:
                seen.add(other.id)
                conf = 0.5 + 0.3 * (1 - dt / self.cfg.temporal_window_s)
                out.append((other.id, conf, "temporal"))
        # (d) transitive topology (2-hop closure score)
        for cand, score in self._transitive(node.id):
            if cand not in seen:
                seen.add(cand)
                out.append((cand, min(0.5 + score / 4.0, 0.95), "transitive"))
        out.sort(key=lambda t: -t[1])
        return out[: self.cfg.max_suggestions_per_store]

    def _transitive(self, nid: str) -> List[Tuple[str, float]]:
        direct = set(self.engine.neighbors(nid))
        scores = []
        two_hop = set()
        for nb in direct:
            two_hop |= set(self.engine.neighbors(nb))
        two_hop -= direct
        two_hop.discard(nid)
        for c in list(two_hop)[:20]:
            s = adamic_adar(self.engine, nid, c)
            if s > 0:
                scores.append((c, s))
        scores.sort(key=lambda kv: -kv[1])
        return scores[:5]

    def _maybe_link(self, a: str, b: str, confidence: float,
                    reason: str, session_id: str = "") -> Optional[Edge]:
        if confidence < self.cfg.min_confidence:
            return None
        # per-node overrides: deny/pin lists, trust-adjusted confidence
        # bar, edge caps (reference pkg/storage/node_config.go)
        if self.node_configs is not None:
            ok, _why = self.node_configs.is_edge_allowed(
                a, b, self.cfg.edge_type, confidence,
                self.cfg.min_confidence)
            if not ok:
                return None
        s, d = sorted((a, b))
        label = _REASON_LABEL.get(reason, "relates_to")
        # cooldown first: a recently-materialized pair accumulates no new
        # evidence churn (reference cooldown.go CanMaterialize)
        if not self.cooldowns.can_materialize(s, d, label):
            return None
        if not self.evidence.add_evidence(s, d, label, confidence, reason,
                                          session_id):
            return None
        # already linked?
        if b in self.engine.neighbors(a):
            self.cooldowns.record_materialization(s, d, label)
            return None
        self.cooldowns.record_materialization(s, d, label)
        now = self.now()
        e = Edge(id=new_id("inf"), type=self.cfg.edge_type, start_node=a,
                 end_node=b,
                 properties={"confidence": confidence, "inferred": True,
                             "reason": reason, "created_at": now})
        try:
            return self.engine.create_edge(e)
        except NotFoundError:
            return None

    # ---- edge decay (reference edge_decay.go) ----
    def decay_inferred_edges(self) -> Dict[str, int]:
        stats = {"decayed": 0, "pruned": 0}
        now = self.now()
        for e in list(self.engine.all_edges()):
            if not e.properties.get("inferred"):
                continue
            age_days = (now - e.properties.get("created_at", now)) / 86400.0
            conf = e.properties.get("confidence", 0.5) - \
                self.cfg.edge_decay_per_day * age_days
            if conf < self.cfg.prune_below:
                self.engine.delete_edge(e.id)
                stats["pruned"] += 1
            else:
                e.properties["confidence"] = conf
                self.engine.update_edge(e)
                stats["decayed"] += 1
        return stats


class HeimdallQC:
    """LLM quality-control of auto-inferred links (reference
    pkg/inference/heimdall_qc.go): before an auto-link is committed, the
    SLM is asked whether the connection is plausible; non-affirmative
    answers veto the edge. Degrades to accept-all when no manager is
    available (same as the reference without a loaded model)."""

    def __init__(self, manager=None, threshold: float = 0.5):
        self.manager = manager
        self.threshold = threshold
        self.stats = {"checked": 0, "vetoed": 0}

    def check(self, a_node, b_node, reason: str) -> bool:
        self.stats["checked"] += 1
        if self.manager is None:
            return True
        try:
            prompt = (f"Should '{a_node.properties.get('name', a_node.id)}' "
                      f"link to '{b_node.properties.get('name', b_node.id)}' "
                      f"because {reason}? Answer yes or no.")
            # greedy: deterministic verdicts, and temperature<=0 takes the
            # in-kernel multi-token decode path (models/heimdall.py)
            out = self.manager.generate(prompt, max_tokens=4, temperature=0.0)
            ok = "no" not in out.lower().split()
        except Exception:
            ok = True
        if not ok:
            self.stats["vetoed"] += 1
        return ok


class ClusterIntegration:
    """k-means cluster assignments as an inference signal (reference
    pkg/inference/cluster_integration.go): nodes sharing a cluster get a
    similarity-floor boost; re-clustering is triggered by the embed queue
    (db.EmbedQueue._pending_recluster)."""

    def __init__(self, search_service, boost: float = 0.1):
        self.search = search_service
        self.boost = boost
        self._assign = {}

    def recluster(self, k: int = None):
        from ..search.kmeans import kmeans_assign_index
        self._assign = kmeans_assign_index(self.search.emb, k=k)
        return len(set(self._assign.values()))

    def same_cluster(self, a: str, b: str) -> bool:
        ca, cb = self._assign.get(a), self._assign.get(b)
        return ca is not None and ca == cb

    def boost_for(self, a: str, b: str) -> float:
        return self.boost if self.same_cluster(a, b) else 0.0
