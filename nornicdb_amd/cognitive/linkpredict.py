"""Topological link prediction.

Parity: reference pkg/linkpredict/topology.go — CommonNeighbors, Jaccard,
AdamicAdar, PreferentialAttachment, ResourceAllocation — plus hybrid
topology+semantic scoring (hybrid.go) using embedding cosine.
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np

from ..storage.types import Engine


def _neighbors(engine: Engine, nid: str) -> set:
    return set(engine.neighbors(nid))


def common_neighbors(engine: Engine, a: str, b: str) -> float:
    return float(len(_neighbors(engine, a) & _neighbors(engine, b)))


def jaccard(engine: Engine, a: str, b: str) -> float:
    na, nb = _neighbors(engine, a), _neighbors(engine, b)
    u = na | nb
    return len(na & nb) / len(u) if u else 0.0


def adamic_adar(engine: Engine, a: str, b: str) -> float:
    score = 0.0
    for z in _neighbors(engine, a) & _neighbors(engine, b):
        deg = len(_neighbors(engine, z))
        if deg > 1:
            score += 1.0 / math.log(deg)
    return score


def preferential_attachment(engine: Engine, a: str, b: str) -> float:
    return float(len(_neighbors(engine, a)) * len(_neighbors(engine, b)))


def resource_allocation(engine: Engine, a: str, b: str) -> float:
    score = 0.0
    for z in _neighbors(engine, a) & _neighbors(engine, b):
        deg = len(_neighbors(engine, z))
        if deg > 0:
            score += 1.0 / deg
    return score


PREDICTORS = {
    "common_neighbors": common_neighbors,
    "jaccard": jaccard,
    "adamic_adar": adamic_adar,
    "preferential_attachment": preferential_attachment,
    "resource_allocation": resource_allocation,
}


def predict_links(engine: Engine, node_id: str, method: str = "adamic_adar",
                  k: int = 10) -> List[Tuple[str, float]]:
    """Rank non-adjacent 2-hop candidates for node_id."""
    fn = PREDICTORS[method]
    direct = _neighbors(engine, node_id)
    candidates = set()
    for nb in direct:
        candidates |= _neighbors(engine, nb)
    candidates -= direct
    candidates.discard(node_id)
    scored = [(c, fn(engine, node_id, c)) for c in candidates]
    scored = [(c, s) for c, s in scored if s > 0]
    scored.sort(key=lambda kv: -kv[1])
    return scored[:k]


def hybrid_score(engine: Engine, a: str, b: str,
                 topo_weight: float = 0.5, method: str = "adamic_adar") -> float:
    """Blend normalized topology score with embedding cosine similarity."""
    topo = PREDICTORS[method](engine, a, b)
    topo_n = 1.0 - math.exp(-topo)  # squash to [0,1)
    sem = 0.0
    try:
        na, nb = engine.get_node(a), engine.get_node(b)
        if na.embedding is not None and nb.embedding is not None:
            va = np.asarray(na.embedding, np.float32)
            vb = np.asarray(nb.embedding, np.float32)
            denom = (np.linalg.norm(va) * np.linalg.norm(vb)) or 1.0
            sem = float(va @ vb / denom)
    except Exception:
        pass
    return topo_weight * topo_n + (1 - topo_weight) * max(sem, 0.0)
