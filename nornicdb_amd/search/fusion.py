"""Result fusion: Reciprocal Rank Fusion + MMR diversification.

Parity: reference pkg/search/search.go RRF (:1432, k=60 per :169) and MMR
(:1544, lambda 0.7).
"""

from __future__ import annotations

from typing import Dict, List, Sequence, Tuple

import numpy as np

RRF_K = 60


def rrf_fuse(rankings: Sequence[List[Tuple[str, float]]],
             weights: Sequence[float] = None,
             k: int = RRF_K) -> List[Tuple[str, float]]:
    """Fuse ranked (id, score) lists: score = sum w_i / (k + rank_i)."""
    weights = weights or [1.0] * len(rankings)
    scores: Dict[str, float] = {}
    for ranking, w in zip(rankings, weights):
        for rank, (id_, _) in enumerate(ranking):
            scores[id_] = scores.get(id_, 0.0) + w / (k + rank + 1)
    return sorted(scores.items(), key=lambda kv: -kv[1])


def mmr_diversify(candidates: List[Tuple[str, float]],
                  vectors: Dict[str, np.ndarray],
                  k: int, lambda_: float = 0.7) -> List[Tuple[str, float]]:
    """Maximal Marginal Relevance re-ranking over candidate (id, relevance)."""
    if not candidates:
        return []
    chosen: List[Tuple[str, float]] = []
    remaining = list(candidates)
    while remaining and len(chosen) < k:
        best_i, best_v = 0, -1e30
        for i, (cid, rel) in enumerate(remaining):
            vred = 0.0
            cv = vectors.get(cid)
            if cv is not None and chosen:
                sims = [float(np.dot(cv, vectors[sid]))
                        for sid, _ in chosen if sid in vectors]
                if sims:
                    vred = max(sims)
            mmr = lambda_ * rel - (1 - lambda_) * vred
            if mmr > best_v:
                best_v, best_i = mmr, i
        chosen.append(remaining.pop(best_i))
    return chosen
