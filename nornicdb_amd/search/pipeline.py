"""Vector search pipeline: candidate generation x exact scoring.

Parity: reference pkg/search/vector_pipeline.go — brute force below 5K
vectors (:22-24), HNSW in the mid range, k-means cluster routing above
100K (kmeans_candidate_gen.go), candidate set max(20k, 200) capped at
5000 (:26-31), exact re-scoring on GPU (:236 GPUExactScorer).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

from .embedding_index import EmbeddingIndex
from .hnsw import HNSWIndex
from .kmeans import ClusterIndex

BRUTE_MAX = 5_000
KMEANS_MIN = 100_000


def candidate_count(k: int) -> int:
    return min(max(20 * k, 200), 5000)


class VectorSearchPipeline:
    def __init__(self, emb: EmbeddingIndex, hnsw: Optional[HNSWIndex] = None,
                 clusters: Optional[ClusterIndex] = None):
        self.emb = emb
        self.hnsw = hnsw
        self.clusters = clusters

    def search(self, query, k: int) -> List[Tuple[str, float]]:
        n = len(self.emb)
        if n == 0:
            return []
        # brute force: small corpora, or GPU path (fused kernel IS the fast path)
        if n <= BRUTE_MAX or (self.emb.device.type == "cuda"
                              and (self.clusters is None or self.clusters.k == 0)):
            return self.emb.search(query, k)
        if n > KMEANS_MIN and self.clusters is not None and self.clusters.k > 0:
            cands = self.clusters.candidates(query)
            if len(cands) > candidate_count(k):
                # HNSW narrows within the routed set if available, else truncate
                cands = cands[: candidate_count(k) * 4]
            scored = self.emb.score_subset(query, cands)
            return scored[:k]
        if self.hnsw is not None and len(self.hnsw) > 0:
            cands = [i for i, _ in self.hnsw.search(query, candidate_count(k))]
            scored = self.emb.score_subset(query, cands)
            return scored[:k]
        return self.emb.search(query, k)
