"""Hybrid search service: per-node indexing, BM25 + vector + RRF + MMR.

Parity: reference pkg/search/search.go Service (:236): per-node indexing
(:651), BuildIndexes (:767), hybrid Search (:851), RRF fusion (:1432),
MMR diversification (:1544), type filters. Stays in sync with storage via
event callbacks (reference pkg/nornicdb/db.go:994-1035 wiring).
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np

from ..storage.types import Engine, EventType, Node
from .bm25 import FulltextIndex
from .embedding_index import EmbeddingIndex
from .fusion import mmr_diversify, rrf_fuse
from .hnsw import HNSWIndex
from .kmeans import ClusterIndex, optimal_k
from .pipeline import KMEANS_MIN, VectorSearchPipeline

TEXT_PROPS = ("title", "name", "content", "text", "description", "summary", "body")


def node_text(node: Node) -> str:
    parts = []
    for p in TEXT_PROPS:
        v = node.properties.get(p)
        if isinstance(v, str):
            parts.append(v)
    return " ".join(parts)


@dataclass
class SearchResult:
    id: str
    score: float
    node: Optional[Node] = None
    source: str = "hybrid"


class SearchService:
    def __init__(self, engine: Engine, dims: int = 1024,
                 device: Optional[str] = None, use_hnsw: bool = True,
                 embedder=None, quant: Optional[str] = None):
        self.engine = engine
        self.dims = dims
        self.embedder = embedder
        self._lock = threading.RLock()
        self.fulltext = FulltextIndex()
        # NORNICDB_SEARCH_QUANT=int8|fp8 stores the GPU corpus at
        # 1 B/element (half the HBM traffic, 2x capacity). int8 keeps
        # per-row scales (~7 bits, recall@10 >= 0.95 worst-case); fp8
        # e4m3 is scale-free but coarser (~0.91 worst-case)
        import os as _os
        quant = quant or _os.environ.get("NORNICDB_SEARCH_QUANT") or None
        self.emb = EmbeddingIndex(dims, device=device, quant=quant)
        self.hnsw = HNSWIndex(dims) if use_hnsw else None
        self.clusters = ClusterIndex()
        self.pipeline = VectorSearchPipeline(self.emb, self.hnsw, self.clusters)
        engine.register_callback(self._on_event)

    # ---- storage sync ----
    def _on_event(self, ev: str, obj: Any):
        if ev in (EventType.NODE_CREATED, EventType.NODE_UPDATED):
            self.index_node(obj)
        elif ev == EventType.NODE_DELETED:
            self.remove_node(obj.id)

    def index_node(self, node: Node):
        with self._lock:
            text = node_text(node)
            if text:
                self.fulltext.index(node.id, text)
            if node.embedding is not None and len(node.embedding) == self.dims:
                self.emb.add(node.id, node.embedding)
                if self.hnsw is not None and len(self.emb) <= KMEANS_MIN:
                    self.hnsw.add(node.id, node.embedding)
                self.clusters.add(node.id, node.embedding)

    def remove_node(self, node_id: str):
        with self._lock:
            self.fulltext.remove(node_id)
            self.emb.remove(node_id)
            if self.hnsw is not None:
                self.hnsw.remove(node_id)
            self.clusters.remove(node_id)

    def build_indexes(self):
        """Full scan (reference BuildIndexes)."""
        for node in self.engine.all_nodes():
            self.index_node(node)

    def recluster(self, k: int = None):
        ids = self.emb.ids()
        if not ids:
            return
        with self._lock:
            mat = self.emb.matrix().float()
            slot_ids = [i for i in self.emb._ids if i in self.emb._id2slot]
            self.clusters.cluster(slot_ids, mat[: len(self.emb._ids)],
                                  k=k or optimal_k(len(slot_ids)))

    # ---- queries ----
    def vector_search(self, query_vec, k: int = 10,
                      labels: Sequence[str] = None) -> List[SearchResult]:
        hits = self.pipeline.search(np.asarray(query_vec, np.float32), k * 3
                                    if labels else k)
        return self._materialize(hits, k, labels, source="vector")

    def text_search(self, query: str, k: int = 10,
                    labels: Sequence[str] = None) -> List[SearchResult]:
        hits = self.fulltext.search(query, k * 3 if labels else k)
        return self._materialize(hits, k, labels, source="fulltext")

    def search(self, query: str = None, query_vec=None, k: int = 10,
               labels: Sequence[str] = None, mmr: bool = False,
               mmr_lambda: float = 0.7) -> List[SearchResult]:
        """Hybrid search: BM25 + vector fused with RRF; optional MMR."""
        if query_vec is None and query and self.embedder is not None:
            query_vec = self.embedder.embed_query(query)
        rankings = []
        if query:
            rankings.append(self.fulltext.search(query, max(k * 4, 40)))
        if query_vec is not None and len(self.emb) > 0:
            rankings.append(self.pipeline.search(
                np.asarray(query_vec, np.float32), max(k * 4, 40)))
        if not rankings:
            return []
        if len(rankings) == 1:
            fused = rankings[0]
        else:
            fused = rrf_fuse(rankings)
        if mmr and query_vec is not None:
            vecs = {}
            for id_, _ in fused[: max(k * 4, 40)]:
                v = self.emb.get(id_)
                if v is not None:
                    vecs[id_] = v
            fused = mmr_diversify(fused[: max(k * 4, 40)], vecs, k * 3,
                                  lambda_=mmr_lambda)
        return self._materialize(fused, k, labels, source="hybrid")

    def _materialize(self, hits: List[Tuple[str, float]], k: int,
                     labels: Sequence[str], source: str) -> List[SearchResult]:
        out = []
        for id_, score in hits:
            try:
                node = self.engine.get_node(id_)
            except Exception:
                continue
            if labels and not any(lb in node.labels for lb in labels):
                continue
            out.append(SearchResult(id_, float(score), node, source))
            if len(out) >= k:
                break
        return out
