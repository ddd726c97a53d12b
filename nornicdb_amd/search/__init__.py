"""Search & indexing: HNSW, BM25, k-means routing, GPU embedding index,
RRF/MMR fusion, hybrid search service."""

from .bm25 import FulltextIndex, tokenize
from .embedding_index import EmbeddingIndex
from .fusion import mmr_diversify, rrf_fuse
from .hnsw import HNSWIndex
from .kmeans import ClusterIndex, kmeans, optimal_k
from .pipeline import VectorSearchPipeline
from .service import SearchResult, SearchService, node_text
from .vectorspace import COSINE, DOT, EUCLIDEAN, Registry, VectorSpace

__all__ = ["FulltextIndex", "tokenize", "EmbeddingIndex", "HNSWIndex",
           "ClusterIndex", "kmeans", "optimal_k", "VectorSearchPipeline",
           "SearchService", "SearchResult", "node_text", "rrf_fuse",
           "mmr_diversify", "Registry", "VectorSpace", "COSINE", "DOT",
           "EUCLIDEAN"]
