"""NornicDB-AMD: an MI355X-native graph + vector database framework.

A from-scratch rebuild of the capabilities of orneryd/NornicDB (a Neo4j
compatible graph database with native vector search, reference mounted at
/root/reference) designed MI355X-first:

- hand-written HIP/CDNA4 kernels (MFMA + LDS tiling, wave64) for the vector
  search, embedding-model forward, k-means and graph-algorithm hot paths
  (replaces reference pkg/gpu/{cuda,metal,vulkan,opencl} + llama.cpp);
- PyTorch-ROCm as the tensor runtime; RCCL over xGMI (torch.distributed,
  one process per GPU) for multi-GPU sharding of the vector index and CSR
  adjacency (replaces reference pkg/replication's data plane for scaling);
- Python/C++ host layer for storage (WAL + snapshots), Cypher, Bolt and
  HTTP protocol surface (replaces reference pkg/{storage,cypher,bolt,server}).
"""

__version__ = "0.1.0"

from . import ops  # noqa: F401
