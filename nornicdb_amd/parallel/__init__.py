"""Multi-GPU (RCCL over xGMI) sharding: vector index + CSR graph."""

from .graph import bfs_sharded, pagerank_sharded, shard_rows

__all__ = ["pagerank_sharded", "bfs_sharded", "shard_rows"]
