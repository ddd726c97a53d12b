"""CSR adjacency representation of the property graph.

Built from a storage Engine (or raw edge lists) for the analytics layer
(reference apoc/algo builds ad-hoc adjacency maps from the Engine,
apoc/algo/algo.go + pkg/linkpredict/graph_builder.go; here it is a proper
CSR so the HIP kernels and RCCL sharding can consume it).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch


@dataclass
class CSRGraph:
    node_ids: List[str]              # index -> external id
    row_ptr: np.ndarray              # [n+1] int64 (out-edges)
    col_idx: np.ndarray              # [m] int32
    in_row_ptr: Optional[np.ndarray] = None   # [n+1] (in-edges; for pagerank pull)
    in_col_idx: Optional[np.ndarray] = None
    weights: Optional[np.ndarray] = None      # [m] float32 aligned with col_idx

    @property
    def n(self) -> int:
        return len(self.node_ids)

    @property
    def m(self) -> int:
        return int(self.col_idx.shape[0])

    @property
    def id2idx(self) -> Dict[str, int]:
        if not hasattr(self, "_id2idx"):
            self._id2idx = {nid: i for i, nid in enumerate(self.node_ids)}
        return self._id2idx

    def out_degrees(self) -> np.ndarray:
        return np.diff(self.row_ptr).astype(np.int32)

    def with_in_edges(self) -> "CSRGraph":
        if self.in_row_ptr is not None:
            return self
        n, m = self.n, self.m
        rows = np.repeat(np.arange(n, dtype=np.int32), np.diff(self.row_ptr))
        order = np.argsort(self.col_idx, kind="stable")
        self.in_col_idx = rows[order]
        counts = np.bincount(self.col_idx, minlength=n)
        self.in_row_ptr = np.zeros(n + 1, dtype=np.int64)
        np.cumsum(counts, out=self.in_row_ptr[1:])
        return self

    def torch_csr(self, device="cpu", in_edges=False):
        if in_edges:
            self.with_in_edges()
            rp, ci = self.in_row_ptr, self.in_col_idx
        else:
            rp, ci = self.row_ptr, self.col_idx
        return (torch.as_tensor(rp, dtype=torch.int64, device=device),
                torch.as_tensor(ci, dtype=torch.int32, device=device))


def from_edges(n: int, edges: Sequence[Tuple[int, int]],
               node_ids: List[str] = None, undirected: bool = False,
               weights: Sequence[float] = None) -> CSRGraph:
    src = np.fromiter((e[0] for e in edges), dtype=np.int64, count=len(edges))
    dst = np.fromiter((e[1] for e in edges), dtype=np.int64, count=len(edges))
    w = np.asarray(weights, dtype=np.float32) if weights is not None else None
    if undirected:
        src, dst = np.concatenate([src, dst]), np.concatenate([dst, src])
        if w is not None:
            w = np.concatenate([w, w])
    order = np.argsort(src, kind="stable")
    src, dst = src[order], dst[order]
    if w is not None:
        w = w[order]
    counts = np.bincount(src, minlength=n)
    row_ptr = np.zeros(n + 1, dtype=np.int64)
    np.cumsum(counts, out=row_ptr[1:])
    return CSRGraph(node_ids or [str(i) for i in range(n)],
                    row_ptr, dst.astype(np.int32), weights=w)


def from_engine(engine, edge_types: Sequence[str] = None,
                undirected: bool = False, weight_prop: str = None) -> CSRGraph:
    node_ids = [n.id for n in engine.all_nodes()]
    idx = {nid: i for i, nid in enumerate(node_ids)}
    edges, weights = [], []
    for e in engine.all_edges():
        if edge_types and e.type not in edge_types:
            continue
        s, t = idx.get(e.start_node), idx.get(e.end_node)
        if s is None or t is None:
            continue
        edges.append((s, t))
        if weight_prop:
            weights.append(float(e.properties.get(weight_prop, 1.0)))
    g = from_edges(len(node_ids), edges, node_ids, undirected=undirected,
                   weights=weights if weight_prop else None)
    return g


def random_graph(n: int, avg_degree: int, seed: int = 0,
                 device: str = "cpu") -> CSRGraph:
    """Synthetic power-law-ish graph for benchmarks (no datasets offline)."""
    rng = np.random.default_rng(seed)
    m = n * avg_degree
    src = rng.integers(0, n, m)
    # preferential-ish skew on destinations
    dst = (rng.zipf(1.3, m) % n).astype(np.int64)
    order = np.argsort(src, kind="stable")
    src, dst = src[order], dst[order]
    counts = np.bincount(src, minlength=n)
    row_ptr = np.zeros(n + 1, dtype=np.int64)
    np.cumsum(counts, out=row_ptr[1:])
    return CSRGraph([str(i) for i in range(n)], row_ptr, dst.astype(np.int32))
