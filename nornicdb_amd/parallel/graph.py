"""Multi-GPU sharded graph analytics over RCCL/xGMI.

The CSR adjacency is ROW-SHARDED across ranks (rank owns rows
[base, base+n_local)); column indices stay global. Per iteration:
  - PageRank: every rank holds the full rank vector; computes new values
    for its rows from all-gathered contributions; all-gather over xGMI
    reassembles the vector (bandwidth-bound -> big contiguous segments).
  - BFS / WCC: all-reduce(MIN) of the global dist/component vector per
    level (xGMI ring all-reduce).
Works with the gloo backend on CPU for tests (world_size > 1 in-process /
multi-process), with the HIP kernels on MI355X under nccl(=RCCL).

This is the data-plane replacement for the reference's TCP replication
transport in the scale-up direction (the reference has no
intra-node GPU sharding; BASELINE config #4 requires it).
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch
import torch.distributed as dist

from ..ops import native_or_none
from ..graph.csr import CSRGraph


def shard_rows(n: int, rank: int, world: int) -> Tuple[int, int]:
    per = (n + world - 1) // world
    lo = min(rank * per, n)
    hi = min(lo + per, n)
    return lo, hi


def _all_gather_rank_vector(local: torch.Tensor, n: int, world: int):
    """Gather per-rank row slices into the full [n] vector."""
    per = (n + world - 1) // world
    padded = torch.zeros(per, dtype=local.dtype, device=local.device)
    padded[: local.numel()] = local
    out = [torch.empty_like(padded) for _ in range(world)]
    dist.all_gather(out, padded)
    return torch.cat(out)[:n]


def pagerank_sharded(row_ptr_local: torch.Tensor, col_idx_local: torch.Tensor,
                     outdeg_global: torch.Tensor, n: int, row_base: int,
                     damping: float = 0.85, iters: int = 20,
                     tol: float = 1e-6) -> torch.Tensor:
    """Distributed PageRank. Inputs are the LOCAL in-edge CSR rows; returns
    the full rank vector (identical on every rank).

    On GPU the per-iteration gather uses the HIP wave-per-row kernel; on
    CPU (gloo tests) a torch index_add implements the same contraction.
    """
    world = dist.get_world_size() if dist.is_initialized() else 1
    device = row_ptr_local.device
    nat = native_or_none() if device.type == "cuda" else None
    n_local = row_ptr_local.numel() - 1

    rank_vec = torch.full((n,), 1.0 / n, device=device)
    dangling_mask = outdeg_global == 0

    if nat is None:
        # CPU contraction precompute: local edge -> destination row map
        counts = (row_ptr_local[1:] - row_ptr_local[:-1])
        rows_local = torch.repeat_interleave(
            torch.arange(n_local, device=device), counts)

    for _ in range(iters):
        contrib = torch.where(outdeg_global > 0,
                              rank_vec / outdeg_global.clamp_min(1),
                              torch.zeros((), device=device))
        dangling = float(rank_vec[dangling_mask].sum()) / n
        base = (1 - damping) / n + damping * dangling
        if nat is not None:
            new_local = nat.pagerank_gather(row_ptr_local, col_idx_local,
                                            contrib.float(), damping, base)
        else:
            new_local = torch.full((n_local,), base, device=device)
            new_local.index_add_(0, rows_local,
                                 damping * contrib[col_idx_local.long()])
        if world > 1:
            new = _all_gather_rank_vector(new_local, n, world)
        else:
            new = new_local
        delta = float((new - rank_vec).abs().sum())
        rank_vec = new
        if delta < tol:
            break
    return rank_vec


def bfs_sharded(row_ptr_local: torch.Tensor, col_idx_local: torch.Tensor,
                n: int, row_base: int, source: int) -> torch.Tensor:
    """Distributed BFS levels; returns full dist vector (int32, -1 = unreached)."""
    world = dist.get_world_size() if dist.is_initialized() else 1
    device = row_ptr_local.device
    nat = native_or_none() if device.type == "cuda" else None
    n_local = row_ptr_local.numel() - 1

    dist_vec = torch.full((n,), -1, dtype=torch.int32, device=device)
    dist_vec[source] = 0
    INF = 2 ** 30

    level = 0
    while True:
        changed = torch.zeros(1, dtype=torch.int32, device=device)
        if nat is not None:
            nat.bfs_level(row_ptr_local, col_idx_local, dist_vec, changed,
                          row_base, level)
        else:
            active = (dist_vec[row_base:row_base + n_local] == level).nonzero().flatten()
            for u in active.tolist():
                s, e = int(row_ptr_local[u]), int(row_ptr_local[u + 1])
                for j in range(s, e):
                    v = int(col_idx_local[j])
                    if dist_vec[v] < 0:
                        dist_vec[v] = level + 1
                        changed[0] = 1
        if world > 1:
            # merge: unvisited(-1) -> INF, MIN-reduce, back to -1
            tmp = torch.where(dist_vec < 0, torch.full_like(dist_vec, INF), dist_vec)
            dist.all_reduce(tmp, op=dist.ReduceOp.MIN)
            dist_vec = torch.where(tmp >= INF, torch.full_like(tmp, -1), tmp)
            dist.all_reduce(changed, op=dist.ReduceOp.MAX)
        if int(changed.item()) == 0:
            break
        level += 1
    return dist_vec
