"""Multi-GPU sharded graph analytics over RCCL/xGMI.

The CSR adjacency is ROW-SHARDED across ranks (rank owns rows
[base, base+n_local)); column indices stay global. Per iteration:
  - PageRank: the rank vector stays SHARDED; each rank exchanges only
    the halo contrib values its local CSR references (HaloExchange,
    one all_to_all of per-owner segments built from a one-time halo
    index) plus two scalar all-reduces. xGMI is point-to-point, so the
    all_to_all maps onto direct links rather than a ring of the full
    vector.
  - BFS / WCC: all-reduce(MIN) of the global dist/component vector per
    level (xGMI ring all-reduce; frontier vectors are int32, one per
    level, not per edge).
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


def _all_to_all(recv: list, send: list, rank: int, world: int):
    """dist.all_to_all with a gloo fallback (gloo lacks alltoall): post
    all irecvs, then isends, wait — self-slot copied locally."""
    if dist.get_backend() == "nccl":
        dist.all_to_all(recv, send)
        return
    works = []
    for p in range(world):
        if p == rank:
            continue
        if recv[p].numel():
            works.append(dist.irecv(recv[p], src=p))
    for p in range(world):
        if p == rank:
            continue
        if send[p].numel():
            works.append(dist.isend(send[p].contiguous(), dst=p))
    recv[rank].copy_(send[rank])
    for w in works:
        w.wait()


def shard_rows(n: int, rank: int, world: int) -> Tuple[int, int]:
    per = (n + world - 1) // world
    lo = min(rank * per, n)
    hi = min(lo + per, n)
    return lo, hi


class HaloExchange:
    """Per-shard slice exchange for row-sharded CSR iteration.

    Built ONCE from the local column set: each rank needs contrib values
    only for the (unique, sorted) global columns its local CSR touches.
    Because row ownership is by contiguous ranges, the sorted unique
    column array splits into per-owner contiguous segments, so each
    iteration is ONE all_to_all of exactly those values — no rank ever
    ships the full vector (the previous all-gather
    moved the whole n-vector per iteration; at 100M+ nodes that is GBs
    over xGMI each round).
    """

    def __init__(self, col_idx_local: torch.Tensor, n: int, world: int,
                 rank: int, device):
        self.world = world
        self.rank = rank
        per = (n + world - 1) // world
        uniq = torch.unique(col_idx_local.long())  # sorted
        owner_bounds = torch.arange(1, world + 1, device=uniq.device) * per
        # segment p = needed cols owned by rank p
        seg_ends = torch.searchsorted(uniq, owner_bounds.to(uniq.dtype))
        seg_starts = torch.cat([torch.zeros(1, dtype=seg_ends.dtype,
                                            device=seg_ends.device),
                                seg_ends[:-1]])
        self.uniq = uniq
        self.recv_counts = (seg_ends - seg_starts).tolist()
        # tell every owner which of ITS local indices we need
        req = [
            (uniq[int(seg_starts[p]):int(seg_ends[p])] - p * per).to(device)
            for p in range(world)
        ]
        # exchange request sizes, then the request index lists
        send_sizes = torch.tensor(self.recv_counts, dtype=torch.long)
        all_sizes = [torch.zeros(world, dtype=torch.long) for _ in range(world)]
        dist.all_gather(all_sizes, send_sizes)
        self.send_counts = [int(all_sizes[p][rank]) for p in range(world)]
        send_idx = [torch.zeros(c, dtype=torch.long, device=device)
                    for c in self.send_counts]
        _all_to_all(send_idx, req, rank, world)
        self.send_idx = send_idx      # per peer: OUR local indices to send
        self.device = device

    def exchange(self, local_vals: torch.Tensor) -> torch.Tensor:
        """Returns the halo value buffer aligned with self.uniq."""
        send = [local_vals[idx] for idx in self.send_idx]
        recv = [torch.empty(c, dtype=local_vals.dtype, device=self.device)
                for c in self.recv_counts]
        _all_to_all(recv, send, self.rank, self.world)
        return torch.cat(recv) if recv else local_vals.new_empty(0)


def pagerank_sharded(row_ptr_local: torch.Tensor, col_idx_local: torch.Tensor,
                     outdeg_global: torch.Tensor, n: int, row_base: int,
                     damping: float = 0.85, iters: int = 20,
                     tol: float = 1e-6) -> torch.Tensor:
    """Distributed PageRank. Inputs are the LOCAL in-edge CSR rows; returns
    the full rank vector (identical on every rank; gathered once at END).

    The rank vector lives SHARDED: each iteration computes the local row
    slice with the HIP wave-per-row kernel (CPU: torch index_add) and
    exchanges only halo contrib values (HaloExchange) plus two scalar
    all-reduces (dangling mass, convergence delta).
    """
    world = dist.get_world_size() if dist.is_initialized() else 1
    rank = dist.get_rank() if dist.is_initialized() else 0
    device = row_ptr_local.device
    nat = native_or_none() if device.type == "cuda" else None
    n_local = row_ptr_local.numel() - 1
    per = (n + world - 1) // world

    outdeg_local = outdeg_global[row_base:row_base + n_local]
    rank_local = torch.full((n_local,), 1.0 / n, device=device)
    dangling_local = outdeg_local == 0

    halo = None
    col_halo = col_idx_local
    if world > 1:
        halo = HaloExchange(col_idx_local, n, world, rank, device)
        col_halo = torch.searchsorted(
            halo.uniq, col_idx_local.long()).to(col_idx_local.dtype)

    if nat is None:
        counts = (row_ptr_local[1:] - row_ptr_local[:-1])
        rows_local = torch.repeat_interleave(
            torch.arange(n_local, device=device), counts)

    for _ in range(iters):
        contrib_local = torch.where(
            outdeg_local > 0, rank_local / outdeg_local.clamp_min(1),
            torch.zeros((), device=device)).float()
        d = rank_local[dangling_local].sum().reshape(1)
        if world > 1:
            vals = halo.exchange(contrib_local)
            dist.all_reduce(d)
        else:
            vals = contrib_local
        dangling = float(d) / n
        base = (1 - damping) / n + damping * dangling
        if nat is not None:
            new_local = nat.pagerank_gather(row_ptr_local, col_halo,
                                            vals, damping, base)
        else:
            new_local = torch.full((n_local,), base, device=device)
            new_local.index_add_(0, rows_local,
                                 damping * vals[col_halo.long()])
        delta = (new_local - rank_local).abs().sum()
        if world > 1:
            dist.all_reduce(delta)
        rank_local = new_local
        if float(delta) < tol:
            break

    if world == 1:
        return rank_local
    # reassemble the full vector once at the end
    padded = torch.zeros(per, dtype=rank_local.dtype, device=device)
    padded[:n_local] = rank_local
    out = [torch.empty_like(padded) for _ in range(world)]
    dist.all_gather(out, padded)
    return torch.cat(out)[:n]


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
