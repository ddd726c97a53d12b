"""K-means cluster index for candidate routing (IVF-style).

Parity: reference pkg/gpu/kmeans.go ClusterIndex (:144) — k-means++ init
(:364), GPU assignment (:491), incremental updates (:910-1009),
optimal k = sqrt(n/2) (:323). On MI355X the inner ops are the
hand-written HIP suite in csrc/kmeans.hip (fused distance+argmin assign,
atomic accumulate + finalize-with-drift, k-means++ min-distance update,
single-point incremental reassign — the CDNA4 replacement for the
reference's 9-kernel Metal suite, kmeans_kernels_darwin.metal:71-370);
the torch expressions below remain the CPU path and numerics oracle.
"""

from __future__ import annotations

import math
import threading
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from ..ops import native_or_none


def _native_for(x: torch.Tensor):
    """HIP kernel suite usable? (CUDA + d % 512 == 0, d <= 4096)."""
    nat = native_or_none()
    if nat is not None and x.is_cuda and x.shape[-1] % 512 == 0             and x.shape[-1] <= 4096:
        return nat
    return None


def optimal_k(n: int) -> int:
    return max(1, int(math.sqrt(n / 2)))


def kmeans(x: torch.Tensor, k: int, iters: int = 25, tol: float = 1e-4,
           seed: int = 0, verbose: bool = False):
    """Spherical-ish k-means on rows of x (any device). Returns (centroids,
    assignments). Distance = squared euclidean via the |x|^2+|c|^2-2xc trick;
    the GEMM is the hot op and runs on MFMA via hipBLASLt."""
    n, d = x.shape
    k = min(k, n)
    g = torch.Generator(device="cpu").manual_seed(seed)
    xf = x.float()

    # k-means++ init (on a sample for large n)
    sample = xf if n <= 100_000 else xf[torch.randperm(n, generator=g)[:100_000].to(x.device)]
    c = _kmeanspp(sample, k, g)

    nat = _native_for(x)
    if nat is not None:
        xb = x if x.dtype == torch.bfloat16 else xf.to(torch.bfloat16)
        prev_inertia = None
        assign = None
        for it in range(iters):
            cb = c.to(torch.bfloat16)
            a32, d2 = nat.kmeans_assign(xb, cb, (c * c).sum(-1))
            inertia = float(d2.sum())
            sums, counts = nat.kmeans_accum(xb, a32, k)
            c, _drift2 = nat.kmeans_finalize(sums, counts, c)
            assign = a32.long()
            if prev_inertia is not None and                     abs(prev_inertia - inertia) <= tol * max(prev_inertia, 1e-12):
                break
            prev_inertia = inertia
        return c, assign

    x_sq = (xf * xf).sum(-1, keepdim=True)  # [n,1]
    prev_inertia = None
    assign = None
    for it in range(iters):
        c_sq = (c * c).sum(-1)  # [k]
        # process in chunks to bound the [n,k] matrix
        chunk = max(1, min(n, (1 << 24) // max(k, 1)))
        assigns = []
        inertia = 0.0
        for s in range(0, n, chunk):
            e = min(s + chunk, n)
            d2 = x_sq[s:e] + c_sq[None, :] - 2.0 * (xf[s:e] @ c.T)
            m, a = d2.min(dim=1)
            inertia += float(m.clamp_min(0).sum())
            assigns.append(a)
        assign = torch.cat(assigns)
        # update: scatter-add
        new_c = torch.zeros_like(c)
        new_c.index_add_(0, assign, xf)
        counts = torch.bincount(assign, minlength=k).float().clamp_min(1)
        new_c /= counts[:, None]
        # keep empty clusters where they were
        empty = torch.bincount(assign, minlength=k) == 0
        new_c[empty] = c[empty]
        c = new_c
        if prev_inertia is not None and abs(prev_inertia - inertia) <= tol * max(prev_inertia, 1e-12):
            break
        prev_inertia = inertia
    return c, assign


def _kmeanspp(x: torch.Tensor, k: int, g) -> torch.Tensor:
    """k-means++ seeding; distances via |x|^2+|c|^2-2xc (GEMV on MFMA),
    sampling stays on-device — no [n,d] broadcast temporaries, no per-step
    host sync (the reference's init is the same algorithm on CPU,
    kmeans.go:364)."""
    n = x.shape[0]
    nat = _native_for(x)
    if nat is not None:
        xb = x.to(torch.bfloat16)
        first = int(torch.randint(n, (1,), generator=g))
        idxs = [torch.as_tensor([first], device=x.device)]
        d2 = torch.full((n,), 3.4e38, device=x.device)
        cvec = xb[first].contiguous()
        nat.kmeanspp_update(xb, cvec, float((x[first] ** 2).sum()), d2)
        for _ in range(1, k):
            w = d2.clamp_min(1e-12)
            cdf = torch.cumsum(w, 0)
            r = torch.rand(1, device=x.device) * cdf[-1]
            idx = torch.searchsorted(cdf, r).clamp_max_(n - 1)
            idxs.append(idx)
            cvec = xb[idx[0]].contiguous()
            nat.kmeanspp_update(xb, cvec, float((x[idx[0]] ** 2).sum()), d2)
        return x[torch.cat(idxs)].clone().float()

    x_sq = (x * x).sum(-1)
    first = int(torch.randint(n, (1,), generator=g))
    idxs = [torch.as_tensor([first], device=x.device)]
    c = x[first]
    d2 = (x_sq + (c * c).sum() - 2.0 * (x @ c)).clamp_min_(0)
    for _ in range(1, k):
        # inverse-CDF sampling on-device (multinomial internally sorts; a
        # cumsum+searchsorted is ~100x cheaper at n=100K)
        w = d2.clamp_min(1e-12)
        cdf = torch.cumsum(w, 0)
        r = torch.rand(1, device=x.device) * cdf[-1]
        idx = torch.searchsorted(cdf, r).clamp_max_(n - 1)
        idxs.append(idx)
        c = x[idx[0]]
        nd = (x_sq + (c * c).sum() - 2.0 * (x @ c)).clamp_min_(0)
        d2 = torch.minimum(d2, nd)
    return x[torch.cat(idxs)].clone()


class ClusterIndex:
    """IVF routing over an EmbeddingIndex: cluster once, route queries to
    the nprobe nearest centroids, exact-score members on GPU."""

    def __init__(self, nprobe: int = 8, reassign_drift: float = 0.2):
        self._lock = threading.RLock()
        self.centroids: Optional[torch.Tensor] = None
        self.members: List[List[str]] = []
        self._id2cluster: Dict[str, int] = {}
        self.counts: Optional[torch.Tensor] = None
        self.nprobe = nprobe
        self._drift = 0
        self._drift_limit_ratio = reassign_drift

    @property
    def k(self) -> int:
        return 0 if self.centroids is None else self.centroids.shape[0]

    def cluster(self, ids: Sequence[str], mat: torch.Tensor, k: int = None,
                iters: int = 25, seed: int = 0):
        with self._lock:
            n = len(ids)
            if n == 0:
                return
            k = k or optimal_k(n)
            c, assign = kmeans(mat, k, iters=iters, seed=seed)
            self.centroids = c
            self.members = [[] for _ in range(c.shape[0])]
            a = assign.tolist()
            self._id2cluster = {}
            for i, id_ in enumerate(ids):
                self.members[a[i]].append(id_)
                self._id2cluster[id_] = a[i]
            self.counts = torch.bincount(assign, minlength=c.shape[0]) \
                .to(dtype=torch.int32, device=c.device)
            self._drift = 0

    def add(self, id_: str, vec) -> None:
        """Incremental assignment to nearest centroid, updating the
        centroid itself (reference kmeans.go:910-1009 incremental path;
        HIP single-point kernel on MI355X)."""
        with self._lock:
            if self.centroids is None:
                return
            v = torch.as_tensor(np.asarray(vec, dtype=np.float32),
                                device=self.centroids.device)
            nat = _native_for(self.centroids)
            if nat is not None:
                vb = v.to(torch.bfloat16).contiguous()
                a32, _ = nat.kmeans_assign(
                    vb[None, :], self.centroids.to(torch.bfloat16),
                    (self.centroids * self.centroids).sum(-1))
                cl = int(a32[0])
            else:
                d2 = ((self.centroids - v) ** 2).sum(-1)
                cl = int(d2.argmin())
            old = self._id2cluster.get(id_)
            if old is not None and old != cl:
                try:
                    self.members[old].remove(id_)
                except ValueError:
                    pass
                self._point_update(old, v, -1)
            if old != cl:
                self.members[cl].append(id_)
                self._id2cluster[id_] = cl
                self._point_update(cl, v, +1)
            self._drift += 1

    def _point_update(self, cl: int, v: torch.Tensor, sign: int) -> None:
        """c = (c*cnt +/- v) / (cnt +/- 1) — incremental centroid move."""
        if getattr(self, "counts", None) is None:
            return
        nat = _native_for(self.centroids)
        if nat is not None:
            nat.kmeans_point_update(self.centroids, self.counts,
                                    v.to(torch.bfloat16).contiguous(), cl, sign)
            return
        cnt = int(self.counts[cl])
        new = cnt + sign
        if new <= 0:
            self.counts[cl] = 0
            return
        self.centroids[cl] = (self.centroids[cl] * cnt + sign * v) / new
        self.counts[cl] = new

    def remove(self, id_: str, vec=None) -> None:
        with self._lock:
            cl = self._id2cluster.pop(id_, None)
            if cl is not None:
                try:
                    self.members[cl].remove(id_)
                except ValueError:
                    pass
                if vec is not None:
                    v = torch.as_tensor(np.asarray(vec, dtype=np.float32),
                                        device=self.centroids.device)
                    self._point_update(cl, v, -1)

    def needs_recluster(self) -> bool:
        with self._lock:
            total = len(self._id2cluster)
            return total > 0 and self._drift > self._drift_limit_ratio * total

    def candidates(self, query, nprobe: int = None) -> List[str]:
        """Member ids of the nprobe nearest clusters."""
        with self._lock:
            if self.centroids is None:
                return []
            q = torch.as_tensor(np.asarray(query, dtype=np.float32),
                                device=self.centroids.device)
            d2 = ((self.centroids - q) ** 2).sum(-1)
            np_ = min(nprobe or self.nprobe, self.k)
            top = torch.topk(-d2, np_).indices.tolist()
            out: List[str] = []
            for t in top:
                out.extend(self.members[t])
            return out


def kmeans_assign_index(emb_index, k: int = None) -> dict:
    """Cluster an EmbeddingIndex; returns {id: cluster}. k defaults to
    ~sqrt(n/2) (the reference's heuristic)."""
    import math

    ids = [i for i in emb_index._ids
           if emb_index._id2slot.get(i) is not None
           and emb_index._id2slot[i] not in emb_index._dead]
    if len(ids) < 2:
        return {i: 0 for i in ids}
    slots = [emb_index._id2slot[i] for i in ids]
    x = emb_index._buf[slots].float()
    kk = k or max(1, int(math.sqrt(len(ids) / 2)))
    kk = min(kk, len(ids))
    centroids, assign = kmeans(x, kk)
    return {i: int(c) for i, c in zip(ids, assign.tolist())}
