"""GPU-resident embedding matrix with top-k scoring.

Replaces reference pkg/gpu/gpu.go EmbeddingIndex (:1224: dirty-tracking
auto-sync, ScoreSubset :1554) with a single CDNA4 path: vectors live in a
bf16 torch tensor on the MI355X (fp32 numpy on CPU test runs), scored by
the fused HIP kNN kernels (ops.knn_search). No multi-backend probing.
"""

from __future__ import annotations

import threading
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch


class EmbeddingIndex:
    def __init__(self, dims: int, device: Optional[str] = None,
                 capacity: int = 1024, quant: Optional[str] = None):
        self.dims = dims
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        # quant="fp8": store vectors as OCP e4m3fn — half the HBM traffic
        # and 2x capacity (200M+ 1024-d per 288 GB GPU), scored by the
        # fp8 MFMA kernel (csrc/knn_fp8.hip); ~1% recall@10 cost on
        # normalized vectors. No reference analogue (pkg/gpu is fp32).
        self.quant = quant
        if quant == "fp8":
            self.dtype = torch.float8_e4m3fn
        elif quant == "int8":
            # symmetric per-row int8 (recommended): ~7 effective bits,
            # scored by the i8 MFMA at 2x the bf16 rate
            self.dtype = torch.int8
        elif quant is not None:
            raise ValueError(f"unknown quant mode {quant!r}")
        else:
            self.dtype = (torch.bfloat16 if self.device.type == "cuda"
                          else torch.float32)
        self._lock = threading.RLock()
        self._buf = torch.zeros(capacity, dims, device=self.device, dtype=self.dtype)
        self._scales = (torch.zeros(capacity, device=self.device)
                        if quant == "int8" else None)
        self._n = 0
        self._ids: List[str] = []
        self._id2slot: Dict[str, int] = {}
        self._dead: set = set()

    def __len__(self):
        return self._n - len(self._dead)

    def _grow(self, need: int):
        cap = self._buf.shape[0]
        if need <= cap:
            return
        new_cap = max(need, cap * 2)
        nb = torch.zeros(new_cap, self.dims, device=self.device, dtype=self.dtype)
        nb[:self._n] = self._buf[:self._n]
        self._buf = nb
        if self._scales is not None:
            ns = torch.zeros(new_cap, device=self.device)
            ns[:self._n] = self._scales[:self._n]
            self._scales = ns

    @staticmethod
    def _normalize(t: torch.Tensor) -> torch.Tensor:
        t = t.float()
        return t / torch.linalg.vector_norm(t, dim=-1, keepdim=True).clamp_min(1e-12)

    def _store_rows(self, m: torch.Tensor):
        """normalized fp32 rows -> (rows in storage dtype, scales|None)"""
        if self.quant == "int8":
            from ..ops.knn import quantize_int8
            return quantize_int8(m)
        return m.to(self.dtype), None

    def add(self, id_: str, vec) -> None:
        with self._lock:
            v = torch.as_tensor(np.asarray(vec, dtype=np.float32),
                                device=self.device)
            v, sc = self._store_rows(self._normalize(v.reshape(1, -1)))
            slot = self._id2slot.get(id_)
            if slot is not None:
                self._buf[slot] = v[0]
                if sc is not None:
                    self._scales[slot] = sc[0]
                self._dead.discard(slot)
                return
            self._grow(self._n + 1)
            self._buf[self._n] = v[0]
            if sc is not None:
                self._scales[self._n] = sc[0]
            self._ids.append(id_)
            self._id2slot[id_] = self._n
            self._n += 1

    def add_batch(self, ids: Sequence[str], mat) -> None:
        with self._lock:
            m = torch.as_tensor(np.asarray(mat, dtype=np.float32), device=self.device)
            m, sc = self._store_rows(self._normalize(m))
            self._grow(self._n + len(ids))
            for i, id_ in enumerate(ids):
                slot = self._id2slot.get(id_)
                if slot is not None:
                    self._buf[slot] = m[i]
                    if sc is not None:
                        self._scales[slot] = sc[i]
                    self._dead.discard(slot)
                else:
                    self._buf[self._n] = m[i]
                    if sc is not None:
                        self._scales[self._n] = sc[i]
                    self._ids.append(id_)
                    self._id2slot[id_] = self._n
                    self._n += 1

    def remove(self, id_: str) -> bool:
        with self._lock:
            slot = self._id2slot.pop(id_, None)
            if slot is None:
                return False
            self._dead.add(slot)
            return True

    def search(self, query, k: int) -> List[Tuple[str, float]]:
        """Brute-force top-k over all vectors (fused HIP kernel on GPU)."""
        from ..ops import knn_search

        with self._lock:
            if self._n == 0:
                return []
            q = torch.as_tensor(np.asarray(query, dtype=np.float32),
                                device=self.device).reshape(1, -1)
            q = self._normalize(q)
            kk = min(k + len(self._dead), self._n)
            if self.quant == "int8":
                from ..ops.knn import knn_search_int8
                s, i = knn_search_int8(self._buf[:self._n],
                                       self._scales[:self._n], q, kk)
            else:
                s, i = knn_search(self._buf[:self._n], q.to(self.dtype), kk)
            out = []
            for score, slot in zip(s[0].tolist(), i[0].tolist()):
                if slot < 0 or slot in self._dead:
                    continue
                out.append((self._ids[slot], float(score)))
                if len(out) >= k:
                    break
            return out

    def search_batch(self, queries, k: int) -> List[List[Tuple[str, float]]]:
        from ..ops import knn_search

        with self._lock:
            if self._n == 0:
                return [[] for _ in range(len(queries))]
            q = torch.as_tensor(np.asarray(queries, dtype=np.float32),
                                device=self.device)
            q = self._normalize(q)
            kk = min(k + len(self._dead), self._n)
            if self.quant == "int8":
                from ..ops.knn import knn_search_int8
                s, i = knn_search_int8(self._buf[:self._n],
                                       self._scales[:self._n], q, kk)
            else:
                s, i = knn_search(self._buf[:self._n], q.to(self.dtype), kk)
            outs = []
            for r in range(q.shape[0]):
                out = []
                for score, slot in zip(s[r].tolist(), i[r].tolist()):
                    if slot < 0 or slot in self._dead:
                        continue
                    out.append((self._ids[slot], float(score)))
                    if len(out) >= k:
                        break
                outs.append(out)
            return outs

    def score_subset(self, query, ids: Sequence[str]) -> List[Tuple[str, float]]:
        """Exact re-scoring of a candidate subset (reference ScoreSubset)."""
        with self._lock:
            slots = [self._id2slot[i] for i in ids if i in self._id2slot]
            if not slots:
                return []
            q = torch.as_tensor(np.asarray(query, dtype=np.float32),
                                device=self.device).reshape(-1)
            q = self._normalize(q.reshape(1, -1))[0]
            sl = torch.as_tensor(slots, device=self.device)
            sub = self._buf[sl].float()
            if self._scales is not None:
                sub = sub * self._scales[sl, None]
            scores = (sub @ q.float()).tolist()
            kept = [i for i in ids if i in self._id2slot]
            return sorted(zip(kept, scores), key=lambda kv: -kv[1])

    def get(self, id_: str) -> Optional[np.ndarray]:
        with self._lock:
            slot = self._id2slot.get(id_)
            if slot is None:
                return None
            v = self._buf[slot].float()
            if self._scales is not None:
                v = v * self._scales[slot]
            return v.cpu().numpy()

    def ids(self) -> List[str]:
        with self._lock:
            return [i for i in self._ids if i in self._id2slot]

    def matrix(self) -> torch.Tensor:
        with self._lock:
            m = self._buf[:self._n]
            if self._scales is not None:
                # int8 storage: return dequantized values — callers
                # (recluster/k-means) expect real vector magnitudes
                return m.float() * self._scales[:self._n, None]
            return m
