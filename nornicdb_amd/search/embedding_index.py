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
                 capacity: int = 1024):
        self.dims = dims
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        self._lock = threading.RLock()
        self._buf = torch.zeros(capacity, dims, device=self.device, dtype=self.dtype)
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

    @staticmethod
    def _normalize(t: torch.Tensor) -> torch.Tensor:
        t = t.float()
        return t / torch.linalg.vector_norm(t, dim=-1, keepdim=True).clamp_min(1e-12)

    def add(self, id_: str, vec) -> None:
        with self._lock:
            v = torch.as_tensor(np.asarray(vec, dtype=np.float32),
                                device=self.device)
            v = self._normalize(v.reshape(1, -1)).to(self.dtype)
            slot = self._id2slot.get(id_)
            if slot is not None:
                self._buf[slot] = v[0]
                self._dead.discard(slot)
                return
            self._grow(self._n + 1)
            self._buf[self._n] = v[0]
            self._ids.append(id_)
            self._id2slot[id_] = self._n
            self._n += 1

    def add_batch(self, ids: Sequence[str], mat) -> None:
        with self._lock:
            m = torch.as_tensor(np.asarray(mat, dtype=np.float32), device=self.device)
            m = self._normalize(m).to(self.dtype)
            self._grow(self._n + len(ids))
            for i, id_ in enumerate(ids):
                slot = self._id2slot.get(id_)
                if slot is not None:
                    self._buf[slot] = m[i]
                    self._dead.discard(slot)
                else:
                    self._buf[self._n] = m[i]
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
            q = self._normalize(q).to(self.dtype)
            kk = min(k + len(self._dead), self._n)
            s, i = knn_search(self._buf[:self._n], q, kk)
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
            q = self._normalize(q).to(self.dtype)
            kk = min(k + len(self._dead), self._n)
            s, i = knn_search(self._buf[:self._n], q, kk)
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
            q = self._normalize(q.reshape(1, -1))[0].to(self.dtype)
            sub = self._buf[torch.as_tensor(slots, device=self.device)]
            scores = (sub.float() @ q.float()).tolist()
            kept = [i for i in ids if i in self._id2slot]
            return sorted(zip(kept, scores), key=lambda kv: -kv[1])

    def get(self, id_: str) -> Optional[np.ndarray]:
        with self._lock:
            slot = self._id2slot.get(id_)
            if slot is None:
                return None
            return self._buf[slot].float().cpu().numpy()

    def ids(self) -> List[str]:
        with self._lock:
            return [i for i in self._ids if i in self._id2slot]

    def matrix(self) -> torch.Tensor:
        with self._lock:
            return self._buf[:self._n]
