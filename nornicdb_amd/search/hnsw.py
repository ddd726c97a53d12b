"""HNSW approximate nearest-neighbor index (host-side, numpy distances).

Parity: reference pkg/search/hnsw_index.go (SoA layout :68, config
M=16/efConstruction=200/efSearch=100 :36-50, tombstone deletes + auto
rebuild :144,:298). Cosine distance over L2-normalized vectors.

Design note (MI355X-first): HNSW remains a CPU-side candidate generator,
exactly as in the reference — exact re-scoring of candidates happens on
the GPU (ops.knn / EmbeddingIndex.score_subset). The graph walk is
sequential and latency-bound; the corpus-scale scoring is the GPU's job.
"""

from __future__ import annotations

import heapq
import random
import threading
from typing import Dict, List, Optional, Tuple

import numpy as np


class HNSWIndex:
    def __init__(self, dims: int, m: int = 16, ef_construction: int = 200,
                 ef_search: int = 100, seed: int = 42,
                 tombstone_rebuild_ratio: float = 0.3):
        self.dims = dims
        self.m = m
        self.m0 = 2 * m
        self.efc = ef_construction
        self.efs = ef_search
        self._rng = random.Random(seed)
        self._lock = threading.RLock()
        self._vecs = np.zeros((0, dims), dtype=np.float32)  # SoA storage
        self._ids: List[str] = []
        self._id2slot: Dict[str, int] = {}
        self._levels: List[int] = []
        self._links: List[List[List[int]]] = []  # slot -> level -> neighbors
        self._dead: set = set()
        self._entry: Optional[int] = None
        self._max_level = -1
        self._tomb_ratio = tombstone_rebuild_ratio

    def __len__(self):
        return len(self._ids) - len(self._dead)

    # ---- internals ----
    def _dist(self, q: np.ndarray, slot: int) -> float:
        return 1.0 - float(np.dot(self._vecs[slot], q))

    def _dists(self, q: np.ndarray, slots: List[int]) -> np.ndarray:
        return 1.0 - self._vecs[slots] @ q

    def _random_level(self) -> int:
        lvl = 0
        while self._rng.random() < 0.5 and lvl < 32:
            lvl += 1
        return lvl

    def _search_layer(self, q: np.ndarray, entry: int, ef: int,
                      level: int) -> List[Tuple[float, int]]:
        visited = {entry}
        d0 = self._dist(q, entry)
        cand = [(d0, entry)]                 # min-heap by distance
        best = [(-d0, entry)]                # max-heap (neg) of current top-ef
        while cand:
            d, c = heapq.heappop(cand)
            if d > -best[0][0]:
                break
            neigh = [n for n in self._links[c][level] if n not in visited]
            if not neigh:
                continue
            visited.update(neigh)
            ds = self._dists(q, neigh)
            for nd, n in zip(ds, neigh):
                nd = float(nd)
                if len(best) < ef or nd < -best[0][0]:
                    heapq.heappush(cand, (nd, n))
                    heapq.heappush(best, (-nd, n))
                    if len(best) > ef:
                        heapq.heappop(best)
        return sorted([(-d, s) for d, s in best])

    def _select_neighbors(self, q: np.ndarray, cands: List[Tuple[float, int]],
                          m: int) -> List[int]:
        """Heuristic neighbor selection (keep diverse set)."""
        out: List[int] = []
        for d, c in sorted(cands):
            if len(out) >= m:
                break
            ok = True
            for o in out:
                if 1.0 - float(np.dot(self._vecs[c], self._vecs[o])) < d:
                    ok = False
                    break
            if ok:
                out.append(c)
        if len(out) < m:  # backfill with closest remaining
            chosen = set(out)
            for d, c in sorted(cands):
                if len(out) >= m:
                    break
                if c not in chosen:
                    out.append(c)
        return out

    # ---- public API ----
    def add(self, id_: str, vec) -> None:
        v = np.asarray(vec, dtype=np.float32)
        n = np.linalg.norm(v)
        if n > 0:
            v = v / n
        with self._lock:
            if id_ in self._id2slot:
                slot = self._id2slot[id_]
                self._vecs[slot] = v
                self._dead.discard(slot)
                return
            slot = len(self._ids)
            self._ids.append(id_)
            self._id2slot[id_] = slot
            if slot >= self._vecs.shape[0]:
                grow = max(1024, self._vecs.shape[0])
                self._vecs = np.vstack([self._vecs,
                                        np.zeros((grow, self.dims), np.float32)])
            self._vecs[slot] = v
            lvl = self._random_level()
            self._levels.append(lvl)
            self._links.append([[] for _ in range(lvl + 1)])

            if self._entry is None:
                self._entry = slot
                self._max_level = lvl
                return

            ep = self._entry
            for level in range(self._max_level, lvl, -1):
                res = self._search_layer(v, ep, 1, level)
                if res:
                    ep = res[0][1]
            for level in range(min(lvl, self._max_level), -1, -1):
                cands = self._search_layer(v, ep, self.efc, level)
                mm = self.m0 if level == 0 else self.m
                neigh = self._select_neighbors(v, cands, mm)
                self._links[slot][level] = list(neigh)
                for nb in neigh:
                    nl = self._links[nb][level]
                    nl.append(slot)
                    if len(nl) > mm:
                        ds = self._dists(self._vecs[nb], nl)
                        order = np.argsort(ds)
                        self._links[nb][level] = [nl[i] for i in order[:mm]]
                if cands:
                    ep = cands[0][1]
            if lvl > self._max_level:
                self._max_level = lvl
                self._entry = slot

    def remove(self, id_: str) -> bool:
        """Tombstone delete; triggers rebuild when ratio exceeded."""
        with self._lock:
            slot = self._id2slot.get(id_)
            if slot is None:
                return False
            self._dead.add(slot)
            del self._id2slot[id_]
            if (len(self._ids) > 64
                    and len(self._dead) / len(self._ids) > self._tomb_ratio):
                self._rebuild()
            return True

    def _rebuild(self):
        alive = [(self._ids[s], self._vecs[s])
                 for s in range(len(self._ids)) if s not in self._dead
                 and self._ids[s] in self._id2slot]
        self.__init__(self.dims, self.m, self.efc, self.efs,
                      tombstone_rebuild_ratio=self._tomb_ratio)
        for i, v in alive:
            self.add(i, v)

    def search(self, vec, k: int, ef: Optional[int] = None) -> List[Tuple[str, float]]:
        """Returns [(id, cosine_similarity)] best-first."""
        with self._lock:
            if self._entry is None:
                return []
            q = np.asarray(vec, dtype=np.float32)
            n = np.linalg.norm(q)
            if n > 0:
                q = q / n
            ef = max(ef or self.efs, k)
            ep = self._entry
            for level in range(self._max_level, 0, -1):
                res = self._search_layer(q, ep, 1, level)
                if res:
                    ep = res[0][1]
            res = self._search_layer(q, ep, ef + len(self._dead), 0)
            out = []
            for d, s in res:
                if s in self._dead:
                    continue
                out.append((self._ids[s], 1.0 - d))
                if len(out) >= k:
                    break
            return out

    def ids(self) -> List[str]:
        with self._lock:
            return [self._ids[s] for s in range(len(self._ids))
                    if s not in self._dead and self._ids[s] in self._id2slot]

    def get_vector(self, id_: str):
        with self._lock:
            slot = self._id2slot.get(id_)
            return None if slot is None else self._vecs[slot].copy()
