"""Vector space registry.

Parity: reference pkg/vectorspace/registry.go — spaces keyed by
(database, entity type, vector name) with dims + distance metric.
"""

from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

COSINE = "cosine"
DOT = "dot"
EUCLIDEAN = "euclidean"


@dataclass
class VectorSpace:
    db: str
    entity_type: str
    name: str
    dims: int
    distance: str = COSINE

    @property
    def key(self) -> Tuple[str, str, str]:
        return (self.db, self.entity_type, self.name)


class Registry:
    def __init__(self):
        self._lock = threading.Lock()
        self._spaces: Dict[Tuple[str, str, str], VectorSpace] = {}

    def register(self, space: VectorSpace) -> VectorSpace:
        with self._lock:
            existing = self._spaces.get(space.key)
            if existing is not None:
                if existing.dims != space.dims or existing.distance != space.distance:
                    raise ValueError(
                        f"vector space {space.key} already registered with "
                        f"dims={existing.dims}/{existing.distance}")
                return existing
            self._spaces[space.key] = space
            return space

    def get(self, db: str, entity_type: str, name: str) -> Optional[VectorSpace]:
        with self._lock:
            return self._spaces.get((db, entity_type, name))

    def list(self, db: str = None) -> List[VectorSpace]:
        with self._lock:
            return [s for s in self._spaces.values() if db is None or s.db == db]

    def drop(self, db: str, entity_type: str, name: str) -> bool:
        with self._lock:
            return self._spaces.pop((db, entity_type, name), None) is not None


GLOBAL = Registry()
