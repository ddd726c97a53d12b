"""LRU + TTL caches (reference pkg/cache/query_cache.go)."""

from __future__ import annotations

import threading
import time
from collections import OrderedDict
from typing import Any, Optional


class LRUCache:
    def __init__(self, capacity: int = 1024, ttl: Optional[float] = None,
                 now_fn=time.time):
        self.capacity = capacity
        self.ttl = ttl
        self.now = now_fn
        self._lock = threading.Lock()
        self._data: OrderedDict = OrderedDict()
        self.hits = 0
        self.misses = 0

    def get(self, key, default=None):
        with self._lock:
            item = self._data.get(key)
            if item is None:
                self.misses += 1
                return default
            value, ts = item
            if self.ttl is not None and self.now() - ts > self.ttl:
                del self._data[key]
                self.misses += 1
                return default
            self._data.move_to_end(key)
            self.hits += 1
            return value

    def put(self, key, value):
        with self._lock:
            self._data[key] = (value, self.now())
            self._data.move_to_end(key)
            while len(self._data) > self.capacity:
                self._data.popitem(last=False)

    def invalidate(self, key=None):
        with self._lock:
            if key is None:
                self._data.clear()
            else:
                self._data.pop(key, None)

    def __len__(self):
        return len(self._data)


class QueryCache(LRUCache):
    """Parsed-plan / result cache keyed by (query, params-ish)."""

    def __init__(self, capacity: int = 512, ttl: float = 60.0, **kw):
        super().__init__(capacity, ttl, **kw)
