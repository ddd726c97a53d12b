"""Prometheus-format metrics registry (reference /metrics endpoint,
pkg/server/server_router.go:108)."""

from __future__ import annotations

import threading
from collections import defaultdict
from typing import Dict, List, Tuple


class MetricsRegistry:
    def __init__(self):
        self._lock = threading.Lock()
        self._counters: Dict[str, float] = defaultdict(float)
        self._gauges: Dict[str, float] = {}
        self._hist: Dict[str, List[float]] = defaultdict(list)

    def inc(self, name: str, value: float = 1.0):
        with self._lock:
            self._counters[name] += value

    def gauge(self, name: str, value: float):
        with self._lock:
            self._gauges[name] = value

    def observe(self, name: str, value: float):
        with self._lock:
            h = self._hist[name]
            h.append(value)
            if len(h) > 10000:
                del h[:5000]

    def render(self) -> str:
        out = []
        with self._lock:
            for k, v in sorted(self._counters.items()):
                out.append(f"# TYPE {k} counter")
                out.append(f"{k} {v}")
            for k, v in sorted(self._gauges.items()):
                out.append(f"# TYPE {k} gauge")
                out.append(f"{k} {v}")
            for k, vals in sorted(self._hist.items()):
                if not vals:
                    continue
                s = sorted(vals)
                out.append(f"# TYPE {k} summary")
                for q in (0.5, 0.95, 0.99):
                    idx = min(int(q * len(s)), len(s) - 1)
                    out.append(f'{k}{{quantile="{q}"}} {s[idx]}')
                out.append(f"{k}_sum {sum(s)}")
                out.append(f"{k}_count {len(s)}")
        return "\n".join(out) + "\n"
