"""Temporal access tracking: Kalman next-access prediction, session
boundaries, co-access detection, cyclic pattern detection.

Parity: reference pkg/temporal (tracker.go, pattern_detector.go,
query_load.go).
"""

from __future__ import annotations

import math
import threading
import time
from collections import defaultdict, deque
from typing import Dict, List, Optional, Tuple

from .kalman import Kalman1D

SESSION_GAP = 1800.0  # 30 min


class AccessTracker:
    def __init__(self, max_history: int = 256, now_fn=time.time):
        self.now = now_fn
        self._lock = threading.Lock()
        self._hist: Dict[str, deque] = defaultdict(lambda: deque(maxlen=max_history))
        self._interval_kf: Dict[str, Kalman1D] = {}
        self._sessions: List[Tuple[float, float]] = []
        self._last_access_time: Optional[float] = None

    def record(self, node_id: str, ts: float = None):
        ts = ts if ts is not None else self.now()
        with self._lock:
            h = self._hist[node_id]
            if h:
                interval = ts - h[-1]
                kf = self._interval_kf.setdefault(
                    node_id, Kalman1D(q=10.0, r=100.0, initial=interval))
                kf.update(interval)
            h.append(ts)
            # session boundary
            if (self._last_access_time is None
                    or ts - self._last_access_time > SESSION_GAP):
                self._sessions.append((ts, ts))
            else:
                s, _ = self._sessions[-1]
                self._sessions[-1] = (s, ts)
            self._last_access_time = ts

    def predict_next_access(self, node_id: str) -> Optional[float]:
        with self._lock:
            h = self._hist.get(node_id)
            kf = self._interval_kf.get(node_id)
            if not h or kf is None:
                return None
            return h[-1] + max(kf.x, 0.0)

    def access_count(self, node_id: str) -> int:
        with self._lock:
            return len(self._hist.get(node_id, ()))

    def co_accessed(self, window: float = 60.0) -> List[Tuple[str, str, int]]:
        """Pairs accessed within `window` seconds of each other."""
        with self._lock:
            events = sorted(
                (ts, nid) for nid, h in self._hist.items() for ts in h)
        pairs = defaultdict(int)
        for i, (ts, nid) in enumerate(events):
            j = i + 1
            while j < len(events) and events[j][0] - ts <= window:
                other = events[j][1]
                if other != nid:
                    pairs[tuple(sorted((nid, other)))] += 1
                j += 1
        return [(a, b, c) for (a, b), c in
                sorted(pairs.items(), key=lambda kv: -kv[1])]

    def sessions(self) -> List[Tuple[float, float]]:
        with self._lock:
            return list(self._sessions)

    @property
    def session_id(self) -> str:
        """Stable identifier of the CURRENT session (start timestamp);
        feeds evidence buffering's unique-session counting."""
        with self._lock:
            if not self._sessions:
                return "s0"
            return f"s{int(self._sessions[-1][0])}"

    def is_session_boundary(self, ts: float = None) -> bool:
        """True when the next access at `ts` would start a new session
        (reference tracker.go IsSessionBoundary: inactivity gap)."""
        ts = ts if ts is not None else self.now()
        with self._lock:
            return (self._last_access_time is None
                    or ts - self._last_access_time > SESSION_GAP)

    def detect_period(self, node_id: str) -> Optional[float]:
        """Dominant access period via mean/variance test on intervals."""
        with self._lock:
            h = list(self._hist.get(node_id, ()))
        if len(h) < 4:
            return None
        intervals = [b - a for a, b in zip(h, h[1:])]
        mean = sum(intervals) / len(intervals)
        if mean <= 0:
            return None
        var = sum((x - mean) ** 2 for x in intervals) / len(intervals)
        cv = math.sqrt(var) / mean
        return mean if cv < 0.5 else None  # regular enough to be a cycle


class QueryLoadTracker:
    """Adaptive decay pacing by query load (reference query_load.go)."""

    def __init__(self, window: float = 60.0, now_fn=time.time):
        self.window = window
        self.now = now_fn
        self._times: deque = deque()
        self._lock = threading.Lock()

    def record_query(self):
        with self._lock:
            t = self.now()
            self._times.append(t)
            while self._times and self._times[0] < t - self.window:
                self._times.popleft()

    def qps(self) -> float:
        with self._lock:
            t = self.now()
            while self._times and self._times[0] < t - self.window:
                self._times.popleft()
            return len(self._times) / self.window

    def decay_interval(self, base: float = 300.0) -> float:
        """Back off decay cycles under load."""
        load = self.qps()
        return base * (1.0 + min(load / 10.0, 10.0))


class RelationshipEvolution:
    """Tracks how relationship strength evolves with co-access
    (reference pkg/temporal/relationship_evolution.go): every co-access
    of two connected nodes reinforces the edge; idle time decays it.
    Strength feeds decay protection and link-prediction priors."""

    def __init__(self, engine, reinforce: float = 0.1,
                 half_life_s: float = 7 * 86400.0, now_fn=time.time):
        self.engine = engine
        self.reinforce = reinforce
        self.half_life_s = half_life_s
        self.now = now_fn
        self._strength: Dict[str, float] = {}
        self._last: Dict[str, float] = {}

    def _decayed(self, eid: str) -> float:
        s = self._strength.get(eid, 0.5)
        last = self._last.get(eid)
        if last is None:
            return s
        dt = max(0.0, self.now() - last)
        return s * (0.5 ** (dt / self.half_life_s))

    def record_coaccess(self, edge_id: str) -> float:
        s = self._decayed(edge_id)
        s = min(1.0, s + self.reinforce * (1.0 - s))
        self._strength[edge_id] = s
        self._last[edge_id] = self.now()
        return s

    def strength(self, edge_id: str) -> float:
        return self._decayed(edge_id)

    def evolution_class(self, edge_id: str) -> str:
        s = self.strength(edge_id)
        if s >= 0.8:
            return "strengthening"
        if s >= 0.4:
            return "stable"
        return "fading"

    def persist(self) -> int:
        """Write current strengths into edge properties (_strength)."""
        n = 0
        for eid in list(self._strength):
            try:
                e = self.engine.get_edge(eid)
                e.properties["_strength"] = round(self.strength(eid), 4)
                self.engine.update_edge(e)
                n += 1
            except Exception:
                self._strength.pop(eid, None)
        return n
