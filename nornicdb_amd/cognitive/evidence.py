"""Evidence buffering + materialization cooldown for inferred edges.

Parity: reference pkg/inference/evidence.go (per-label thresholds on
count / cumulative score / unique sessions / age; materialize only when
ALL are met) and pkg/inference/cooldown.go (per-(src,dst,label) cooldown
window after a materialization so flapping signals don't re-create
edges). Both are first-class, engine-independent components so the
behavior tests mirror the reference's.
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

HOUR = 3600.0


@dataclass
class EvidenceThreshold:
    min_count: int = 3
    min_score: float = 0.5        # on the cumulative average
    min_sessions: int = 2
    max_age: float = 24 * HOUR    # evidence expires

# reference evidence.go DefaultThresholds
DEFAULT_THRESHOLDS: Dict[str, EvidenceThreshold] = {
    "relates_to": EvidenceThreshold(3, 0.5, 2, 24 * HOUR),
    "similar_to": EvidenceThreshold(2, 0.7, 1, 48 * HOUR),
    "coaccess": EvidenceThreshold(5, 0.3, 3, 12 * HOUR),
    "topology": EvidenceThreshold(2, 0.6, 1, 72 * HOUR),
    "depends_on": EvidenceThreshold(3, 0.6, 2, 168 * HOUR),
}
DEFAULT_THRESHOLD = EvidenceThreshold()


@dataclass
class Evidence:
    src: str
    dst: str
    label: str
    count: int = 0
    score_sum: float = 0.0
    first_ts: float = 0.0
    last_ts: float = 0.0
    sessions: set = field(default_factory=set)
    signals: List[str] = field(default_factory=list)

    @property
    def score_avg(self) -> float:
        return self.score_sum / self.count if self.count else 0.0


class EvidenceBuffer:
    """Accumulates signals for potential edges; returns True from
    add_evidence only when the label's full threshold is met."""

    def __init__(self, thresholds: Dict[str, EvidenceThreshold] = None,
                 now_fn=time.time):
        self._lock = threading.Lock()
        self._entries: Dict[Tuple[str, str, str], Evidence] = {}
        self.thresholds = dict(DEFAULT_THRESHOLDS)
        if thresholds:
            self.thresholds.update(thresholds)
        self.now = now_fn
        self.total_added = 0
        self.total_materialized = 0
        self.total_expired = 0

    def _thr(self, label: str) -> EvidenceThreshold:
        return self.thresholds.get(label, DEFAULT_THRESHOLD)

    def set_threshold(self, label: str, thr: EvidenceThreshold):
        with self._lock:
            self.thresholds[label] = thr

    def add_evidence(self, src: str, dst: str, label: str, score: float,
                     signal_type: str, session_id: str = "") -> bool:
        key = (src, dst, label)
        now = self.now()
        thr = self._thr(label)
        with self._lock:
            self.total_added += 1
            ev = self._entries.get(key)
            if ev is not None and now - ev.first_ts > thr.max_age:
                self.total_expired += 1
                ev = None
            if ev is None:
                ev = Evidence(src, dst, label, first_ts=now)
                self._entries[key] = ev
            ev.count += 1
            ev.score_sum += score
            ev.last_ts = now
            if session_id:
                ev.sessions.add(session_id)
            ev.signals.append(signal_type)
            if self._ready(ev, thr):
                del self._entries[key]
                self.total_materialized += 1
                return True
            return False

    @staticmethod
    def _ready(ev: Evidence, thr: EvidenceThreshold) -> bool:
        return (ev.count >= thr.min_count
                and ev.score_avg >= thr.min_score
                and len(ev.sessions) >= thr.min_sessions)

    def check_threshold(self, src, dst, label) -> Tuple[bool, str]:
        """(ready, human reason) — reference CheckThreshold."""
        thr = self._thr(label)
        with self._lock:
            ev = self._entries.get((src, dst, label))
        if ev is None:
            return False, f"no evidence yet (0/{thr.min_count})"
        if ev.count < thr.min_count:
            need = thr.min_count - ev.count
            return False, f"need {need} more signal(s) ({ev.count}/{thr.min_count})"
        if ev.score_avg < thr.min_score:
            return False, (f"avg score {ev.score_avg:.2f} below "
                           f"{thr.min_score:.2f}")
        if len(ev.sessions) < thr.min_sessions:
            return False, (f"need {thr.min_sessions - len(ev.sessions)} more "
                           f"session(s) ({len(ev.sessions)}/{thr.min_sessions})")
        return True, "ready"

    def get_evidence(self, src, dst, label) -> Optional[Evidence]:
        with self._lock:
            return self._entries.get((src, dst, label))

    def pending(self, min_progress: float = 0.0) -> List[Evidence]:
        out = []
        with self._lock:
            for ev in self._entries.values():
                thr = self._thr(ev.label)
                prog = min(ev.count / max(thr.min_count, 1), 1.0)
                if prog >= min_progress:
                    out.append(ev)
        return out

    def cleanup(self) -> int:
        """Drop expired evidence; returns number removed."""
        now = self.now()
        removed = 0
        with self._lock:
            for key in list(self._entries):
                ev = self._entries[key]
                if now - ev.first_ts > self._thr(ev.label).max_age:
                    del self._entries[key]
                    removed += 1
                    self.total_expired += 1
        return removed

    def stats(self) -> dict:
        with self._lock:
            added = self.total_added
            return {
                "entries": len(self._entries),
                "added": added,
                "materialized": self.total_materialized,
                "expired": self.total_expired,
                "materialize_rate": (self.total_materialized / added
                                     if added else 0.0),
            }

    def __len__(self):
        with self._lock:
            return len(self._entries)


# reference cooldown.go DefaultCooldowns
DEFAULT_COOLDOWNS: Dict[str, float] = {
    "relates_to": 10 * 60.0,
    "similar_to": 30 * 60.0,
    "coaccess": 5 * 60.0,
    "topology": 60 * 60.0,
    "depends_on": 2 * 3600.0,
}
DEFAULT_COOLDOWN = 5 * 60.0


class CooldownTable:
    """Per-(src, dst, label) re-materialization cooldown
    (reference pkg/inference/cooldown.go)."""

    def __init__(self, cooldowns: Dict[str, float] = None, now_fn=time.time):
        self._lock = threading.Lock()
        self._last: Dict[Tuple[str, str, str], float] = {}
        self.cooldowns = dict(DEFAULT_COOLDOWNS)
        if cooldowns:
            self.cooldowns.update(cooldowns)
        self.now = now_fn

    def _window(self, label: str) -> float:
        return self.cooldowns.get(label, DEFAULT_COOLDOWN)

    def set_label_cooldown(self, label: str, seconds: float):
        with self._lock:
            self.cooldowns[label] = seconds

    def can_materialize(self, src, dst, label) -> bool:
        with self._lock:
            t = self._last.get((src, dst, label))
        return t is None or self.now() - t >= self._window(label)

    def time_until_allowed(self, src, dst, label) -> float:
        with self._lock:
            t = self._last.get((src, dst, label))
        if t is None:
            return 0.0
        return max(0.0, self._window(label) - (self.now() - t))

    def record_materialization(self, src, dst, label, ts: float = None):
        with self._lock:
            self._last[(src, dst, label)] = ts if ts is not None else self.now()

    def cleanup(self) -> int:
        now = self.now()
        removed = 0
        with self._lock:
            for key in list(self._last):
                if now - self._last[key] >= self._window(key[2]):
                    del self._last[key]
                    removed += 1
        return removed

    def __len__(self):
        with self._lock:
            return len(self._last)
