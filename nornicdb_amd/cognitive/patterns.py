"""Access-pattern detection + query-load prediction.

Parity: reference pkg/temporal/pattern_detector.go (daily / weekly /
burst / trend patterns from hour- and day-histograms with
concentration-based confidence) and pkg/temporal/query_load.go
(Kalman-smoothed QPS with velocity trend, per-hour baselines, anomaly
flags and adaptive-decay hooks). Session boundaries live in
cognitive.temporal.AccessTracker.
"""

from __future__ import annotations

import math
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .kalman import VelocityKalman

DAILY = "daily"
WEEKLY = "weekly"
BURST = "burst"
GROWING = "growing"
DECAYING = "decaying"


@dataclass
class DetectedPattern:
    type: str
    confidence: float
    peak_hour: int = -1
    peak_day: int = -1
    period: float = 0.0
    last_seen: float = 0.0


@dataclass
class PatternConfig:
    min_samples: int = 10
    daily_confidence: float = 0.3
    weekly_confidence: float = 0.4
    burst_window: float = 60.0
    burst_min_accesses: int = 5
    growth_threshold: float = 0.05
    decay_threshold: float = -0.05


class PatternDetector:
    """Per-node daily/weekly/burst/trend pattern detection
    (reference pattern_detector.go:165-343)."""

    def __init__(self, config: PatternConfig = None, now_fn=time.time):
        self.cfg = config or PatternConfig()
        self.now = now_fn
        self._lock = threading.Lock()
        self._nodes: Dict[str, dict] = {}

    def _data(self, node_id):
        d = self._nodes.get(node_id)
        if d is None:
            d = {"hours": [0] * 24, "days": [0] * 7, "recent": [], "total": 0}
            self._nodes[node_id] = d
        return d

    def record_access(self, node_id: str, ts: float = None):
        ts = ts if ts is not None else self.now()
        lt = time.localtime(ts)
        with self._lock:
            d = self._data(node_id)
            d["hours"][lt.tm_hour] += 1
            d["days"][lt.tm_wday] += 1  # note: python Mon=0; stable mapping
            d["total"] += 1
            d["recent"].append(ts)
            cut = ts - self.cfg.burst_window
            d["recent"] = [t for t in d["recent"] if t >= cut][-64:]

    @staticmethod
    def _concentration_confidence(counts, buckets, scale=3.0):
        total = sum(counts)
        if total == 0:
            return 0.0, 0
        peak = max(range(len(counts)), key=counts.__getitem__)
        expected = total / buckets
        concentration = counts[peak] / expected
        conf = (concentration - 1.0) / scale  # reference: 4x conc = 1.0
        return max(0.0, min(1.0, conf)), peak

    def detect_patterns(self, node_id: str,
                        velocity: float = 0.0) -> List[DetectedPattern]:
        with self._lock:
            d = self._nodes.get(node_id)
            if d is None or d["total"] < self.cfg.min_samples:
                # trend patterns need no history (reference behavior)
                return self._trend_only(velocity)
            out: List[DetectedPattern] = []
            conf, peak = self._concentration_confidence(d["hours"], 24)
            if conf >= self.cfg.daily_confidence:
                out.append(DetectedPattern(DAILY, conf, peak_hour=peak,
                                           period=86400.0,
                                           last_seen=self.now()))
            conf, peak = self._concentration_confidence(d["days"], 7, 2.0)
            if conf >= self.cfg.weekly_confidence:
                out.append(DetectedPattern(WEEKLY, conf, peak_day=peak,
                                           period=7 * 86400.0,
                                           last_seen=self.now()))
            now = self.now()
            recent = [t for t in d["recent"]
                      if t >= now - self.cfg.burst_window]
            if len(recent) >= self.cfg.burst_min_accesses:
                conf = min(1.0, len(recent)
                           / (2.0 * self.cfg.burst_min_accesses))
                out.append(DetectedPattern(BURST, conf,
                                           period=self.cfg.burst_window,
                                           last_seen=now))
            out.extend(self._trend_only(velocity))
            return out

    def _trend_only(self, velocity: float) -> List[DetectedPattern]:
        if velocity > self.cfg.growth_threshold:
            return [DetectedPattern(GROWING,
                                    min(1.0, velocity
                                        / (4 * self.cfg.growth_threshold)),
                                    last_seen=self.now())]
        if velocity < self.cfg.decay_threshold:
            return [DetectedPattern(DECAYING,
                                    min(1.0, abs(velocity)
                                        / (4 * abs(self.cfg.decay_threshold))),
                                    last_seen=self.now())]
        return []

    def has_pattern(self, node_id: str, ptype: str,
                    velocity: float = 0.0) -> bool:
        return any(p.type == ptype
                   for p in self.detect_patterns(node_id, velocity))

    def peak_access_time(self, node_id: str):
        """(hour, day, confidence) of the node's access concentration."""
        with self._lock:
            d = self._nodes.get(node_id)
            if d is None or d["total"] == 0:
                return -1, -1, 0.0
            hconf, hour = self._concentration_confidence(d["hours"], 24)
            dconf, day = self._concentration_confidence(d["days"], 7, 2.0)
            return hour, day, max(hconf, dconf)

    def reset_node(self, node_id: str):
        with self._lock:
            self._nodes.pop(node_id, None)


@dataclass
class LoadPrediction:
    current_qps: float
    raw_qps: float
    velocity: float
    trend: str                 # increasing / decreasing / stable
    predicted_qps_5m: float
    predicted_qps_15m: float
    predicted_qps_1h: float
    is_anomaly: bool
    anomaly_type: str          # spike / drop / ""
    peak_hour: int
    is_near_peak: bool
    total_queries: int


class QueryLoadPredictor:
    """Kalman-smoothed query-load model with trend, short-horizon
    prediction, per-hour peak tracking and spike/drop anomaly flags
    (reference pkg/temporal/query_load.go). decay_interval() is the
    query-load-adaptive decay hook: high load stretches the decay
    sweep so background maintenance yields to query traffic."""

    def __init__(self, bucket_seconds: float = 1.0, spike_threshold: float = 5.0,
                 drop_threshold: float = -5.0, now_fn=time.time):
        self.bucket = bucket_seconds
        self.spike_threshold = spike_threshold
        self.drop_threshold = drop_threshold
        self.now = now_fn
        self._lock = threading.Lock()
        self._kf = VelocityKalman(process_pos=0.5, process_vel=0.1,
                                  measurement=2.0)
        self._bucket_start = self.now()
        self._bucket_count = 0
        self._total = 0
        self._hour_counts = [0.0] * 24
        self._qps = 0.0
        self._vel = 0.0

    def record_query(self, ts: float = None):
        self.record_queries(1, ts)

    def record_queries(self, count: int, ts: float = None):
        ts = ts if ts is not None else self.now()
        with self._lock:
            while ts - self._bucket_start >= self.bucket:
                self._flush_locked()
                self._bucket_start += self.bucket
            self._bucket_count += count
            self._total += count
            self._hour_counts[time.localtime(ts).tm_hour] += count

    def _flush_locked(self):
        qps = self._bucket_count / self.bucket
        self._qps, self._vel = self._kf.update(qps, dt=self.bucket)
        self._bucket_count = 0

    def prediction(self) -> LoadPrediction:
        with self._lock:
            now = self.now()
            # flush any elapsed buckets so idle periods decay the estimate
            while now - self._bucket_start >= self.bucket:
                self._flush_locked()
                self._bucket_start += self.bucket
            raw = self._bucket_count / max(now - self._bucket_start, 1e-9)
            qps, vel = self._qps, self._vel
            trend = ("increasing" if vel > 0.1
                     else "decreasing" if vel < -0.1 else "stable")
            anomaly = ""
            if vel > self.spike_threshold:
                anomaly = "spike"
            elif vel < self.drop_threshold:
                anomaly = "drop"
            peak_hour = max(range(24), key=self._hour_counts.__getitem__) \
                if any(self._hour_counts) else -1
            cur_hour = time.localtime(now).tm_hour
            near = peak_hour >= 0 and min((cur_hour - peak_hour) % 24,
                                          (peak_hour - cur_hour) % 24) <= 1
            clamp = lambda v: max(0.0, v)
            return LoadPrediction(
                current_qps=qps, raw_qps=raw, velocity=vel, trend=trend,
                predicted_qps_5m=clamp(qps + vel * 300),
                predicted_qps_15m=clamp(qps + vel * 900),
                predicted_qps_1h=clamp(qps + vel * 3600),
                is_anomaly=bool(anomaly), anomaly_type=anomaly,
                peak_hour=peak_hour, is_near_peak=near,
                total_queries=self._total)

    # ---- adaptive decay hook (reference query_load.go + decay wiring) ----
    def decay_interval(self, base: float = 300.0, max_stretch: float = 8.0,
                       target_qps: float = 50.0) -> float:
        """Stretch the decay sweep interval under load: at 0 QPS the
        sweep runs at `base`; at >= target_qps it runs `max_stretch`
        slower (background work yields to queries)."""
        p = self.prediction()
        load = min(p.current_qps / target_qps, 1.0)
        return base * (1.0 + (max_stretch - 1.0) * load)

    def should_scale_up(self, threshold_qps: float) -> bool:
        p = self.prediction()
        return p.predicted_qps_5m > threshold_qps and p.velocity > 0

    def should_scale_down(self, threshold_qps: float, min_qps: float) -> bool:
        p = self.prediction()
        return p.predicted_qps_15m < threshold_qps and p.current_qps < min_qps
