"""Lightweight 1-D Kalman filter.

Parity: reference pkg/filter/kalman.go (imu-f derived), used by decay,
temporal tracking and search ranking.
"""

from __future__ import annotations


class Kalman1D:
    def __init__(self, q: float = 0.01, r: float = 0.1,
                 initial: float = 0.0, p: float = 1.0):
        self.q = q  # process noise
        self.r = r  # measurement noise
        self.x = initial
        self.p = p

    def predict(self) -> float:
        self.p += self.q
        return self.x

    def update(self, measurement: float) -> float:
        self.predict()
        k = self.p / (self.p + self.r)
        self.x += k * (measurement - self.x)
        self.p *= (1 - k)
        return self.x

    @property
    def gain(self) -> float:
        return self.p / (self.p + self.r)
