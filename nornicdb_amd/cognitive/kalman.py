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


class VelocityKalman:
    """2-state (position, velocity) Kalman filter — the smoother behind
    query-load prediction (reference pkg/filter VelocityConfig used by
    pkg/temporal/query_load.go). Constant-velocity model."""

    def __init__(self, process_pos: float = 0.5, process_vel: float = 0.1,
                 measurement: float = 2.0, initial_pos_var: float = 100.0,
                 initial_vel_var: float = 10.0):
        self.qp = process_pos
        self.qv = process_vel
        self.r = measurement
        self.x = 0.0   # position (e.g. QPS)
        self.v = 0.0   # velocity (QPS/s)
        # covariance [[pxx, pxv], [pxv, pvv]]
        self.pxx = initial_pos_var
        self.pxv = 0.0
        self.pvv = initial_vel_var

    def update(self, measurement: float, dt: float = 1.0):
        # predict
        x = self.x + self.v * dt
        pxx = self.pxx + dt * (2 * self.pxv + dt * self.pvv) + self.qp
        pxv = self.pxv + dt * self.pvv
        pvv = self.pvv + self.qv
        # update
        s = pxx + self.r
        kx = pxx / s
        kv = pxv / s
        innov = measurement - x
        self.x = x + kx * innov
        self.v = self.v + kv * innov
        self.pxx = (1 - kx) * pxx
        self.pxv = (1 - kx) * pxv
        self.pvv = pvv - kv * pxv
        return self.x, self.v
