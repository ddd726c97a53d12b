"""Cypher scalar + aggregate function registry.

Parity: reference pkg/cypher/functions.go + fn/registry.go. APOC functions
register into the same table under their dotted names (apoc/ registry).
"""

from __future__ import annotations

import json
import math
import random
import re
import time
import uuid
from typing import Any, Callable, Dict, List

from ..storage.types import Edge, Node


class CypherRuntimeError(Exception):
    pass


FUNCTIONS: Dict[str, Callable] = {}


def register(name: str, fn: Callable = None):
    if fn is not None:
        FUNCTIONS[name.lower()] = fn
        return fn

    def deco(f):
        FUNCTIONS[name.lower()] = f
        return f
    return deco


def _num(x):
    if isinstance(x, bool) or not isinstance(x, (int, float)):
        raise CypherRuntimeError(f"expected number, got {type(x).__name__}")
    return x


# ---- entity functions ----
register("id", lambda x: None if x is None else getattr(x, "id", None))
register("elementid", lambda x: None if x is None else getattr(x, "id", None))
register("labels", lambda n: None if n is None else list(n.labels))
register("type", lambda e: None if e is None else e.type)
register("properties", lambda x: None if x is None else
         (dict(x.properties) if isinstance(x, (Node, Edge)) else dict(x)))
register("keys", lambda x: None if x is None else
         (sorted(x.properties.keys()) if isinstance(x, (Node, Edge)) else sorted(x.keys())))
register("startnode", lambda e: None if e is None else e._start_ref if hasattr(e, "_start_ref") else e.start_node)
register("endnode", lambda e: None if e is None else e._end_ref if hasattr(e, "_end_ref") else e.end_node)


# ---- scalar ----
register("coalesce", lambda *a: next((x for x in a if x is not None), None))
register("head", lambda l: None if not l else l[0])
register("last", lambda l: None if not l else l[-1])
register("tail", lambda l: None if l is None else list(l[1:]))
register("size", lambda x: None if x is None else len(x))
register("length", lambda x: None if x is None else len(x))
register("reverse", lambda x: None if x is None else
         (x[::-1] if isinstance(x, str) else list(reversed(x))))
register("range", lambda a, b, step=1: list(range(int(a), int(b) + (1 if step > 0 else -1), int(step))))
register("abs", lambda x: None if x is None else abs(_num(x)))
register("sign", lambda x: None if x is None else (0 if x == 0 else math.copysign(1, _num(x))))
register("rand", lambda: random.random())
register("randomuuid", lambda: str(uuid.uuid4()))
register("timestamp", lambda: int(time.time() * 1000))

# ---- temporal (see temporal.py) ----
from . import temporal as _tp  # noqa: E402

register("date", lambda arg=None: _tp.make_date(arg))
register("datetime", lambda arg=None: _tp.make_datetime(arg))
register("localdatetime", lambda arg=None: _tp.make_datetime(arg, local=True))
register("time", lambda arg=None: _tp.make_time(arg))
register("localtime", lambda arg=None: _tp.make_time(arg, local=True))
register("duration", lambda arg: _tp.make_duration(arg))
register("duration.between", lambda a, b: _tp.duration_between(a, b))
def _dur_norm(a, b=None):
    """1-arg form converts a duration; 2-arg form measures between temporals
    (reference: functions_eval_functions.go duration.inDays both arities)."""
    return _tp.make_duration(a) if b is None and isinstance(
        a, (_tp.CypherDuration, str)) else _tp.duration_between(a, b)


register("duration.indays", lambda a, b=None: _tp.CypherDuration(
    0, int(_dur_norm(a, b).total_seconds_approx() // 86400), 0, 0))
register("duration.inseconds", lambda a, b=None: _tp.CypherDuration(
    0, 0, int(_dur_norm(a, b).total_seconds_approx()), 0))
register("duration.inmonths", lambda a, b=None: _tp.CypherDuration(
    int(_dur_norm(a, b).total_seconds_approx() // (30.4375 * 86400)), 0, 0, 0))
register("datetime.truncate", lambda unit, v=None:
         _tp.truncate(unit, v if v is not None else _tp.make_datetime(None)))
register("date.truncate", lambda unit, v=None:
         _tp.truncate(unit, v if v is not None else _tp.make_datetime(None),
                      kind="date"))
register("datetime.fromepoch", lambda s, ns=0: _tp.make_datetime(
    (float(s) + float(ns) / 1e9) * 1000))
register("datetime.fromepochmillis", lambda ms: _tp.make_datetime(float(ms)))
register("toInteger".lower(), lambda x: _to_int(x))
register("tofloat", lambda x: _to_float(x))
register("tostring", lambda x: None if x is None else
         (str(x).lower() if isinstance(x, bool) else str(x)))
register("toboolean", lambda x: _to_bool(x))


def _to_int(x):
    if x is None:
        return None
    try:
        if isinstance(x, str):
            return int(float(x)) if ("." in x or "e" in x.lower()) else int(x)
        if isinstance(x, bool):
            return 1 if x else 0
        return int(x)
    except (ValueError, TypeError):
        return None


def _to_float(x):
    if x is None:
        return None
    try:
        return float(x)
    except (ValueError, TypeError):
        return None


def _to_bool(x):
    if x is None or isinstance(x, bool):
        return x
    if isinstance(x, str):
        return {"true": True, "false": False}.get(x.lower())
    return None


# ---- math ----
for n, f in dict(
    ceil=math.ceil, floor=math.floor, round=round, sqrt=math.sqrt,
    exp=math.exp, log=math.log, log10=math.log10, sin=math.sin, cos=math.cos,
    tan=math.tan, asin=math.asin, acos=math.acos, atan=math.atan,
    degrees=math.degrees, radians=math.radians,
).items():
    register(n, (lambda f: lambda x: None if x is None else f(_num(x)))(f))
register("atan2", lambda y, x: math.atan2(_num(y), _num(x)))
register("pi", lambda: math.pi)
register("e", lambda: math.e)
register("haversin", lambda x: None if x is None else (1 - math.cos(_num(x))) / 2)


# ---- strings ----
register("toupper", lambda s: None if s is None else s.upper())
register("tolower", lambda s: None if s is None else s.lower())
register("upper", lambda s: None if s is None else s.upper())
register("lower", lambda s: None if s is None else s.lower())
register("trim", lambda s: None if s is None else s.strip())
register("ltrim", lambda s: None if s is None else s.lstrip())
register("rtrim", lambda s: None if s is None else s.rstrip())
register("replace", lambda s, a, b: None if s is None else s.replace(a, b))
register("split", lambda s, d: None if s is None else s.split(d))
register("substring", lambda s, start, length=None:
         None if s is None else (s[start:start + length] if length is not None else s[start:]))
register("left", lambda s, n: None if s is None else s[:n])
register("right", lambda s, n: None if s is None else s[-n:] if n else "")


# ---- list/aggregation helpers ----
register("__haslabels", lambda n, labels: n is not None and all(lb in n.labels for lb in labels))
register("nodes", lambda p: None if p is None else p.nodes)
register("relationships", lambda p: None if p is None else p.edges)
register("reduce", None)  # handled in evaluator (needs lazy eval)


# ---- aggregates (handled by executor; names listed for detection) ----
AGGREGATES = {"count", "sum", "avg", "min", "max", "collect", "stdev",
              "stdevp", "percentilecont", "percentiledisc"}

# extension point: apoc.agg.* finalizers, registered by apoc.breadth
# (values list, extra arg) -> result
AGG_FINALIZERS: Dict[str, Callable] = {}


def is_aggregate(name: str) -> bool:
    return name.lower() in AGGREGATES


class Aggregator:
    def __init__(self, name: str, distinct: bool = False):
        self.name = name.lower()
        self.distinct = distinct
        self.values: List[Any] = []
        self.seen = set()
        self.count = 0

    def add(self, v):
        if self.name == "count" and v is None:
            return
        if self.distinct:
            key = repr(v)
            if key in self.seen:
                return
            self.seen.add(key)
        if v is not None or self.name == "collect":
            self.values.append(v)
        self.count += 1

    def result(self, extra=None):
        vals = [v for v in self.values if v is not None]
        n = self.name
        if n == "count":
            return self.count
        if n == "collect":
            return [v for v in self.values if v is not None]
        if n == "sum":
            return sum(vals) if vals else 0
        if n == "avg":
            return sum(vals) / len(vals) if vals else None
        if n == "min":
            return min(vals) if vals else None
        if n == "max":
            return max(vals) if vals else None
        if n in ("stdev", "stdevp"):
            if len(vals) < 2:
                return 0.0
            m = sum(vals) / len(vals)
            var = sum((v - m) ** 2 for v in vals)
            var /= (len(vals) - 1) if n == "stdev" else len(vals)
            return math.sqrt(var)
        if n in ("percentilecont", "percentiledisc"):
            if not vals:
                return None
            p = extra if extra is not None else 0.5
            s = sorted(vals)
            if n == "percentiledisc":
                return s[min(int(p * len(s)), len(s) - 1)]
            idx = p * (len(s) - 1)
            lo, hi = int(math.floor(idx)), int(math.ceil(idx))
            if lo == hi:
                return s[lo]
            return s[lo] + (s[hi] - s[lo]) * (idx - lo)
        fin = AGG_FINALIZERS.get(n)
        if fin is not None:
            return fin(self.values, extra)
        raise CypherRuntimeError(f"unknown aggregate {n}")


# ---- Cypher 5 additions: OrNull casts, valueType, point, normalize ----
register("char_length", lambda s: None if s is None else len(s))
register("character_length", lambda s: None if s is None else len(s))
register("btrim", lambda s, chars=None: None if s is None else
         (s.strip() if chars is None else s.strip(chars)))
register("normalize", lambda s, form="NFC": None if s is None else
         __import__("unicodedata").normalize(form, s))
register("isnan", lambda x: isinstance(x, float) and x != x)
register("nullif", lambda a, b: None if a == b else a)
register("tointegerornull", lambda x: _to_int(x))
register("tofloatornull", lambda x: _to_float(x))
register("tobooleanornull", lambda x: _to_bool(x))
register("tostringornull", lambda x: None if not isinstance(
    x, (str, bool, int, float)) else (str(x).lower() if isinstance(x, bool)
                                      else str(x)))


def _value_type(x):
    if x is None:
        return "NULL"
    if isinstance(x, bool):
        return "BOOLEAN NOT NULL"
    if isinstance(x, int):
        return "INTEGER NOT NULL"
    if isinstance(x, float):
        return "FLOAT NOT NULL"
    if isinstance(x, str):
        return "STRING NOT NULL"
    if isinstance(x, list):
        return "LIST<ANY> NOT NULL"
    if isinstance(x, dict):
        return "MAP NOT NULL"
    if isinstance(x, Node):
        return "NODE NOT NULL"
    if isinstance(x, Edge):
        return "RELATIONSHIP NOT NULL"
    return type(x).__name__.upper() + " NOT NULL"


register("valuetype", _value_type)


class CypherPoint:
    """2D/3D point (cartesian or WGS-84); reference supports points via
    its Cypher layer. Accessors via component() like temporal values."""

    __slots__ = ("x", "y", "z", "crs")

    def __init__(self, m):
        m = {k.lower(): v for k, v in dict(m).items()}
        if "latitude" in m or "longitude" in m:
            self.crs = "wgs-84"
            self.x = float(m.get("longitude", 0.0))
            self.y = float(m.get("latitude", 0.0))
        else:
            self.crs = m.get("crs", "cartesian")
            self.x = float(m.get("x", 0.0))
            self.y = float(m.get("y", 0.0))
        self.z = float(m["z"]) if "z" in m else (
            float(m["height"]) if "height" in m else None)

    def component(self, key):
        k = key.lower()
        vals = {"x": self.x, "y": self.y, "z": self.z, "crs": self.crs,
                "longitude": self.x, "latitude": self.y, "height": self.z,
                "srid": 4326 if self.crs == "wgs-84" else 7203}
        if k in vals:
            return vals[k]
        raise KeyError(key)

    def __eq__(self, o):
        return (isinstance(o, CypherPoint) and self.crs == o.crs
                and (self.x, self.y, self.z) == (o.x, o.y, o.z))

    def __hash__(self):
        return hash(("point", self.crs, self.x, self.y, self.z))

    def __str__(self):
        z = f", z: {self.z}" if self.z is not None else ""
        return f"point({{x: {self.x}, y: {self.y}{z}, crs: '{self.crs}'}})"

    __repr__ = __str__


def _point_distance(a, b):
    if a is None or b is None:
        return None
    if a.crs == "wgs-84" and b.crs == "wgs-84":
        # haversine metres
        import math as _m
        p1, p2 = _m.radians(a.y), _m.radians(b.y)
        dp = _m.radians(b.y - a.y)
        dl = _m.radians(b.x - a.x)
        h = _m.sin(dp / 2) ** 2 + _m.cos(p1) * _m.cos(p2) * _m.sin(dl / 2) ** 2
        return 2 * 6371008.8 * _m.asin(_m.sqrt(h))
    dz = ((a.z or 0) - (b.z or 0)) ** 2 if (a.z is not None or
                                            b.z is not None) else 0
    return math.sqrt((a.x - b.x) ** 2 + (a.y - b.y) ** 2 + dz)


register("point", lambda m: None if m is None else CypherPoint(m))
register("point.distance", _point_distance)
register("distance", _point_distance)
register("point.withinbbox", lambda p, lo, hi: None if p is None else
         (lo.x <= p.x <= hi.x and lo.y <= p.y <= hi.y))


# Neo4j math semantics: out-of-domain returns NaN (not an error)
register("sqrt", lambda x: None if x is None else (
    float("nan") if x < 0 else math.sqrt(_num(x))))
register("log", lambda x: None if x is None else (
    float("nan") if x <= 0 else math.log(_num(x))))
# hyperbolic family + power (Neo4j 2025.06 surface,
# reference functions_eval_math.go:188-230)
register("sinh", lambda x: None if x is None else math.sinh(_num(x)))
register("cosh", lambda x: None if x is None else math.cosh(_num(x)))
register("tanh", lambda x: None if x is None else math.tanh(_num(x)))
register("coth", lambda x: None if x is None else (
    float("nan") if _num(x) == 0 else math.cosh(_num(x)) / math.sinh(_num(x))))
register("power", lambda b, e: None if b is None or e is None
         else _num(b) ** _num(e))


def _date_comp(arg, comp):
    d = arg if isinstance(arg, (_tp.CypherDate, _tp.CypherDateTime)) \
        else _tp.make_date(arg)
    return d.component(comp)


# date.year/month/day accessors on ISO strings or temporals
# (reference functions_eval_functions.go:1402-1496)
register("date.year", lambda a: _date_comp(a, "year"))
register("date.month", lambda a: _date_comp(a, "month"))
register("date.day", lambda a: _date_comp(a, "day"))
register("date.week", lambda a: _date_comp(a, "week"))
register("date.quarter", lambda a: _date_comp(a, "quarter"))
register("date.dayofweek", lambda a: _date_comp(a, "dayOfWeek"))
register("date.dayofyear", lambda a: _date_comp(a, "ordinalDay"))
register("apoc.create.uuid", lambda: str(uuid.uuid4()))

register("log10", lambda x: None if x is None else (
    float("nan") if x <= 0 else math.log10(_num(x))))
register("asin", lambda x: None if x is None else (
    float("nan") if abs(x) > 1 else math.asin(_num(x))))
register("acos", lambda x: None if x is None else (
    float("nan") if abs(x) > 1 else math.acos(_num(x))))


# ---- kalman.* functions (JSON-state filters, pkg/cypher/kalman_functions.go) ----
# State travels as a JSON string the caller stores in a node property:
#   s = kalman.init({processNoise: 0.1}); r = kalman.process(23.5, s);
#   r.value is the smoothed estimate, r.state the updated JSON.
# Three families: kalman.* (1-state scalar), kalman.velocity.* (2-state
# position+velocity), kalman.adaptive.* (auto-switches between the two).


def _kj_default():
    return {"x": 0.0, "lx": 0.0, "p": 30.0, "k": 0.0, "e": 1.0,
            "q": 0.0001, "r": 88.0, "vs": 10.0, "n": 0}


def _kvj_default():
    return {"pos": 0.0, "vel": 0.0, "p": [100.0, 0.0, 0.0, 10.0],
            "qp": 0.1, "qv": 0.01, "r": 1.0, "dt": 1.0, "n": 0}


def _kaj_default():
    return {"basic": _kj_default(), "velocity": _kvj_default(),
            "mode": "basic", "ss": 0, "tt": 0.1, "st": 0.02, "hy": 10,
            "n": 0, "lf": 0.0, "ts": 0.0}


def _kalman_init(config=None):
    st = _kj_default()
    cfg = config if isinstance(config, dict) else {}
    if "processNoise" in cfg:
        st["q"] = float(cfg["processNoise"]) * 0.001
    if "measurementNoise" in cfg:
        st["r"] = float(cfg["measurementNoise"])
    if "initialCovariance" in cfg:
        st["p"] = float(cfg["initialCovariance"])
    if "varianceScale" in cfg:
        st["vs"] = float(cfg["varianceScale"])
    return json.dumps(st)


def _kalman_process(measurement, state, target=0.0):
    m = float(measurement)
    try:
        st = json.loads(state)
        assert isinstance(st, dict) and "x" in st
    except Exception:
        return {"value": m, "state": state, "error": "invalid state"}
    velocity = st["x"] - st["lx"]
    st["x"] += velocity
    st["lx"] = st["x"]
    if target and st["lx"]:
        st["e"] = abs(1.0 - (float(target) / st["lx"]))
    else:
        st["e"] = 1.0
    st["p"] = st["p"] + st["q"] * st["e"]
    st["k"] = st["p"] / (st["p"] + st["r"])
    st["x"] += st["k"] * (m - st["x"])
    st["p"] = (1.0 - st["k"]) * st["p"]
    st["n"] += 1
    return {"value": st["x"], "state": json.dumps(st)}


def _kalman_predict(state, steps=1):
    try:
        st = json.loads(state)
        return st["x"] + float(steps) * (st["x"] - st["lx"])
    except Exception:
        return 0.0


def _kalman_state_value(state):
    try:
        return json.loads(state)["x"]
    except Exception:
        return 0.0


def _kalman_rate(state):
    try:
        st = json.loads(state)
        return st["x"] - st["lx"]
    except Exception:
        return 0.0


def _kalman_vel_init(initial_pos=None, initial_vel=None):
    st = _kvj_default()
    if initial_pos is not None:
        st["pos"] = float(initial_pos)
        st["vel"] = float(initial_vel or 0.0)
    return json.dumps(st)


def _kalman_vel_process(measurement, state):
    m = float(measurement)
    try:
        st = json.loads(state)
        assert isinstance(st, dict) and "pos" in st
    except Exception:
        return {"value": m, "velocity": 0.0, "state": state,
                "error": "invalid state"}
    dt = st["dt"] if st["dt"] > 0 else 1.0
    pred_pos = st["pos"] + st["vel"] * dt
    pred_vel = st["vel"]
    p00, p01, p10, p11 = st["p"]
    pp00 = p00 + dt * p10 + dt * p01 + dt * dt * p11 + st["qp"]
    pp01 = p01 + dt * p11
    pp10 = p10 + dt * p11
    pp11 = p11 + st["qv"]
    innov = m - pred_pos
    sc = pp00 + st["r"]
    k0, k1 = pp00 / sc, pp10 / sc
    st["pos"] = pred_pos + k0 * innov
    st["vel"] = pred_vel + k1 * innov
    st["p"] = [(1 - k0) * pp00, (1 - k0) * pp01,
               pp10 - k1 * pp00, pp11 - k1 * pp01]
    st["n"] += 1
    return {"value": st["pos"], "velocity": st["vel"],
            "state": json.dumps(st)}


def _kalman_vel_predict(state, steps=1):
    try:
        st = json.loads(state)
        dt = st["dt"] if st["dt"] > 0 else 1.0
        return st["pos"] + st["vel"] * float(steps) * dt
    except Exception:
        return 0.0


def _kalman_adaptive_init(config=None):
    st = _kaj_default()
    cfg = config if isinstance(config, dict) else {}
    if "trendThreshold" in cfg:
        st["tt"] = float(cfg["trendThreshold"])
    if "stabilityThreshold" in cfg:
        st["st"] = float(cfg["stabilityThreshold"])
    if "hysteresis" in cfg:
        st["hy"] = int(cfg["hysteresis"])
    if cfg.get("initialMode") == "velocity":
        st["mode"] = "velocity"
    return json.dumps(st)


def _kalman_adaptive_process(measurement, state):
    m = float(measurement)
    try:
        st = json.loads(state)
        assert isinstance(st, dict) and "mode" in st
    except Exception:
        return {"value": m, "mode": "error", "state": state,
                "error": "invalid state"}
    if st["mode"] == "velocity":
        r = _kalman_vel_process(m, json.dumps(st["velocity"]))
        filtered = r["value"]
        st["velocity"] = json.loads(r["state"])
        st["ts"] = st["velocity"]["vel"]
    else:
        r = _kalman_process(m, json.dumps(st["basic"]))
        filtered = r["value"]
        st["basic"] = json.loads(r["state"])
        st["ts"] = st["basic"]["x"] - st["basic"]["lx"]
    st["n"] += 1
    st["ss"] += 1
    if st["ss"] >= st["hy"]:
        mag = abs(st["ts"])
        if st["mode"] == "basic" and mag > st["tt"]:
            st["mode"] = "velocity"
            st["ss"] = 0
            st["velocity"]["pos"] = st["basic"]["x"]
            st["velocity"]["vel"] = st["ts"]
        elif st["mode"] == "velocity" and mag < st["st"]:
            st["mode"] = "basic"
            st["ss"] = 0
            st["basic"]["x"] = st["velocity"]["pos"]
            st["basic"]["lx"] = st["velocity"]["pos"] - st["velocity"]["vel"]
    st["lf"] = filtered
    return {"value": filtered, "mode": st["mode"], "state": json.dumps(st)}


def _kalman_reset(state):
    try:
        st = json.loads(state)
    except Exception:
        return _kalman_init()
    if isinstance(st, dict) and "pos" in st:
        return _kalman_vel_init()
    if isinstance(st, dict) and "mode" in st:
        return _kalman_adaptive_init()
    return _kalman_init()


register("kalman.init", _kalman_init)
register("kalman.process", _kalman_process)
register("kalman.predict", _kalman_predict)
register("kalman.state", _kalman_state_value)
register("kalman.rate", _kalman_rate)
register("kalman.reset", _kalman_reset)
register("kalman.velocity.init", _kalman_vel_init)
register("kalman.velocity.process", _kalman_vel_process)
register("kalman.velocity.predict", _kalman_vel_predict)
register("kalman.adaptive.init", _kalman_adaptive_init)
register("kalman.adaptive.process", _kalman_adaptive_process)

# ---- string format/pad (reference util) ----
register("lpad", lambda s, n, pad=" ": None if s is None else
         str(s).rjust(int(n), str(pad)[:1] or " "))
register("rpad", lambda s, n, pad=" ": None if s is None else
         str(s).ljust(int(n), str(pad)[:1] or " "))


def _java_format(fmt, *args):
    out = str(fmt)
    # %s/%d/%f subset
    for a in args:
        for tok in ("%s", "%d", "%f"):
            i = out.find(tok)
            if i >= 0:
                rep = (str(int(a)) if tok == "%d" else
                       (f"{float(a):f}" if tok == "%f" else str(a)))
                out = out[:i] + rep + out[i + 2:]
                break
    return out


register("format", _java_format)


# Neo4j round(): optional precision and mode arguments
def _round(x, precision=0, mode="HALF_UP"):
    if x is None:
        return None
    import decimal
    p = int(precision)
    modes = {"UP": decimal.ROUND_UP, "DOWN": decimal.ROUND_DOWN,
             "CEILING": decimal.ROUND_CEILING, "FLOOR": decimal.ROUND_FLOOR,
             "HALF_UP": decimal.ROUND_HALF_UP,
             "HALF_DOWN": decimal.ROUND_HALF_DOWN,
             "HALF_EVEN": decimal.ROUND_HALF_EVEN}
    m = modes.get(str(mode).upper(), decimal.ROUND_HALF_UP)
    q = decimal.Decimal(1).scaleb(-p)
    return float(decimal.Decimal(str(float(x))).quantize(q, rounding=m))


register("round", _round)


register("ltrim", lambda s, chars=None: None if s is None else
         (s.lstrip() if chars is None else s.lstrip(chars)))
register("rtrim", lambda s, chars=None: None if s is None else
         (s.rstrip() if chars is None else s.rstrip(chars)))
