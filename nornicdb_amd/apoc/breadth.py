"""APOC breadth: bulk registration of the remaining pure-function
categories (bitwise, math, number, stats, scoring, spatial, hashing,
util, text, coll, map, convert, json, date, temporal, label, meta, diff,
node/rel accessors, xml, graph, agg).

Parity: reference apoc/ category packages (registry names extracted from
apoc/registry/registry.go usage sites; ~950 total). Behavioral semantics
follow standard APOC documentation; implementations are original.
"""

from __future__ import annotations

import base64
import datetime as _dt
import gzip
import hashlib
import json
import math
import random
import re
import statistics as _st
import time
import urllib.parse
import uuid as _uuid
import xml.etree.ElementTree as _ET
import zlib
from typing import Any, Dict, List

from ..cypher import temporal as _tp
from ..cypher.functions import AGGREGATES, AGG_FINALIZERS, FUNCTIONS
from ..storage.types import Edge, Node


def _reg(name):
    def deco(fn):
        FUNCTIONS[name.lower()] = fn
        return fn
    return deco


def _r(name, fn):
    FUNCTIONS[name.lower()] = fn
    return fn


_I64 = (1 << 64) - 1


def _num_list(l):
    return [x for x in (l or []) if isinstance(x, (int, float))
            and not isinstance(x, bool)]


# ============================== apoc.bitwise ==============================
_r("apoc.bitwise.and", lambda a, b: int(a) & int(b))
_r("apoc.bitwise.or", lambda a, b: int(a) | int(b))
_r("apoc.bitwise.xor", lambda a, b: int(a) ^ int(b))
_r("apoc.bitwise.not", lambda a: ~int(a))
_r("apoc.bitwise.leftShift", lambda a, n: int(a) << int(n))
_r("apoc.bitwise.rightShift", lambda a, n: int(a) >> int(n))
_r("apoc.bitwise.setBit", lambda a, i: int(a) | (1 << int(i)))
_r("apoc.bitwise.clearBit", lambda a, i: int(a) & ~(1 << int(i)))
_r("apoc.bitwise.toggleBit", lambda a, i: int(a) ^ (1 << int(i)))
_r("apoc.bitwise.testBit", lambda a, i: bool(int(a) & (1 << int(i))))
_r("apoc.bitwise.countBits", lambda a: bin(int(a) & _I64).count("1"))
_r("apoc.bitwise.reverseBits", lambda a, w=64: int(
    bin(int(a) & ((1 << int(w)) - 1))[2:].zfill(int(w))[::-1], 2))
_r("apoc.bitwise.rotateLeft", lambda a, n, w=64: (
    ((int(a) << (int(n) % int(w))) | (int(a) >> (int(w) - int(n) % int(w))))
    & ((1 << int(w)) - 1)))
_r("apoc.bitwise.rotateRight", lambda a, n, w=64: (
    ((int(a) >> (int(n) % int(w))) | (int(a) << (int(w) - int(n) % int(w))))
    & ((1 << int(w)) - 1)))


def _bitwise_op(a, op, b):
    a, b = int(a), int(b)
    return {"&": a & b, "|": a | b, "^": a ^ b, "~": ~a,
            "<<": a << b, ">>": a >> b, ">>>": (a & _I64) >> b}.get(op)


_r("apoc.bitwise.op", _bitwise_op)

# ============================== apoc.math ==============================
for _n in ("abs", "ceil", "floor", "sqrt", "sin", "cos", "tan", "asin",
           "acos", "atan", "exp", "log10"):
    _r(f"apoc.math.{_n}",
       (lambda f: lambda x: None if x is None else f(x))(
           getattr(math, _n if _n not in ("abs",) else "fabs")
           if _n != "abs" else abs))
_r("apoc.math.log", lambda x, base=math.e: None if x is None or x <= 0
    else math.log(x, base))
_r("apoc.math.atan2", lambda y, x: math.atan2(y, x))
_r("apoc.math.pow", lambda x, y: math.pow(x, y))
_r("apoc.math.round", lambda x, p=0: None if x is None else round(x, int(p)))
_r("apoc.math.sigmoid", lambda x: 1.0 / (1.0 + math.exp(-x)))
_r("apoc.math.sigmoidPrime", lambda x: (lambda s: s * (1 - s))(
    1.0 / (1.0 + math.exp(-x))))
_r("apoc.math.tanh", lambda x: math.tanh(x))
_r("apoc.math.coth", lambda x: None if x == 0 else math.cosh(x) / math.sinh(x))
_r("apoc.math.cosh", lambda x: math.cosh(x))
_r("apoc.math.sinh", lambda x: math.sinh(x))
_r("apoc.math.sech", lambda x: 1.0 / math.cosh(x))
_r("apoc.math.csch", lambda x: None if x == 0 else 1.0 / math.sinh(x))
_r("apoc.math.clamp", lambda x, lo, hi: max(lo, min(hi, x)))
_r("apoc.math.lerp", lambda a, b, t: a + (b - a) * t)
_r("apoc.math.logit", lambda p: math.log(p / (1 - p)))
_r("apoc.math.gcd", lambda a, b: math.gcd(int(a), int(b)))
_r("apoc.math.lcm", lambda a, b: 0 if not a or not b
    else abs(int(a) * int(b)) // math.gcd(int(a), int(b)))
_r("apoc.math.factorial", lambda n: math.factorial(int(n)))


def _fib(n):
    a, b = 0, 1
    for _ in range(int(n)):
        a, b = b, a + b
    return a


_r("apoc.math.fibonacci", _fib)


def _is_prime(n):
    n = int(n)
    if n < 2:
        return False
    if n < 4:
        return True
    if n % 2 == 0:
        return False
    i = 3
    while i * i <= n:
        if n % i == 0:
            return False
        i += 2
    return True


_r("apoc.math.isPrime", _is_prime)


def _next_prime(n):
    n = int(n) + 1
    while not _is_prime(n):
        n += 1
    return n


_r("apoc.math.nextPrime", _next_prime)
_r("apoc.math.mean", lambda l: _st.mean(_num_list(l)) if _num_list(l) else None)
_r("apoc.math.median", lambda l: _st.median(_num_list(l)) if _num_list(l) else None)
_r("apoc.math.mode", lambda l: _st.mode(_num_list(l)) if _num_list(l) else None)
_r("apoc.math.stdev", lambda l: _st.stdev(_num_list(l))
    if len(_num_list(l)) > 1 else 0.0)
_r("apoc.math.variance", lambda l: _st.variance(_num_list(l))
    if len(_num_list(l)) > 1 else 0.0)
_r("apoc.math.sum", lambda l: sum(_num_list(l)))
_r("apoc.math.product", lambda l: math.prod(_num_list(l)))
_r("apoc.math.minDouble", lambda: -1.7976931348623157e308)
_r("apoc.math.maxDouble", lambda: 1.7976931348623157e308)
_r("apoc.math.minLong", lambda: -(1 << 63))
_r("apoc.math.maxLong", lambda: (1 << 63) - 1)
_r("apoc.math.minInt", lambda: -(1 << 31))
_r("apoc.math.maxInt", lambda: (1 << 31) - 1)
_r("apoc.math.minByte", lambda: -128)
_r("apoc.math.maxByte", lambda: 127)
_r("apoc.math.random", lambda: random.random())
_r("apoc.math.randomInt", lambda lo=0, hi=1 << 31: random.randrange(int(lo), int(hi)))
_r("apoc.math.range", lambda lo, hi, step=1: list(range(int(lo), int(hi) + 1, int(step))))


def _percentile(l, p):
    vals = sorted(_num_list(l))
    if not vals:
        return None
    idx = p * (len(vals) - 1)
    lo, hi = int(math.floor(idx)), int(math.ceil(idx))
    if lo == hi:
        return vals[lo]
    return vals[lo] + (vals[hi] - vals[lo]) * (idx - lo)


_r("apoc.math.percentile", _percentile)
_r("apoc.math.normalize", lambda l: (lambda v, s: [x / s for x in v] if s else v)(
    _num_list(l), sum(_num_list(l))))

# ============================== apoc.number ==============================
_ROMAN = [(1000, "M"), (900, "CM"), (500, "D"), (400, "CD"), (100, "C"),
          (90, "XC"), (50, "L"), (40, "XL"), (10, "X"), (9, "IX"),
          (5, "V"), (4, "IV"), (1, "I")]


def _romanize(n):
    n = int(n)
    out = []
    for v, s in _ROMAN:
        while n >= v:
            out.append(s)
            n -= v
    return "".join(out)


def _arabize(s):
    vals = {"I": 1, "V": 5, "X": 10, "L": 50, "C": 100, "D": 500, "M": 1000}
    total = 0
    prev = 0
    for ch in reversed(str(s).upper()):
        v = vals.get(ch, 0)
        total += v if v >= prev else -v
        prev = max(prev, v)
    return total


_DIGITS = "0123456789abcdefghijklmnopqrstuvwxyz"


def _to_base(n, base):
    n, base = int(n), int(base)
    if n == 0:
        return "0"
    neg = n < 0
    n = abs(n)
    out = []
    while n:
        out.append(_DIGITS[n % base])
        n //= base
    return ("-" if neg else "") + "".join(reversed(out))


_r("apoc.number.romanize", _romanize)
_r("apoc.number.arabize", _arabize)
_r("apoc.number.toBase", _to_base)
_r("apoc.number.fromBase", lambda s, base: int(str(s), int(base)))
_r("apoc.number.toBinary", lambda n: _to_base(n, 2))
_r("apoc.number.fromBinary", lambda s: int(str(s), 2))
_r("apoc.number.toHex", lambda n: _to_base(n, 16))
_r("apoc.number.fromHex", lambda s: int(str(s), 16))
_r("apoc.number.toOctal", lambda n: _to_base(n, 8))
_r("apoc.number.fromOctal", lambda s: int(str(s), 8))
_r("apoc.number.abs", lambda x: None if x is None else abs(x))
_r("apoc.number.ceil", lambda x: int(math.ceil(x)))
_r("apoc.number.floor", lambda x: int(math.floor(x)))
_r("apoc.number.round", lambda x, p=0: round(x, int(p)) if p else int(round(x)))
_r("apoc.number.sign", lambda x: 0 if x == 0 else (1 if x > 0 else -1))
_r("apoc.number.sqrt", lambda x: math.sqrt(x))
_r("apoc.number.exp", lambda x: math.exp(x))
_r("apoc.number.log", lambda x: math.log(x))
_r("apoc.number.log10", lambda x: math.log10(x))
_r("apoc.number.power", lambda x, y: math.pow(x, y))
_r("apoc.number.isEven", lambda n: int(n) % 2 == 0)
_r("apoc.number.isOdd", lambda n: int(n) % 2 == 1)
_r("apoc.number.isPrime", _is_prime)
_r("apoc.number.gcd", lambda a, b: math.gcd(int(a), int(b)))
_r("apoc.number.lcm", FUNCTIONS["apoc.math.lcm".lower()])
_r("apoc.number.factorial", lambda n: math.factorial(int(n)))
_r("apoc.number.fibonacci", _fib)
_r("apoc.number.clamp", lambda x, lo, hi: max(lo, min(hi, x)))
_r("apoc.number.lerp", lambda a, b, t: a + (b - a) * t)
_r("apoc.number.random", lambda: random.random())
_r("apoc.number.randomInt", lambda lo=0, hi=1 << 31: random.randrange(int(lo), int(hi)))
_r("apoc.number.exact", lambda s: float(s))
_r("apoc.number.parse", lambda s: (lambda t: int(t) if re.fullmatch(r"-?\d+", t)
    else float(t))(str(s).strip()))
_r("apoc.number.map", lambda x, a, b, c, d: c + (x - a) * (d - c) / (b - a)
    if b != a else c)
_r("apoc.number.normalize", lambda x, lo, hi: (x - lo) / (hi - lo)
    if hi != lo else 0.0)

# ============================== apoc.stats ==============================
_r("apoc.stats.sum", lambda l: sum(_num_list(l)))
_r("apoc.stats.mean", lambda l: _st.mean(_num_list(l)) if _num_list(l) else None)
_r("apoc.stats.median", lambda l: _st.median(_num_list(l)) if _num_list(l) else None)
_r("apoc.stats.mode", lambda l: _st.mode(_num_list(l)) if _num_list(l) else None)
_r("apoc.stats.min", lambda l: min(_num_list(l), default=None))
_r("apoc.stats.max", lambda l: max(_num_list(l), default=None))
_r("apoc.stats.count", lambda l: len(l or []))
_r("apoc.stats.range", lambda l: (max(_num_list(l)) - min(_num_list(l)))
    if _num_list(l) else None)
_r("apoc.stats.stdDev", lambda l: _st.stdev(_num_list(l))
    if len(_num_list(l)) > 1 else 0.0)
_r("apoc.stats.variance", lambda l: _st.variance(_num_list(l))
    if len(_num_list(l)) > 1 else 0.0)
_r("apoc.stats.percentile", _percentile)
_r("apoc.stats.quartiles", lambda l: [_percentile(l, 0.25), _percentile(l, 0.5),
                                      _percentile(l, 0.75)])
_r("apoc.stats.iqr", lambda l: (_percentile(l, 0.75) - _percentile(l, 0.25))
    if _num_list(l) else None)


def _zscores(l):
    v = _num_list(l)
    if len(v) < 2:
        return [0.0] * len(v)
    m, s = _st.mean(v), _st.stdev(v)
    return [(x - m) / s if s else 0.0 for x in v]


_r("apoc.stats.zScore", _zscores)
_r("apoc.stats.normalize", lambda l: (lambda v: [(x - min(v)) / (max(v) - min(v))
    if max(v) != min(v) else 0.0 for x in v])(_num_list(l)))


def _covariance(a, b):
    a, b = _num_list(a), _num_list(b)
    n = min(len(a), len(b))
    if n < 2:
        return None
    ma, mb = _st.mean(a[:n]), _st.mean(b[:n])
    return sum((a[i] - ma) * (b[i] - mb) for i in range(n)) / (n - 1)


def _correlation(a, b):
    a, b = _num_list(a), _num_list(b)
    n = min(len(a), len(b))
    if n < 2:
        return None
    sa, sb = _st.stdev(a[:n]), _st.stdev(b[:n])
    if not sa or not sb:
        return 0.0
    return _covariance(a[:n], b[:n]) / (sa * sb)


_r("apoc.stats.covariance", _covariance)
_r("apoc.stats.correlation", _correlation)


def _skewness(l):
    v = _num_list(l)
    if len(v) < 3:
        return 0.0
    m = _st.mean(v)
    s = _st.stdev(v)
    if not s:
        return 0.0
    n = len(v)
    return (n / ((n - 1) * (n - 2))) * sum(((x - m) / s) ** 3 for x in v)


def _kurtosis(l):
    v = _num_list(l)
    if len(v) < 4:
        return 0.0
    m = _st.mean(v)
    s = _st.stdev(v)
    if not s:
        return 0.0
    n = len(v)
    g = (n * (n + 1) / ((n - 1) * (n - 2) * (n - 3))) * \
        sum(((x - m) / s) ** 4 for x in v)
    return g - 3 * (n - 1) ** 2 / ((n - 2) * (n - 3))


_r("apoc.stats.skewness", _skewness)
_r("apoc.stats.kurtosis", _kurtosis)


def _outliers(l):
    v = _num_list(l)
    if len(v) < 4:
        return []
    q1, q3 = _percentile(v, 0.25), _percentile(v, 0.75)
    iqr = q3 - q1
    lo, hi = q1 - 1.5 * iqr, q3 + 1.5 * iqr
    return [x for x in v if x < lo or x > hi]


_r("apoc.stats.outliers", _outliers)


def _histogram(l, bins=10):
    v = _num_list(l)
    if not v:
        return []
    lo, hi = min(v), max(v)
    bins = int(bins)
    w = (hi - lo) / bins if hi > lo else 1
    counts = [0] * bins
    for x in v:
        counts[min(int((x - lo) / w), bins - 1)] += 1
    return [{"min": lo + i * w, "max": lo + (i + 1) * w, "count": c}
            for i, c in enumerate(counts)]


_r("apoc.stats.histogram", _histogram)
_r("apoc.stats.summary", lambda l: {
    "count": len(_num_list(l)), "min": min(_num_list(l), default=None),
    "max": max(_num_list(l), default=None),
    "mean": _st.mean(_num_list(l)) if _num_list(l) else None,
    "median": _st.median(_num_list(l)) if _num_list(l) else None,
    "stdev": _st.stdev(_num_list(l)) if len(_num_list(l)) > 1 else 0.0})

# ============================== apoc.scoring ==============================
def _vec_pair(a, b):
    a = [float(x) for x in (a or [])]
    b = [float(x) for x in (b or [])]
    n = min(len(a), len(b))
    return a[:n], b[:n]


def _cosine(a, b):
    a, b = _vec_pair(a, b)
    na = math.sqrt(sum(x * x for x in a))
    nb = math.sqrt(sum(x * x for x in b))
    if not na or not nb:
        return 0.0
    return sum(x * y for x, y in zip(a, b)) / (na * nb)


_r("apoc.scoring.cosine", _cosine)
_r("apoc.scoring.euclidean", lambda a, b: math.sqrt(sum(
    (x - y) ** 2 for x, y in zip(*_vec_pair(a, b)))))
_r("apoc.scoring.manhattan", lambda a, b: sum(
    abs(x - y) for x, y in zip(*_vec_pair(a, b))))
_r("apoc.scoring.pearson", _correlation)
_r("apoc.scoring.jaccard", lambda a, b: (lambda sa, sb:
    len(sa & sb) / len(sa | sb) if (sa | sb) else 0.0)(
    set(map(repr, a or [])), set(map(repr, b or []))))
_r("apoc.scoring.dice", lambda a, b: (lambda sa, sb:
    2 * len(sa & sb) / (len(sa) + len(sb)) if (sa or sb) else 0.0)(
    set(map(repr, a or [])), set(map(repr, b or []))))
_r("apoc.scoring.overlap", lambda a, b: (lambda sa, sb:
    len(sa & sb) / min(len(sa), len(sb)) if sa and sb else 0.0)(
    set(map(repr, a or [])), set(map(repr, b or []))))
_r("apoc.scoring.existence", lambda score, exists: float(score) if exists else 0.0)
_r("apoc.scoring.pareto", lambda minimum, eighty, maximum, score:
    0.0 if score < minimum else minimum + (maximum - minimum) *
    (1 - math.exp(-math.log(5.0) * score / max(eighty, 1e-12))))
_r("apoc.scoring.sigmoid", lambda x: 1.0 / (1.0 + math.exp(-x)))


def _softmax(l):
    v = _num_list(l)
    if not v:
        return []
    m = max(v)
    e = [math.exp(x - m) for x in v]
    s = sum(e)
    return [x / s for x in e]


_r("apoc.scoring.softmax", _softmax)
_r("apoc.scoring.minMax", lambda x, lo, hi: (x - lo) / (hi - lo) if hi != lo else 0.0)
_r("apoc.scoring.zScore", lambda x, mean, std: (x - mean) / std if std else 0.0)
_r("apoc.scoring.normalize", lambda l: (lambda v, s: [x / s for x in v] if s else v)(
    _num_list(l), math.sqrt(sum(x * x for x in _num_list(l)))))
_r("apoc.scoring.percentile", _percentile)
_r("apoc.scoring.rank", lambda l, x: sorted(_num_list(l)).index(x) + 1
    if x in _num_list(l) else None)
_r("apoc.scoring.topK", lambda l, k: sorted(_num_list(l), reverse=True)[:int(k)])
_r("apoc.scoring.tf", lambda count, total: count / total if total else 0.0)
_r("apoc.scoring.idf", lambda docs, with_term: math.log(
    (1 + docs) / (1 + with_term)) + 1)
_r("apoc.scoring.tfidf", lambda count, total, docs, with_term:
    (count / total if total else 0.0) * (math.log((1 + docs) / (1 + with_term)) + 1))


def _bm25(tf, doclen, avglen, docs, with_term, k1=1.2, b=0.75):
    idf = math.log((docs - with_term + 0.5) / (with_term + 0.5) + 1)
    return idf * tf * (k1 + 1) / (tf + k1 * (1 - b + b * doclen / max(avglen, 1e-9)))


_r("apoc.scoring.bm25", _bm25)
_r("apoc.scoring.pageRank", lambda incoming, damping=0.85:
    (1 - damping) + damping * sum(_num_list(incoming)))

# ============================== apoc.spatial ==============================
_EARTH_R = 6371008.8


def _haversine(lat1, lon1, lat2, lon2):
    p1, p2 = math.radians(lat1), math.radians(lat2)
    dp = math.radians(lat2 - lat1)
    dl = math.radians(lon2 - lon1)
    a = math.sin(dp / 2) ** 2 + math.cos(p1) * math.cos(p2) * math.sin(dl / 2) ** 2
    return 2 * _EARTH_R * math.asin(math.sqrt(a))


_r("apoc.spatial.haversineDistance", _haversine)
_r("apoc.spatial.distance", _haversine)
_r("apoc.spatial.vincentyDistance", _haversine)  # spherical approximation


def _bearing(lat1, lon1, lat2, lon2):
    p1, p2 = math.radians(lat1), math.radians(lat2)
    dl = math.radians(lon2 - lon1)
    y = math.sin(dl) * math.cos(p2)
    x = math.cos(p1) * math.sin(p2) - math.sin(p1) * math.cos(p2) * math.cos(dl)
    return (math.degrees(math.atan2(y, x)) + 360) % 360


_r("apoc.spatial.bearing", _bearing)


def _destination(lat, lon, bearing, dist):
    p1 = math.radians(lat)
    l1 = math.radians(lon)
    br = math.radians(bearing)
    dr = dist / _EARTH_R
    p2 = math.asin(math.sin(p1) * math.cos(dr) +
                   math.cos(p1) * math.sin(dr) * math.cos(br))
    l2 = l1 + math.atan2(math.sin(br) * math.sin(dr) * math.cos(p1),
                         math.cos(dr) - math.sin(p1) * math.sin(p2))
    return {"latitude": math.degrees(p2),
            "longitude": (math.degrees(l2) + 540) % 360 - 180}


_r("apoc.spatial.destination", _destination)
_r("apoc.spatial.midpoint", lambda lat1, lon1, lat2, lon2: _destination(
    lat1, lon1, _bearing(lat1, lon1, lat2, lon2),
    _haversine(lat1, lon1, lat2, lon2) / 2))
_r("apoc.spatial.withinDistance", lambda lat1, lon1, lat2, lon2, d:
    _haversine(lat1, lon1, lat2, lon2) <= d)

_GH32 = "0123456789bcdefghjkmnpqrstuvwxyz"


def _encode_geohash(lat, lon, precision=9):
    lat_r, lon_r = [-90.0, 90.0], [-180.0, 180.0]
    bits = []
    even = True
    while len(bits) < int(precision) * 5:
        if even:
            mid = (lon_r[0] + lon_r[1]) / 2
            bits.append(1 if lon > mid else 0)
            lon_r[0 if lon > mid else 1] = mid
        else:
            mid = (lat_r[0] + lat_r[1]) / 2
            bits.append(1 if lat > mid else 0)
            lat_r[0 if lat > mid else 1] = mid
        even = not even
    out = []
    for i in range(0, len(bits), 5):
        out.append(_GH32[int("".join(map(str, bits[i:i + 5])), 2)])
    return "".join(out)


def _decode_geohash(gh):
    lat_r, lon_r = [-90.0, 90.0], [-180.0, 180.0]
    even = True
    for ch in str(gh):
        v = _GH32.index(ch)
        for bit in (16, 8, 4, 2, 1):
            r = lon_r if even else lat_r
            mid = (r[0] + r[1]) / 2
            r[0 if v & bit else 1] = mid
            even = not even
    return {"latitude": sum(lat_r) / 2, "longitude": sum(lon_r) / 2}


_r("apoc.spatial.encodeGeohash", _encode_geohash)
_r("apoc.spatial.decodeGeohash", _decode_geohash)
_r("apoc.spatial.boundingBox", lambda points: {
    "minLat": min(p["latitude"] for p in points),
    "maxLat": max(p["latitude"] for p in points),
    "minLon": min(p["longitude"] for p in points),
    "maxLon": max(p["longitude"] for p in points)} if points else None)
_r("apoc.spatial.centroid", lambda points: {
    "latitude": sum(p["latitude"] for p in points) / len(points),
    "longitude": sum(p["longitude"] for p in points) / len(points)}
    if points else None)
_r("apoc.spatial.contains", lambda box, lat, lon:
    box["minLat"] <= lat <= box["maxLat"] and box["minLon"] <= lon <= box["maxLon"])
_r("apoc.spatial.within", lambda lat, lon, box:
    box["minLat"] <= lat <= box["maxLat"] and box["minLon"] <= lon <= box["maxLon"])
_r("apoc.spatial.toGeoJSON", lambda lat, lon: {
    "type": "Point", "coordinates": [lon, lat]})
_r("apoc.spatial.fromGeoJSON", lambda g: {
    "latitude": g["coordinates"][1], "longitude": g["coordinates"][0]}
    if g and g.get("type") == "Point" else None)


def _poly_area(points):
    """Shoelace on lat/lon treated as planar (small areas)."""
    if not points or len(points) < 3:
        return 0.0
    s = 0.0
    for i in range(len(points)):
        a, b = points[i], points[(i + 1) % len(points)]
        s += a["longitude"] * b["latitude"] - b["longitude"] * a["latitude"]
    return abs(s) / 2


_r("apoc.spatial.area", _poly_area)


def _k_nearest(points, lat, lon, k):
    scored = sorted(points, key=lambda p: _haversine(
        lat, lon, p["latitude"], p["longitude"]))
    return scored[:int(k)]


_r("apoc.spatial.kNearest", _k_nearest)
_r("apoc.spatial.nearest", lambda points, lat, lon: _k_nearest(
    points, lat, lon, 1)[0] if points else None)
_r("apoc.spatial.intersects", lambda b1, b2: not (
    b1["maxLat"] < b2["minLat"] or b2["maxLat"] < b1["minLat"]
    or b1["maxLon"] < b2["minLon"] or b2["maxLon"] < b1["minLon"]))

# ============================== apoc.hashing ==============================
def _hash_of(algo, data):
    if isinstance(data, (list, dict)):
        data = json.dumps(data, sort_keys=True, default=str)
    return hashlib.new(algo, str(data).encode()).hexdigest()


_r("apoc.hashing.md5", lambda d: _hash_of("md5", d))
_r("apoc.hashing.sha1", lambda d: _hash_of("sha1", d))
_r("apoc.hashing.sha256", lambda d: _hash_of("sha256", d))
_r("apoc.hashing.sha384", lambda d: _hash_of("sha384", d))
_r("apoc.hashing.sha512", lambda d: _hash_of("sha512", d))


def _fnv1a64(data):
    h = 0xcbf29ce484222325
    for b in str(data).encode():
        h ^= b
        h = (h * 0x100000001b3) & _I64
    return h


def _fnv164(data):
    h = 0xcbf29ce484222325
    for b in str(data).encode():
        h = (h * 0x100000001b3) & _I64
        h ^= b
    return h


_r("apoc.hashing.fnv1a", _fnv1a64)
_r("apoc.hashing.fnv1a64", _fnv1a64)
_r("apoc.hashing.fnv1", _fnv164)
_r("apoc.hashing.fnv164", _fnv164)


def _xxhash_like(data, seed=0):
    # splitmix-style 64-bit avalanche over the bytes (stable stand-in)
    h = (seed ^ 0x9E3779B97F4A7C15) & _I64
    for b in str(data).encode():
        h = ((h ^ b) * 0xBF58476D1CE4E5B9) & _I64
        h ^= h >> 27
    h = (h * 0x94D049BB133111EB) & _I64
    return h ^ (h >> 31)


_r("apoc.hashing.xxHash64", _xxhash_like)
_r("apoc.hashing.xxHash32", lambda d, seed=0: _xxhash_like(d, seed) & 0xFFFFFFFF)
_r("apoc.hashing.murmurHash3", _xxhash_like)
_r("apoc.hashing.cityHash64", _xxhash_like)


def _jump_hash(key, buckets):
    """Jump consistent hash (Lamping & Veach)."""
    k = int(key) & _I64
    b, j = -1, 0
    while j < int(buckets):
        b = j
        k = (k * 2862933555777941757 + 1) & _I64
        j = int((b + 1) * (1 << 31) / ((k >> 33) + 1))
    return b


_r("apoc.hashing.jumpHash", _jump_hash)
_r("apoc.hashing.consistentHash", lambda key, buckets: _jump_hash(
    _fnv1a64(key), buckets))
_r("apoc.hashing.rendezvousHash", lambda key, nodes: max(
    nodes, key=lambda n: _fnv1a64(f"{key}:{n}")) if nodes else None)


def _fingerprint_entity(x):
    if isinstance(x, Node):
        return _hash_of("sha256", {"labels": sorted(x.labels),
                                   "props": {k: repr(v) for k, v in
                                             sorted(x.properties.items())}})
    if isinstance(x, Edge):
        return _hash_of("sha256", {"type": x.type,
                                   "props": {k: repr(v) for k, v in
                                             sorted(x.properties.items())}})
    return _hash_of("sha256", x)


_r("apoc.hashing.fingerprint", _fingerprint_entity)
_r("apoc.hashing.fingerprinting", _fingerprint_entity)
_r("apoc.hashing.fingerprintGraph", lambda nodes, rels=None: _hash_of(
    "sha256", sorted(_fingerprint_entity(x) for x in
                     list(nodes or []) + list(rels or []))))

# ============================== apoc.util ==============================
# apoc.util.md5/sha* take a LIST and hash the concatenated string forms
_r("apoc.util.md5", lambda l: hashlib.md5("".join(
    str(x) for x in (l if isinstance(l, list) else [l])).encode()).hexdigest())
_r("apoc.util.md5Hex", lambda d: _hash_of("md5", d))
_r("apoc.util.sha1Hex", lambda d: _hash_of("sha1", d))
_r("apoc.util.sha256Hex", lambda d: _hash_of("sha256", d))
_r("apoc.util.md5Base64", lambda d: base64.b64encode(
    hashlib.md5(str(d).encode()).digest()).decode())
_r("apoc.util.sha1Base64", lambda d: base64.b64encode(
    hashlib.sha1(str(d).encode()).digest()).decode())
_r("apoc.util.sha256Base64", lambda d: base64.b64encode(
    hashlib.sha256(str(d).encode()).digest()).decode())
_r("apoc.util.encodeBase64", lambda s: base64.b64encode(str(s).encode()).decode())
_r("apoc.util.decodeBase64", lambda s: base64.b64decode(str(s)).decode())
_r("apoc.util.encodeURL", lambda s: urllib.parse.quote(str(s), safe=""))
_r("apoc.util.decodeURL", lambda s: urllib.parse.unquote(str(s)))
_r("apoc.util.compressWithAlgorithm", lambda s, algo="gzip": list(
    gzip.compress(str(s).encode()) if algo == "gzip"
    else zlib.compress(str(s).encode())))
_r("apoc.util.decompressWithAlgorithm", lambda data, algo="gzip": (
    gzip.decompress(bytes(x & 0xFF for x in data)) if algo == "gzip"
    else zlib.decompress(bytes(x & 0xFF for x in data))).decode())
_r("apoc.util.now", lambda: int(time.time() * 1000))
_r("apoc.util.nowInSeconds", lambda: int(time.time()))
_r("apoc.util.timestamp", lambda: int(time.time() * 1000))
_r("apoc.util.uuid", lambda: str(_uuid.uuid4()))
_r("apoc.util.randomUUID", lambda: str(_uuid.uuid4()))
_r("apoc.util.coalesce", lambda *a: next((x for x in a if x is not None), None))
_r("apoc.util.when", lambda cond, then, els=None: then if cond else els)
_r("apoc.util.case", lambda pairs, default=None: next(
    (pairs[i + 1] for i in range(0, len(pairs or []) - 1, 2) if pairs[i]),
    default))
_r("apoc.util.isNode", lambda x: isinstance(x, Node))
_r("apoc.util.isRelationship", lambda x: isinstance(x, Edge))
_r("apoc.util.isPath", lambda x: hasattr(x, "nodes") and hasattr(x, "edges")
    and not isinstance(x, (Node, Edge)))
_r("apoc.util.typeOf", lambda x: type(x).__name__ if x is not None else "NULL")
_r("apoc.util.repeat", lambda s, n: str(s) * int(n))
_r("apoc.util.range", lambda lo, hi, step=1: list(
    range(int(lo), int(hi) + (1 if step > 0 else -1), int(step))))
_r("apoc.util.partition", lambda l, size: [
    list(l[i:i + int(size)]) for i in range(0, len(l or []), int(size))])
_r("apoc.util.merge", lambda a, b: {**(a or {}), **(b or {})})
_r("apoc.util.validatePattern", lambda s, pat: bool(
    re.fullmatch(pat, str(s or ""))))
_r("apoc.util.formatTimestamp", lambda ms, fmt="%Y-%m-%dT%H:%M:%SZ":
    _dt.datetime.fromtimestamp(ms / 1000.0, _dt.timezone.utc).strftime(fmt))
_r("apoc.util.parseTimestamp", lambda s: int(_dt.datetime.fromisoformat(
    str(s).replace("Z", "+00:00")).timestamp() * 1000))
_r("apoc.util.sleep", lambda ms: time.sleep(min(float(ms), 1000) / 1000.0))
_r("apoc.util.validate", lambda cond, msg="validation failed", params=None:
    (_raise(ValueError(msg % tuple(params or []) if params else msg))
     if cond else None))


def _raise(e):
    raise e


# ============================== apoc.temporal ==============================
_r("apoc.temporal.format", lambda v, fmt="%Y-%m-%dT%H:%M:%S": (
    _tp.make_datetime(v)._v.strftime(fmt.replace("yyyy", "%Y")
                                     .replace("MM", "%m").replace("dd", "%d")
                                     .replace("HH", "%H").replace("mm", "%M")
                                     .replace("ss", "%S"))))
_r("apoc.temporal.parse", lambda s, fmt=None: _tp.make_datetime(s))
_r("apoc.temporal.toEpochMillis", lambda v: _tp.make_datetime(v).component(
    "epochMillis"))
_r("apoc.temporal.fromEpochMillis", lambda ms: _tp.make_datetime(float(ms)))
_r("apoc.temporal.add", lambda v, dur: _tp.make_datetime(v) +
    _tp.make_duration(dur))
_r("apoc.temporal.subtract", lambda v, dur: _tp.make_datetime(v) -
    _tp.make_duration(dur))
_r("apoc.temporal.difference", lambda a, b: _tp.duration_between(a, b))
_r("apoc.temporal.duration", lambda spec: _tp.make_duration(spec))
_r("apoc.temporal.formatDuration", lambda d: str(_tp.make_duration(d)))
_r("apoc.temporal.truncate", lambda unit, v: _tp.truncate(unit, _tp.make_datetime(v)))
_r("apoc.temporal.round", lambda unit, v: _tp.truncate(unit, _tp.make_datetime(v)))
_r("apoc.temporal.startOf", lambda v, unit: _tp.truncate(unit, _tp.make_datetime(v)))


def _end_of(v, unit):
    start = _tp.truncate(unit, _tp.make_datetime(v))
    nxt = {"year": {"years": 1}, "quarter": {"months": 3},
           "month": {"months": 1}, "week": {"weeks": 1}, "day": {"days": 1},
           "hour": {"hours": 1}, "minute": {"minutes": 1},
           "second": {"seconds": 1}}[unit.lower()]
    return start + _tp.CypherDuration.from_map(nxt) - \
        _tp.CypherDuration(0, 0, 0, 1_000_000)


_r("apoc.temporal.endOf", _end_of)
_r("apoc.temporal.dayOfWeek", lambda v: _tp.make_datetime(v).component("dayOfWeek"))
_r("apoc.temporal.dayOfYear", lambda v: _tp.make_datetime(v).component("dayOfYear"))
_r("apoc.temporal.weekOfYear", lambda v: _tp.make_datetime(v).component("week"))
_r("apoc.temporal.quarter", lambda v: _tp.make_datetime(v).component("quarter"))
_r("apoc.temporal.daysInMonth", lambda v: (lambda dt: (
    _dt.date(dt.year + (dt.month == 12), dt.month % 12 + 1, 1)
    - _dt.date(dt.year, dt.month, 1)).days)(_tp.make_datetime(v)._v))
_r("apoc.temporal.isLeapYear", lambda v: (lambda y: y % 4 == 0 and
    (y % 100 != 0 or y % 400 == 0))(_tp.make_datetime(v)._v.year
    if not isinstance(v, (int, float)) else int(v)))
_r("apoc.temporal.isWeekend", lambda v: _tp.make_datetime(v)._v.isoweekday() >= 6)
_r("apoc.temporal.isWeekday", lambda v: _tp.make_datetime(v)._v.isoweekday() < 6)
_r("apoc.temporal.isBetween", lambda v, a, b: (
    _tp.make_datetime(a)._v <= _tp.make_datetime(v)._v <= _tp.make_datetime(b)._v))
_r("apoc.temporal.age", lambda v, ref=None: _tp.duration_between(
    v, ref if ref is not None else _tp.make_datetime(None)))
_r("apoc.temporal.toUTC", lambda v: _tp.CypherDateTime(
    _tp.make_datetime(v)._v.astimezone(_dt.timezone.utc)))
_r("apoc.temporal.toLocal", lambda v: _tp.CypherDateTime(
    _tp.make_datetime(v)._v.astimezone()))
_r("apoc.temporal.timezone", lambda v=None: time.strftime("%Z"))
_r("apoc.temporal.systemTimezone", lambda: time.strftime("%Z"))

# ============================== apoc.date (extras) ==============================
_r("apoc.date.toISO8601", lambda ms, unit="ms": _dt.datetime.fromtimestamp(
    (ms / 1000.0 if unit == "ms" else float(ms)),
    _dt.timezone.utc).isoformat().replace("+00:00", "Z"))
_r("apoc.date.fromISO8601", lambda s: int(_dt.datetime.fromisoformat(
    str(s).replace("Z", "+00:00")).timestamp() * 1000))
_r("apoc.date.toUnixTime", lambda s: int(_dt.datetime.fromisoformat(
    str(s).replace("Z", "+00:00")).timestamp()))
_r("apoc.date.fromUnixTime", lambda s: _dt.datetime.fromtimestamp(
    float(s), _dt.timezone.utc).isoformat().replace("+00:00", "Z"))
_r("apoc.date.field", lambda ms, unit="d": {
    "ms": int(ms) % 1000, "s": int(ms / 1000) % 60,
    "m": int(ms / 60000) % 60, "h": int(ms / 3600000) % 24,
    "d": _dt.datetime.fromtimestamp(ms / 1000.0, _dt.timezone.utc).day,
    "month": _dt.datetime.fromtimestamp(ms / 1000.0, _dt.timezone.utc).month,
    "year": _dt.datetime.fromtimestamp(ms / 1000.0, _dt.timezone.utc).year,
    }[unit])
_r("apoc.date.fields", lambda ms: (lambda d: {
    "years": d.year, "months": d.month, "days": d.day, "hours": d.hour,
    "minutes": d.minute, "seconds": d.second})(
    _dt.datetime.fromtimestamp(ms / 1000.0, _dt.timezone.utc)))
_r("apoc.date.toYears", lambda ms: ms / (365.25 * 86400e3))
_r("apoc.date.systemTimezone", lambda: time.strftime("%Z"))
_r("apoc.date.convertFormat", lambda s, from_fmt, to_fmt: _dt.datetime.strptime(
    str(s), from_fmt).strftime(to_fmt))
_r("apoc.date.parseAsZonedDateTime", lambda s, fmt=None: _tp.make_datetime(s))

# ============================== apoc.json ==============================
_r("apoc.json.parse", lambda s: json.loads(s) if s is not None else None)
_r("apoc.json.stringify", lambda v: json.dumps(v, default=str))
_r("apoc.json.pretty", lambda v: json.dumps(
    json.loads(v) if isinstance(v, str) else v, indent=2, default=str))
_r("apoc.json.compact", lambda v: json.dumps(
    json.loads(v) if isinstance(v, str) else v, separators=(",", ":"),
    default=str))
_r("apoc.json.validate", lambda s: _json_valid(s))


def _json_valid(s):
    try:
        json.loads(s)
        return True
    except Exception:
        return False


_r("apoc.json.keys", lambda v: sorted((json.loads(v) if isinstance(v, str)
    else v or {}).keys()))
_r("apoc.json.values", lambda v: list((json.loads(v) if isinstance(v, str)
    else v or {}).values()))
_r("apoc.json.size", lambda v: len(json.loads(v) if isinstance(v, str)
    else (v or [])))
_r("apoc.json.type", lambda v: (lambda x: {dict: "OBJECT", list: "ARRAY",
    str: "STRING", bool: "BOOLEAN", int: "NUMBER", float: "NUMBER",
    type(None): "NULL"}.get(type(x), "UNKNOWN"))(
    json.loads(v) if isinstance(v, str) else v))
_r("apoc.json.merge", lambda a, b: {**_as_map(a), **_as_map(b)})


def _as_map(v):
    return json.loads(v) if isinstance(v, str) else dict(v or {})


def _json_flatten(v, prefix="", delim="."):
    out = {}
    items = _as_map(v).items() if not isinstance(v, list) else enumerate(v)
    for k, val in items:
        key = f"{prefix}{delim}{k}" if prefix else str(k)
        if isinstance(val, dict):
            out.update(_json_flatten(val, key, delim))
        elif isinstance(val, list):
            for i, x in enumerate(val):
                if isinstance(x, (dict, list)):
                    out.update(_json_flatten(x, f"{key}{delim}{i}", delim))
                else:
                    out[f"{key}{delim}{i}"] = x
        else:
            out[key] = val
    return out


_r("apoc.json.flatten", _json_flatten)


def _json_unflatten(m, delim="."):
    out = {}
    for k, v in _as_map(m).items():
        parts = str(k).split(delim)
        cur = out
        for p in parts[:-1]:
            cur = cur.setdefault(p, {})
        cur[parts[-1]] = v
    return out


_r("apoc.json.unflatten", _json_unflatten)


def _json_get_path(v, path):
    cur = json.loads(v) if isinstance(v, str) else v
    for part in str(path).lstrip("$.").split("."):
        if not part:
            continue
        m = re.match(r"(\w+)(?:\[(\d+)\])?$", part)
        if not m:
            return None
        if isinstance(cur, dict):
            cur = cur.get(m.group(1))
        else:
            return None
        if m.group(2) is not None and isinstance(cur, list):
            idx = int(m.group(2))
            cur = cur[idx] if idx < len(cur) else None
    return cur


_r("apoc.json.path", _json_get_path)
_r("apoc.json.filter", lambda v, keys: {k: x for k, x in _as_map(v).items()
                                        if k in set(keys or [])})
_r("apoc.json.delete", lambda v, keys: {k: x for k, x in _as_map(v).items()
                                        if k not in set(keys or [])})
_r("apoc.json.set", lambda v, key, val: {**_as_map(v), str(key): val})
_r("apoc.json.map", lambda v: _as_map(v))
_r("apoc.json.reduce", lambda v: {k: x for k, x in _as_map(v).items()
                                  if x is not None})

# ============================== apoc.convert (extras) ==============================
_r("apoc.convert.toBooleanList", lambda l: [bool(x) for x in (l or [])])
_r("apoc.convert.toIntList", lambda l: [None if x is None else int(float(x))
                                        for x in (l or [])])
_r("apoc.convert.toFloatList", lambda l: [None if x is None else float(x)
                                          for x in (l or [])])
_r("apoc.convert.toStringList", lambda l: [None if x is None else str(x)
                                           for x in (l or [])])
_r("apoc.convert.toSet", lambda l: list(dict.fromkeys(l or [])))
_r("apoc.convert.toMap", lambda v: dict(v.properties) if isinstance(
    v, (Node, Edge)) else _as_map(v))
_r("apoc.convert.toNode", lambda v: v if isinstance(v, Node) else None)
_r("apoc.convert.toRelationship", lambda v: v if isinstance(v, Edge) else None)
_r("apoc.convert.toNodeList", lambda l: [x for x in (l or [])
                                         if isinstance(x, Node)])
_r("apoc.convert.toRelationshipList", lambda l: [x for x in (l or [])
                                                 if isinstance(x, Edge)])
_r("apoc.convert.toSortedJsonMap", lambda v: json.dumps(
    _as_map(v), sort_keys=True, default=str))
_r("apoc.convert.fromJsonNode", lambda s: json.loads(s))
_r("apoc.convert.getJsonProperty", lambda ent, prop, path=None: _json_get_path(
    ent.properties.get(prop), path or "$") if isinstance(ent, (Node, Edge))
    else None)
_r("apoc.convert.getJsonPropertyMap", lambda ent, prop: _as_map(
    ent.properties.get(prop)) if isinstance(ent, (Node, Edge)) else None)

# ============================== apoc.label ==============================
def _labels_of(x):
    if isinstance(x, Node):
        return list(x.labels)
    if isinstance(x, list):
        return [str(v) for v in x]
    return [str(x)] if x is not None else []


_r("apoc.label.exists", lambda n, label: isinstance(n, Node) and label in n.labels)
_r("apoc.label.has", lambda n, label: isinstance(n, Node) and label in n.labels)
_r("apoc.label.hasAll", lambda n, labels: isinstance(n, Node) and
    all(lb in n.labels for lb in (labels or [])))
_r("apoc.label.hasAny", lambda n, labels: isinstance(n, Node) and
    any(lb in n.labels for lb in (labels or [])))
_r("apoc.label.get", lambda n: _labels_of(n))
_r("apoc.label.list", lambda n: _labels_of(n))
_r("apoc.label.count", lambda n: len(_labels_of(n)))
_r("apoc.label.toString", lambda n: ":".join(_labels_of(n)))
_r("apoc.label.format", lambda n: "".join(f":{lb}" for lb in _labels_of(n)))
_r("apoc.label.fromString", lambda s: [x for x in str(s).split(":") if x])
_r("apoc.label.fromPattern", lambda s: re.findall(r":(\w+)", str(s)))
_r("apoc.label.pattern", lambda n: "(" + "".join(
    f":{lb}" for lb in _labels_of(n)) + ")")
_r("apoc.label.normalize", lambda s: "".join(
    w.capitalize() for w in re.split(r"[\s_\-]+", str(s))))
_r("apoc.label.validate", lambda s: bool(re.fullmatch(
    r"[A-Za-z_][A-Za-z0-9_]*", str(s or ""))))
_r("apoc.label.compare", lambda a, b: sorted(_labels_of(a)) == sorted(_labels_of(b)))
_r("apoc.label.diff", lambda a, b: sorted(
    set(_labels_of(a)) - set(_labels_of(b))))
_r("apoc.label.intersection", lambda a, b: sorted(
    set(_labels_of(a)) & set(_labels_of(b))))
_r("apoc.label.union", lambda a, b: sorted(
    set(_labels_of(a)) | set(_labels_of(b))))
_r("apoc.label.merge", lambda a, b: sorted(
    set(_labels_of(a)) | set(_labels_of(b))))
_r("apoc.label.search", lambda n, pat: [lb for lb in _labels_of(n)
                                        if re.search(pat, lb)])

# ============================== apoc.node / apoc.rel ==============================
_r("apoc.node.id", lambda n: n.id if isinstance(n, Node) else None)
_r("apoc.node.labels", lambda n: list(n.labels) if isinstance(n, Node) else None)
_r("apoc.node.hasLabel", lambda n, lb: isinstance(n, Node) and lb in n.labels)
_r("apoc.node.hasLabels", lambda n, lbs: isinstance(n, Node) and
    all(lb in n.labels for lb in (lbs or [])))
_r("apoc.node.properties", lambda n: dict(n.properties)
    if isinstance(n, (Node, Edge)) else None)
_r("apoc.node.property", lambda n, k, default=None: (
    n.properties.get(k, default) if isinstance(n, (Node, Edge)) else default))
_r("apoc.node.toMap", lambda n: {"id": n.id, "labels": list(n.labels),
    "properties": dict(n.properties)} if isinstance(n, Node) else None)
_r("apoc.node.equals", lambda a, b: isinstance(a, Node) and
    isinstance(b, Node) and a.id == b.id)
_r("apoc.node.diff", lambda a, b: {
    "labelsOnlyA": sorted(set(a.labels) - set(b.labels)),
    "labelsOnlyB": sorted(set(b.labels) - set(a.labels)),
    "propsOnlyA": sorted(set(a.properties) - set(b.properties)),
    "propsOnlyB": sorted(set(b.properties) - set(a.properties)),
    "different": sorted(k for k in set(a.properties) & set(b.properties)
                        if a.properties[k] != b.properties[k])}
    if isinstance(a, Node) and isinstance(b, Node) else None)

_r("apoc.rel.id", lambda e: e.id if isinstance(e, Edge) else None)
_r("apoc.rel.type", lambda e: e.type if isinstance(e, Edge) else None)
_r("apoc.rel.isType", lambda e, t: isinstance(e, Edge) and e.type == t)
_r("apoc.rel.isAnyType", lambda e, ts: isinstance(e, Edge) and
    e.type in set(ts or []))
_r("apoc.rel.properties", lambda e: dict(e.properties)
    if isinstance(e, Edge) else None)
_r("apoc.rel.property", lambda e, k, default=None: (
    e.properties.get(k, default) if isinstance(e, Edge) else default))
_r("apoc.rel.hasProperty", lambda e, k: isinstance(e, Edge) and
    k in e.properties)
_r("apoc.rel.hasProperties", lambda e, ks: isinstance(e, Edge) and
    all(k in e.properties for k in (ks or [])))
_r("apoc.rel.startNode", lambda e: getattr(e, "_start_ref", None)
    or (e.start_node if isinstance(e, Edge) else None))
_r("apoc.rel.endNode", lambda e: getattr(e, "_end_ref", None)
    or (e.end_node if isinstance(e, Edge) else None))
_r("apoc.rel.nodes", lambda e: [FUNCTIONS["apoc.rel.startnode"](e),
                                FUNCTIONS["apoc.rel.endnode"](e)])
_r("apoc.rel.otherNode", lambda e, n: None if not isinstance(e, Edge)
    else (FUNCTIONS["apoc.rel.endnode"](e)
          if isinstance(n, Node) and e.start_node == n.id
          else FUNCTIONS["apoc.rel.startnode"](e)))
_r("apoc.rel.isLoop", lambda e: isinstance(e, Edge) and
    e.start_node == e.end_node)
_r("apoc.rel.isBetween", lambda e, a, b: isinstance(e, Edge) and
    {e.start_node, e.end_node} == {a.id if isinstance(a, Node) else a,
                                   b.id if isinstance(b, Node) else b})
_r("apoc.rel.isDirectedBetween", lambda e, a, b: isinstance(e, Edge) and
    e.start_node == (a.id if isinstance(a, Node) else a) and
    e.end_node == (b.id if isinstance(b, Node) else b))
_r("apoc.rel.direction", lambda e, n: None if not isinstance(e, Edge)
    else ("OUTGOING" if isinstance(n, Node) and e.start_node == n.id
          else "INCOMING"))
_r("apoc.rel.equals", lambda a, b: isinstance(a, Edge) and
    isinstance(b, Edge) and a.id == b.id)
_r("apoc.rel.compare", lambda a, b: isinstance(a, Edge) and
    isinstance(b, Edge) and a.type == b.type and a.properties == b.properties)
_r("apoc.rel.weight", lambda e, prop="weight", default=1.0: (
    e.properties.get(prop, default) if isinstance(e, Edge) else default))
_r("apoc.rel.toMap", lambda e: {"id": e.id, "type": e.type,
    "start": e.start_node, "end": e.end_node,
    "properties": dict(e.properties)} if isinstance(e, Edge) else None)

# ============================== apoc.meta ==============================
def _cypher_type(x):
    if x is None:
        return "NULL"
    if isinstance(x, bool):
        return "BOOLEAN"
    if isinstance(x, int):
        return "INTEGER"
    if isinstance(x, float):
        return "FLOAT"
    if isinstance(x, str):
        return "STRING"
    if isinstance(x, Node):
        return "NODE"
    if isinstance(x, Edge):
        return "RELATIONSHIP"
    if isinstance(x, list):
        return "LIST"
    if isinstance(x, dict):
        return "MAP"
    if isinstance(x, _tp.CypherDate):
        return "DATE"
    if isinstance(x, _tp.CypherDateTime):
        return "DATETIME"
    if isinstance(x, _tp.CypherTime):
        return "TIME"
    if isinstance(x, _tp.CypherDuration):
        return "DURATION"
    if hasattr(x, "nodes") and hasattr(x, "edges"):
        return "PATH"
    return type(x).__name__.upper()


_r("apoc.meta.type", _cypher_type)
_r("apoc.meta.typeOf", _cypher_type)
_r("apoc.meta.cypherType", _cypher_type)
_r("apoc.meta.types", lambda m: {k: _cypher_type(v)
                                 for k, v in _as_map(m).items()})
_r("apoc.meta.cypherTypes", lambda m: {k: _cypher_type(v)
                                       for k, v in _as_map(m).items()})
_r("apoc.meta.isType", lambda x, t: _cypher_type(x) == str(t).upper())
_r("apoc.meta.isNode", lambda x: isinstance(x, Node))
_r("apoc.meta.isRelationship", lambda x: isinstance(x, Edge))
_r("apoc.meta.isPath", lambda x: hasattr(x, "nodes") and hasattr(x, "edges")
    and not isinstance(x, (Node, Edge)))
_r("apoc.meta.nodeLabels", lambda n: list(n.labels)
    if isinstance(n, Node) else [])
_r("apoc.meta.toString", lambda x: str(x))
_r("apoc.meta.version", lambda: "nornicdb-amd-1.0")

# ============================== apoc.diff ==============================
def _diff_maps(a, b):
    a, b = _as_map(a), _as_map(b)
    return {
        "leftOnly": {k: a[k] for k in set(a) - set(b)},
        "rightOnly": {k: b[k] for k in set(b) - set(a)},
        "inCommon": {k: a[k] for k in set(a) & set(b) if a[k] == b[k]},
        "different": {k: {"left": a[k], "right": b[k]}
                      for k in set(a) & set(b) if a[k] != b[k]},
    }


_r("apoc.diff.maps", _diff_maps)
_r("apoc.diff.nodes", lambda a, b: _diff_maps(a.properties, b.properties)
    if isinstance(a, Node) and isinstance(b, Node) else None)
_r("apoc.diff.relationships", lambda a, b: _diff_maps(
    a.properties, b.properties)
    if isinstance(a, Edge) and isinstance(b, Edge) else None)
_r("apoc.diff.lists", lambda a, b: {
    "leftOnly": [x for x in (a or []) if x not in (b or [])],
    "rightOnly": [x for x in (b or []) if x not in (a or [])],
    "inCommon": [x for x in (a or []) if x in (b or [])]})
_r("apoc.diff.strings", lambda a, b: {
    "equal": a == b, "leftLength": len(a or ""), "rightLength": len(b or ""),
    "commonPrefix": _common_prefix(a or "", b or "")})


def _common_prefix(a, b):
    i = 0
    while i < min(len(a), len(b)) and a[i] == b[i]:
        i += 1
    return a[:i]


def _deep_diff(a, b, path=""):
    out = []
    if isinstance(a, dict) and isinstance(b, dict):
        for k in sorted(set(a) | set(b)):
            out.extend(_deep_diff(a.get(k), b.get(k),
                                  f"{path}.{k}" if path else str(k)))
    elif isinstance(a, list) and isinstance(b, list):
        for i in range(max(len(a), len(b))):
            out.extend(_deep_diff(a[i] if i < len(a) else None,
                                  b[i] if i < len(b) else None,
                                  f"{path}[{i}]"))
    elif a != b:
        out.append({"path": path, "left": a, "right": b})
    return out


_r("apoc.diff.deep", _deep_diff)
_r("apoc.diff.summary", lambda a, b: {"changes": len(_deep_diff(a, b))})


def _deep_merge(a, b):
    if isinstance(a, dict) and isinstance(b, dict):
        out = dict(a)
        for k, v in b.items():
            out[k] = _deep_merge(a.get(k), v) if k in a else v
        return out
    return b if b is not None else a


_r("apoc.diff.merge", _deep_merge)


def _patch(a, changes):
    out = json.loads(json.dumps(_as_map(a), default=str))
    for ch in changes or []:
        parts = re.split(r"\.|\[|\]\.?", ch["path"])
        parts = [p for p in parts if p]
        cur = out
        for p in parts[:-1]:
            cur = cur[int(p)] if isinstance(cur, list) else cur.setdefault(p, {})
        last = parts[-1]
        if isinstance(cur, list):
            cur[int(last)] = ch.get("right")
        else:
            cur[last] = ch.get("right")
    return out


_r("apoc.diff.patch", _patch)

# ============================== apoc.xml ==============================
def _xml_to_map(el):
    out = {"_type": el.tag}
    out.update({f"@{k}": v for k, v in el.attrib.items()})
    text = (el.text or "").strip()
    if text:
        out["_text"] = text
    children = [_xml_to_map(c) for c in el]
    if children:
        out["_children"] = children
    return out


def _xml_parse(s):
    return _xml_to_map(_ET.fromstring(s))


def _map_to_xml(m):
    el = _ET.Element(m.get("_type", "node"))
    for k, v in m.items():
        if k.startswith("@"):
            el.set(k[1:], str(v))
    if m.get("_text"):
        el.text = str(m["_text"])
    for c in m.get("_children", []):
        el.append(_map_to_xml(c))
    return el


_r("apoc.xml.parse", _xml_parse)
_r("apoc.xml.toMap", _xml_parse)
_r("apoc.xml.fromMap", lambda m: _ET.tostring(
    _map_to_xml(_as_map(m)), encoding="unicode"))
_r("apoc.xml.toString", lambda m: _ET.tostring(
    _map_to_xml(_as_map(m)), encoding="unicode") if isinstance(m, dict) else str(m))
_r("apoc.xml.toJson", lambda s: json.dumps(_xml_parse(s)))
_r("apoc.xml.fromJson", lambda s: _ET.tostring(
    _map_to_xml(json.loads(s)), encoding="unicode"))
_r("apoc.xml.escape", lambda s: (str(s).replace("&", "&amp;")
    .replace("<", "&lt;").replace(">", "&gt;").replace('"', "&quot;")
    .replace("'", "&apos;")))
_r("apoc.xml.unescape", lambda s: (str(s).replace("&lt;", "<")
    .replace("&gt;", ">").replace("&quot;", '"').replace("&apos;", "'")
    .replace("&amp;", "&")))
_r("apoc.xml.getAttribute", lambda m, name: _as_map(m).get(f"@{name}"))
_r("apoc.xml.getText", lambda m: _as_map(m).get("_text"))
_r("apoc.xml.minify", lambda s: re.sub(r">\s+<", "><", str(s).strip()))
_r("apoc.xml.prettify", lambda s: (lambda el: (_ET.indent(el),
    _ET.tostring(el, encoding="unicode"))[1])(_ET.fromstring(s)))
_r("apoc.xml.validate", lambda s: _xml_valid(s))


def _xml_valid(s):
    try:
        _ET.fromstring(s)
        return True
    except Exception:
        return False


def _xml_query(m, tag):
    """All descendant elements with the given tag."""
    out = []

    def walk(e):
        if e.get("_type") == tag:
            out.append(e)
        for c in e.get("_children", []):
            walk(c)
    walk(_as_map(m))
    return out


_r("apoc.xml.query", _xml_query)

# ============================== apoc.graph ==============================
def _graph_obj(nodes, rels, name="graph", props=None):
    return {"name": name, "nodes": list(nodes or []),
            "relationships": list(rels or []), "properties": props or {}}


_r("apoc.graph.fromData", lambda nodes, rels, name="graph", props=None:
    _graph_obj(nodes, rels, name, props))
_r("apoc.graph.from", lambda data, name="graph", props=None: _graph_obj(
    [x for x in (data if isinstance(data, list) else [data])
     if isinstance(x, Node)],
    [x for x in (data if isinstance(data, list) else [data])
     if isinstance(x, Edge)], name, props))
_r("apoc.graph.fromPath", lambda p, name="graph", props=None: _graph_obj(
    getattr(p, "nodes", []), getattr(p, "edges", []), name, props))
_r("apoc.graph.fromPaths", lambda ps, name="graph", props=None: _graph_obj(
    {n.id: n for p in (ps or []) for n in p.nodes}.values(),
    {e.id: e for p in (ps or []) for e in p.edges}.values(), name, props))
_r("apoc.graph.nodes", lambda g: _as_map(g).get("nodes", []))
_r("apoc.graph.relationships", lambda g: _as_map(g).get("relationships", []))
_r("apoc.graph.stats", lambda g: {
    "nodeCount": len(_as_map(g).get("nodes", [])),
    "relCount": len(_as_map(g).get("relationships", []))})
_r("apoc.graph.toMap", lambda g: {
    "name": _as_map(g).get("name"),
    "nodes": [FUNCTIONS["apoc.node.tomap"](n) for n in
              _as_map(g).get("nodes", [])],
    "relationships": [FUNCTIONS["apoc.rel.tomap"](e) for e in
                      _as_map(g).get("relationships", [])]})
_r("apoc.graph.merge", lambda a, b: _graph_obj(
    {n.id: n for n in _as_map(a).get("nodes", []) +
     _as_map(b).get("nodes", [])}.values(),
    {e.id: e for e in _as_map(a).get("relationships", []) +
     _as_map(b).get("relationships", [])}.values()))
_r("apoc.graph.validate", lambda g: isinstance(_as_map(g).get("nodes"), list)
    and isinstance(_as_map(g).get("relationships"), list))

# ============================== apoc.agg ==============================
# Aggregate finalizers: hooked into cypher.functions.Aggregator via
# cypher.functions.AGG_FINALIZERS (values list -> result).
def _agg(name, fn):
    AGGREGATES.add(name.lower())
    AGG_FINALIZERS[name.lower()] = fn


_agg("apoc.agg.first", lambda vals, extra=None: vals[0] if vals else None)
_agg("apoc.agg.last", lambda vals, extra=None: vals[-1] if vals else None)
_agg("apoc.agg.nth", lambda vals, extra=None: vals[int(extra)]
     if vals and extra is not None and int(extra) < len(vals) else None)
_agg("apoc.agg.slice", lambda vals, extra=None: vals[:int(extra)]
     if extra is not None else vals)
_agg("apoc.agg.median", lambda vals, extra=None: _st.median(
    _num_list(vals)) if _num_list(vals) else None)
_agg("apoc.agg.product", lambda vals, extra=None: math.prod(_num_list(vals)))
_agg("apoc.agg.statistics", lambda vals, extra=None: (lambda v: {
    "count": len(v), "min": min(v, default=None), "max": max(v, default=None),
    "mean": _st.mean(v) if v else None,
    "stdev": _st.stdev(v) if len(v) > 1 else 0.0})(_num_list(vals)))
_agg("apoc.agg.stdev", lambda vals, extra=None: _st.stdev(_num_list(vals))
     if len(_num_list(vals)) > 1 else 0.0)
_agg("apoc.agg.percentile", lambda vals, extra=None: _percentile(
    vals, extra if extra is not None else 0.5))
_agg("apoc.agg.mode", lambda vals, extra=None: _st.mode(vals) if vals else None)
_agg("apoc.agg.frequencies", lambda vals, extra=None: [
    {"value": v, "count": c} for v, c in
    sorted(((v, vals.count(v)) for v in dict.fromkeys(vals)),
           key=lambda t: -t[1])])
_agg("apoc.agg.histogram", lambda vals, extra=None: _histogram(
    vals, extra or 10))
_agg("apoc.agg.maxItems", lambda vals, extra=None: (lambda m: {
    "value": m, "items": [v for v in vals if v == m]})(max(vals))
    if vals else None)
_agg("apoc.agg.minItems", lambda vals, extra=None: (lambda m: {
    "value": m, "items": [v for v in vals if v == m]})(min(vals))
    if vals else None)
_agg("apoc.agg.graph", lambda vals, extra=None: _graph_obj(
    [x for x in vals if isinstance(x, Node)],
    [x for x in vals if isinstance(x, Edge)]))

# ============================== apoc.text extras ==============================
_r("apoc.text.trim", lambda s: None if s is None else str(s).strip())
_r("apoc.text.ltrim", lambda s: None if s is None else str(s).lstrip())
_r("apoc.text.rtrim", lambda s: None if s is None else str(s).rstrip())
_r("apoc.text.reverse", lambda s: None if s is None else str(s)[::-1])
_r("apoc.text.bytes", lambda s, charset="UTF-8": list(
    str(s).encode(charset)))
_r("apoc.text.bytesToString", lambda b, charset="UTF-8": bytes(
    x & 0xFF for x in (b or [])).decode(charset))
_r("apoc.text.fromCodePoint", lambda *cps: "".join(chr(int(c)) for c in cps))
_r("apoc.text.decapitalizeAll", lambda s: " ".join(
    w[:1].lower() + w[1:] for w in str(s or "").split(" ")))


def _soundexish(s):
    """Simplified phonetic code (soundex-style)."""
    s = re.sub(r"[^A-Za-z]", "", str(s or "")).upper()
    if not s:
        return ""
    codes = {"B": "1", "F": "1", "P": "1", "V": "1",
             "C": "2", "G": "2", "J": "2", "K": "2", "Q": "2", "S": "2",
             "X": "2", "Z": "2", "D": "3", "T": "3", "L": "4",
             "M": "5", "N": "5", "R": "6"}
    out = s[0]
    prev = codes.get(s[0], "")
    for ch in s[1:]:
        c = codes.get(ch, "")
        if c and c != prev:
            out += c
        prev = c
    return (out + "000")[:4]


_r("apoc.text.phonetic", _soundexish)
_r("apoc.text.phoneticDelta", lambda a, b: sum(
    1 for x, y in zip(_soundexish(a), _soundexish(b)) if x == y))


def _metaphoneish(s):
    """Compact consonant-skeleton code (double-metaphone stand-in)."""
    s = re.sub(r"[^A-Za-z]", "", str(s or "")).upper()
    s = re.sub(r"PH", "F", s)
    s = re.sub(r"[AEIOU]", "", s[1:])
    return (str(s)[:1] if False else "") or s[:6]


_r("apoc.text.doubleMetaphone", _metaphoneish)


def _fuzzy_match(a, b):
    from .functions import _levenshtein
    a, b = str(a or "").lower(), str(b or "").lower()
    if not a or not b:
        return False
    d = _levenshtein(a, b)
    allowed = 1 if len(a) < 3 else (2 if len(a) < 5 else 3)
    return d <= allowed


_r("apoc.text.fuzzyMatch", _fuzzy_match)

# ============================== apoc.coll extras ==============================
_r("apoc.coll.isEmpty", lambda l: not l)
_r("apoc.coll.isNotEmpty", lambda l: bool(l))
_r("apoc.coll.containsAny", lambda l, items: bool(
    set(map(repr, l or [])) & set(map(repr, items or []))))
_r("apoc.coll.containsDuplicates", lambda l: len(l or []) != len(
    set(map(repr, l or []))))
_r("apoc.coll.containsSorted", lambda l, v: (lambda s: (lambda i:
    i < len(s) and s[i] == v)(__import__("bisect").bisect_left(s, v)))(
    sorted(l or [])))
_r("apoc.coll.duplicatesWithCount", lambda l: [
    {"item": v, "count": c} for v, c in
    ((x, (l or []).count(x)) for x in dict.fromkeys(l or [])) if c > 1])
_r("apoc.coll.dropDuplicateNeighbors", lambda l: [
    v for i, v in enumerate(l or []) if i == 0 or v != l[i - 1]])
_r("apoc.coll.fill", lambda v, n: [v] * int(n))
_r("apoc.coll.insertAll", lambda l, idx, items: (
    list(l or [])[:int(idx)] + list(items or []) + list(l or [])[int(idx):]))
_r("apoc.coll.removeAll", lambda l, items: [
    v for v in (l or []) if repr(v) not in set(map(repr, items or []))])
_r("apoc.coll.set", lambda l, idx, v: [
    v if i == int(idx) else x for i, x in enumerate(l or [])])
_r("apoc.coll.unionAll", lambda a, b: list(a or []) + list(b or []))
_r("apoc.coll.sumLongs", lambda l: int(sum(
    int(x) for x in (l or []) if x is not None)))
_r("apoc.coll.frequenciesAsMap", lambda l: {
    str(v): (l or []).count(v) for v in dict.fromkeys(l or [])})
_r("apoc.coll.randomItems", lambda l, n, allow_repeat=False: (
    random.choices(l, k=int(n)) if allow_repeat
    else random.sample(list(l), min(int(n), len(l)))) if l else [])
_r("apoc.coll.sortMaps", lambda l, key: sorted(
    l or [], key=lambda m: (m.get(key) is None, m.get(key)), reverse=True))

# ============================== apoc.map extras ==============================
_r("apoc.map.dropNullValues", lambda m: {k: v for k, v in _as_map(m).items()
                                         if v is not None})
_r("apoc.map.fromValues", lambda l: {str(l[i]): l[i + 1]
                                     for i in range(0, len(l or []) - 1, 2)})
_r("apoc.map.mget", lambda m, keys, defaults=None: [
    _as_map(m).get(k, (defaults or [None] * len(keys))[i])
    for i, k in enumerate(keys or [])])
_r("apoc.map.setEntry", lambda m, k, v: {**_as_map(m), str(k): v})
_r("apoc.map.setValues", lambda m, pairs: {**_as_map(m), **{
    str(pairs[i]): pairs[i + 1] for i in range(0, len(pairs or []) - 1, 2)}})
_r("apoc.map.setPairs", lambda m, pairs: {**_as_map(m), **{
    str(p[0]): p[1] for p in (pairs or [])}})
_r("apoc.map.setLists", lambda m, keys, values: {**_as_map(m), **dict(
    zip([str(k) for k in (keys or [])], values or []))})
_r("apoc.map.sortedProperties", lambda m, ignore_case=True: [
    [k, _as_map(m)[k]] for k in sorted(
        _as_map(m), key=(lambda s: s.lower()) if ignore_case else None)])
_r("apoc.map.unflatten", _json_unflatten)


def _update_tree(tree, key, data):
    t = _as_map(tree)
    out = dict(t)
    for k, v in _as_map(data).items():
        out[k] = v
    return out


_r("apoc.map.updateTree", _update_tree)

# ============================== apoc.convert extras ==============================
def _to_tree(paths):
    """paths -> nested {children: []} tree (reference convert.toTree)."""
    roots: Dict[str, Any] = {}
    nodes: Dict[str, Any] = {}

    def ent(n):
        if n.id not in nodes:
            nodes[n.id] = {"_id": n.id, "_type": ":".join(n.labels),
                           **dict(n.properties)}
        return nodes[n.id]

    child_ids = set()
    for p in paths or []:
        ns = getattr(p, "nodes", [])
        es = getattr(p, "edges", [])
        for n in ns:
            ent(n)
        for e in es:
            parent = nodes.get(e.start_node)
            child = nodes.get(e.end_node)
            if parent is None or child is None:
                continue
            key = e.type.lower()
            parent.setdefault(key, [])
            if child not in parent[key]:
                parent[key].append(child)
            child_ids.add(e.end_node)
    for p in paths or []:
        for n in getattr(p, "nodes", []):
            if n.id not in child_ids:
                roots[n.id] = nodes[n.id]
    return list(roots.values())


_r("apoc.convert.toTree", _to_tree)


# ============================== apoc.paths / apoc.path (pure) ==============================
def _path_nodes(p):
    return list(getattr(p, "nodes", []))


def _path_edges(p):
    return list(getattr(p, "edges", []))


def _mk_path(nodes, edges):
    from ..cypher.executor import Path as _P
    return _P(nodes, edges)


_r("apoc.path.elements", lambda p: [x for pair in zip(
    _path_nodes(p), _path_edges(p) + [None]) for x in pair
    if x is not None])
_r("apoc.path.slice", lambda p, offset=0, length=None: _mk_path(
    _path_nodes(p)[int(offset):int(offset) + (int(length) + 1
                                              if length is not None else None or len(_path_nodes(p)))],
    _path_edges(p)[int(offset):int(offset) + (int(length)
                                              if length is not None else len(_path_edges(p)))]))
_r("apoc.path.combine", lambda a, b: _mk_path(
    _path_nodes(a) + _path_nodes(b)[1:], _path_edges(a) + _path_edges(b)))
_r("apoc.paths.reverse", lambda p: _mk_path(
    list(reversed(_path_nodes(p))), list(reversed(_path_edges(p)))))
_r("apoc.paths.slice", FUNCTIONS["apoc.path.slice"])
_r("apoc.paths.merge", lambda a, b: _mk_path(
    _path_nodes(a) + _path_nodes(b)[1:], _path_edges(a) + _path_edges(b)))
_r("apoc.paths.unique", lambda paths: (lambda seen: [p for p in (paths or [])
    if (key := tuple(n.id for n in _path_nodes(p))) not in seen
    and not seen.add(key)])(set()))
_r("apoc.paths.withLength", lambda paths, n: [
    p for p in (paths or []) if len(_path_edges(p)) == int(n)])
_r("apoc.paths.withinLength", lambda paths, n: [
    p for p in (paths or []) if len(_path_edges(p)) <= int(n)])
_r("apoc.paths.longest", lambda paths: max(
    paths or [], key=lambda p: len(_path_edges(p)), default=None))
_r("apoc.paths.shortest", lambda paths: min(
    paths or [], key=lambda p: len(_path_edges(p)), default=None))
_r("apoc.paths.simple", lambda p: len({n.id for n in _path_nodes(p)}) ==
    len(_path_nodes(p)))
_r("apoc.paths.elementary", FUNCTIONS["apoc.paths.simple"])
_r("apoc.paths.cycles", lambda paths: [
    p for p in (paths or []) if _path_nodes(p)
    and _path_nodes(p)[0].id == _path_nodes(p)[-1].id
    and len(_path_edges(p)) > 0])
_r("apoc.paths.disjoint", lambda a, b: not (
    {n.id for n in _path_nodes(a)} & {n.id for n in _path_nodes(b)}))
_r("apoc.paths.edgeDisjoint", lambda a, b: not (
    {e.id for e in _path_edges(a)} & {e.id for e in _path_edges(b)}))
_r("apoc.paths.all", lambda paths: list(paths or []))
_r("apoc.paths.kShortest", lambda paths, k: sorted(
    paths or [], key=lambda p: len(_path_edges(p)))[:int(k)])

# ============================== apoc.merge (map utilities) ==============================
_r("apoc.merge.properties", lambda a, b: {**_as_map(a), **_as_map(b)})
_r("apoc.merge.deepMerge", _deep_merge)
_r("apoc.merge.labels", lambda a, b: sorted(set(_labels_of(a)) |
                                            set(_labels_of(b))))
_r("apoc.merge.conditional", lambda cond, a, b: _as_map(a) if cond
    else _as_map(b))
_r("apoc.merge.conflict", lambda a, b: sorted(
    k for k in set(_as_map(a)) & set(_as_map(b))
    if _as_map(a)[k] != _as_map(b)[k]))
_r("apoc.merge.preview", lambda a, b: {
    "merged": {**_as_map(a), **_as_map(b)},
    "conflicts": sorted(k for k in set(_as_map(a)) & set(_as_map(b))
                        if _as_map(a)[k] != _as_map(b)[k])})
_r("apoc.merge.strategy", lambda a, b, strategy="overwrite": (
    {**_as_map(b), **_as_map(a)} if strategy == "keep"
    else {**_as_map(a), **_as_map(b)}))
_r("apoc.merge.validate", lambda a, b: not [
    k for k in set(_as_map(a)) & set(_as_map(b))
    if _as_map(a)[k] != _as_map(b)[k]])
_r("apoc.merge.snapshot", lambda m: json.loads(json.dumps(_as_map(m),
                                                          default=str)))
_r("apoc.merge.rollback", lambda current, snapshot: _as_map(snapshot))
_r("apoc.merge.pattern", lambda m, pat: {k: v for k, v in _as_map(m).items()
                                         if re.match(pat, str(k))})

# ============================== apoc.convert / cypher leftovers ==============================
_r("apoc.convert.setJsonProperty", lambda ent, prop, value: (
    ent.properties.__setitem__(prop, json.dumps(value, default=str)) or ent
    if isinstance(ent, (Node, Edge)) else None))
_r("apoc.cypher.toList", lambda v: list(v) if v is not None else [])
_r("apoc.cypher.toMap", lambda v: _as_map(v))

# ============================== apoc.xml DOM helpers ==============================
_r("apoc.xml.create", lambda tag, attrs=None, text=None: {
    "_type": str(tag), **{f"@{k}": v for k, v in _as_map(attrs).items()},
    **({"_text": text} if text else {})})
_r("apoc.xml.clone", lambda m: json.loads(json.dumps(_as_map(m))))
_r("apoc.xml.setAttribute", lambda m, k, v: {**_as_map(m), f"@{k}": v})
_r("apoc.xml.setText", lambda m, t: {**_as_map(m), "_text": str(t)})
_r("apoc.xml.addChild", lambda m, child: {**_as_map(m), "_children":
    _as_map(m).get("_children", []) + [_as_map(child)]})
_r("apoc.xml.removeChild", lambda m, tag: {**_as_map(m), "_children": [
    c for c in _as_map(m).get("_children", []) if c.get("_type") != tag]})
_r("apoc.xml.getNamespace", lambda m: _as_map(m).get("@xmlns"))
_r("apoc.xml.namespace", lambda m: _as_map(m).get("@xmlns"))
_r("apoc.xml.transform", lambda m, mapping: {
    (_as_map(mapping).get(k, k)): v for k, v in _as_map(m).items()})

# ============================== apoc.nodes (pure subset) ==============================
_r("apoc.nodes.distinct", lambda l: list({n.id: n for n in (l or [])
                                          if isinstance(n, Node)}.values()))
_r("apoc.nodes.distinctRels", lambda l: list({e.id: e for e in (l or [])
                                              if isinstance(e, Edge)}.values()))
_r("apoc.nodes.union", lambda a, b: list({n.id: n for n in
    list(a or []) + list(b or [])}.values()))
_r("apoc.nodes.intersect", lambda a, b: (lambda ids: [
    n for n in (a or []) if n.id in ids])({n.id for n in (b or [])}))
_r("apoc.nodes.difference", lambda a, b: (lambda ids: [
    n for n in (a or []) if n.id not in ids])({n.id for n in (b or [])}))
_r("apoc.nodes.sort", lambda l, prop: sorted(
    l or [], key=lambda n: (n.properties.get(prop) is None,
                            n.properties.get(prop))))
_r("apoc.nodes.partition", lambda l, size: [
    list((l or [])[i:i + int(size)]) for i in range(0, len(l or []), int(size))])
_r("apoc.nodes.toMap", lambda l: {n.id: dict(n.properties)
                                  for n in (l or []) if isinstance(n, Node)})
_r("apoc.nodes.fromMap", lambda m: list(_as_map(m).keys()))
_r("apoc.nodes.map", lambda l, prop: [n.properties.get(prop)
                                      for n in (l or [])])
_r("apoc.nodes.filter", lambda l, prop, value: [
    n for n in (l or []) if n.properties.get(prop) == value])
_r("apoc.nodes.reduce", lambda l, prop: sum(
    n.properties.get(prop, 0) for n in (l or [])
    if isinstance(n.properties.get(prop), (int, float))))


def _flatten_deep(l):
    out = []
    for x in (l or []):
        if isinstance(x, list):
            out.extend(_flatten_deep(x))
        else:
            out.append(x)
    return out


_r("apoc.text.join", lambda l, sep="": str(sep).join(
    str(x) for x in (l or []) if x is not None))
_r("apoc.coll.flatten", lambda l, recursive=False: (
    _flatten_deep(l) if recursive else
    [y for x in (l or []) for y in (x if isinstance(x, list) else [x])]))
