"""APOC function library (expression-position functions).

Parity: reference apoc/ (47 category packages, ~950 functions,
apoc/registry/registry.go). This module covers the core categories —
coll, map, text, math, number, date, convert/json, hashing, meta, label —
registered into the Cypher function table under their dotted names.
"""

from __future__ import annotations

import base64
import datetime as _dt
import hashlib
import json
import math
import random
import re
import urllib.parse
import zlib
from typing import Any, List

from ..cypher.functions import FUNCTIONS, register
from ..storage.types import Edge, Node


def _reg(name):
    def deco(fn):
        FUNCTIONS[name.lower()] = fn
        return fn
    return deco


# ============================== apoc.coll ==============================
_reg("apoc.coll.sum")(lambda l: sum(x for x in (l or []) if x is not None))
_reg("apoc.coll.avg")(lambda l: (sum(l) / len(l)) if l else None)
_reg("apoc.coll.min")(lambda l: min((x for x in l if x is not None), default=None) if l else None)
_reg("apoc.coll.max")(lambda l: max((x for x in l if x is not None), default=None) if l else None)
_reg("apoc.coll.sort")(lambda l: sorted(l) if l is not None else None)
_reg("apoc.coll.sortNodes")(lambda l, prop: sorted(l, key=lambda n: (n.properties.get(prop) is None, n.properties.get(prop))) if l else [])
_reg("apoc.coll.reverse")(lambda l: list(reversed(l)) if l is not None else None)
_reg("apoc.coll.contains")(lambda l, v: v in (l or []))
_reg("apoc.coll.containsAll")(lambda l, vs: all(v in (l or []) for v in (vs or [])))
_reg("apoc.coll.indexOf")(lambda l, v: (l or []).index(v) if v in (l or []) else -1)
_reg("apoc.coll.toSet")(lambda l: list(dict.fromkeys(l or [])))
_reg("apoc.coll.union")(lambda a, b: list(dict.fromkeys((a or []) + (b or []))))
_reg("apoc.coll.intersection")(lambda a, b: [x for x in dict.fromkeys(a or []) if x in (b or [])])
_reg("apoc.coll.subtract")(lambda a, b: [x for x in dict.fromkeys(a or []) if x not in (b or [])])
_reg("apoc.coll.disjunction")(lambda a, b: [x for x in dict.fromkeys((a or []) + (b or []))
                                            if (x in (a or [])) != (x in (b or []))])
_reg("apoc.coll.flatten")(lambda l: [x for sub in (l or []) for x in (sub if isinstance(sub, list) else [sub])])
_reg("apoc.coll.pairs")(lambda l: [[l[i], l[i + 1] if i + 1 < len(l) else None] for i in range(len(l or []))] if l else [])
_reg("apoc.coll.pairsMin")(lambda l: [[l[i], l[i + 1]] for i in range(len(l) - 1)] if l and len(l) > 1 else [])
_reg("apoc.coll.zip")(lambda a, b: [[x, y] for x, y in zip(a or [], b or [])])
_reg("apoc.coll.frequencies")(lambda l: [{"item": k, "count": v} for k, v in
                                         __import__("collections").Counter(l or []).items()])
_reg("apoc.coll.occurrences")(lambda l, v: (l or []).count(v))
_reg("apoc.coll.duplicates")(lambda l: [k for k, v in __import__("collections").Counter(l or []).items() if v > 1])
_reg("apoc.coll.shuffle")(lambda l: random.sample(l, len(l)) if l else [])
_reg("apoc.coll.randomItem")(lambda l: random.choice(l) if l else None)
_reg("apoc.coll.slice")(lambda l, off, length=None: (l or [])[off:off + length if length is not None else None])
_reg("apoc.coll.partition")(lambda l, size: [l[i:i + size] for i in range(0, len(l or []), size)])
_reg("apoc.coll.split")(lambda l, v: _coll_split(l or [], v))
_reg("apoc.coll.insert")(lambda l, idx, v: (l or [])[:idx] + [v] + (l or [])[idx:])
_reg("apoc.coll.remove")(lambda l, idx, length=1: (l or [])[:idx] + (l or [])[idx + length:])
_reg("apoc.coll.different")(lambda l: len(set(map(repr, l or []))) == len(l or []))
_reg("apoc.coll.isEqualCollection")(lambda a, b: sorted(map(repr, a or [])) == sorted(map(repr, b or [])))


def _coll_split(l, v):
    out, cur = [], []
    for x in l:
        if x == v:
            if cur:
                out.append(cur)
            cur = []
        else:
            cur.append(x)
    if cur:
        out.append(cur)
    return out


# ============================== apoc.map ==============================
_reg("apoc.map.fromPairs")(lambda pairs: {p[0]: p[1] for p in (pairs or [])})
_reg("apoc.map.fromLists")(lambda ks, vs: dict(zip(ks or [], vs or [])))
_reg("apoc.map.merge")(lambda a, b: {**(a or {}), **(b or {})})
_reg("apoc.map.mergeList")(lambda ms: {k: v for m in (ms or []) for k, v in (m or {}).items()})
_reg("apoc.map.setKey")(lambda m, k, v: {**(m or {}), k: v})
_reg("apoc.map.removeKey")(lambda m, k: {x: v for x, v in (m or {}).items() if x != k})
_reg("apoc.map.removeKeys")(lambda m, ks: {x: v for x, v in (m or {}).items() if x not in (ks or [])})
_reg("apoc.map.clean")(lambda m, ks=None, vs=None: {
    x: v for x, v in (m or {}).items()
    if x not in (ks or []) and v is not None and v not in (vs or [])})
_reg("apoc.map.get")(lambda m, k, default=None: (m or {}).get(k, default))
_reg("apoc.map.submap")(lambda m, ks: {k: (m or {}).get(k) for k in (ks or [])})
_reg("apoc.map.keys")(lambda m: sorted((m or {}).keys()))
_reg("apoc.map.values")(lambda m, ks=None: [(m or {}).get(k) for k in (ks or sorted((m or {}).keys()))])
_reg("apoc.map.flatten")(lambda m, delim=".": _map_flatten(m or {}, delim))
_reg("apoc.map.groupBy")(lambda l, key: {str((x or {}).get(key)): x for x in (l or [])})
_reg("apoc.map.groupByMulti")(lambda l, key: _group_multi(l or [], key))


def _map_flatten(m, delim, prefix=""):
    out = {}
    for k, v in m.items():
        kk = f"{prefix}{delim}{k}" if prefix else k
        if isinstance(v, dict):
            out.update(_map_flatten(v, delim, kk))
        else:
            out[kk] = v
    return out


def _group_multi(l, key):
    out = {}
    for x in l:
        out.setdefault(str((x or {}).get(key)), []).append(x)
    return out


# ============================== apoc.text ==============================
_reg("apoc.text.join")(lambda l, sep: sep.join(str(x) for x in (l or []) if x is not None))
_reg("apoc.text.split")(lambda s, rx: re.split(rx, s) if s is not None else None)
_reg("apoc.text.replace")(lambda s, rx, repl: re.sub(rx, repl, s) if s is not None else None)
_reg("apoc.text.regexGroups")(lambda s, rx: [list(m.groups()) if m.groups() else [m.group(0)]
                                             for m in re.finditer(rx, s or "")])
_reg("apoc.text.capitalize")(lambda s: s[:1].upper() + s[1:] if s else s)
_reg("apoc.text.decapitalize")(lambda s: s[:1].lower() + s[1:] if s else s)
_reg("apoc.text.capitalizeAll")(lambda s: " ".join(w.capitalize() for w in s.split(" ")) if s is not None else None)
_reg("apoc.text.swapCase")(lambda s: s.swapcase() if s is not None else None)
_reg("apoc.text.camelCase")(lambda s: _camel(s, False))
_reg("apoc.text.upperCamelCase")(lambda s: _camel(s, True))
_reg("apoc.text.snakeCase")(lambda s: re.sub(r"[\s_-]+", "-", re.sub(r"(?<=[a-z0-9])([A-Z])", r"-\1", s or "")).lower().replace("-", "-") if s is not None else None)
_reg("apoc.text.toUpperCase")(lambda s: re.sub(r"[\s-]+", "_", (s or "")).upper() if s is not None else None)
_reg("apoc.text.random")(lambda length, valid="A-Za-z0-9": "".join(
    random.choice(_expand_ranges(valid)) for _ in range(int(length))))
_reg("apoc.text.lpad")(lambda s, width, pad=" ": (s or "").rjust(width, pad))
_reg("apoc.text.rpad")(lambda s, width, pad=" ": (s or "").ljust(width, pad))
_reg("apoc.text.format")(lambda fmt, params: (fmt or "") % tuple(params or []))
_reg("apoc.text.indexOf")(lambda s, sub, offset=0: (s or "").find(sub, offset))
_reg("apoc.text.indexesOf")(lambda s, sub: [m.start() for m in re.finditer(re.escape(sub), s or "")])
_reg("apoc.text.distance")(lambda a, b: _levenshtein(a or "", b or ""))
_reg("apoc.text.levenshteinDistance")(lambda a, b: _levenshtein(a or "", b or ""))
_reg("apoc.text.levenshteinSimilarity")(lambda a, b: 1.0 - _levenshtein(a or "", b or "") / max(len(a or ""), len(b or ""), 1))
_reg("apoc.text.hammingDistance")(lambda a, b: sum(c1 != c2 for c1, c2 in zip(a or "", b or "")) + abs(len(a or "") - len(b or "")))
_reg("apoc.text.sorensenDiceSimilarity")(lambda a, b: _dice(a or "", b or ""))
_reg("apoc.text.jaroWinklerDistance")(lambda a, b: _jaro_winkler(a or "", b or ""))
_reg("apoc.text.clean")(lambda s: re.sub(r"[^a-z0-9]", "", (s or "").lower()))
_reg("apoc.text.compareCleaned")(lambda a, b: re.sub(r"[^a-z0-9]", "", (a or "").lower()) == re.sub(r"[^a-z0-9]", "", (b or "").lower()))
_reg("apoc.text.urlencode")(lambda s: urllib.parse.quote(s or "", safe=""))
_reg("apoc.text.urldecode")(lambda s: urllib.parse.unquote(s or ""))
_reg("apoc.text.base64Encode")(lambda s: base64.b64encode((s or "").encode()).decode())
_reg("apoc.text.base64Decode")(lambda s: base64.b64decode(s or "").decode())
_reg("apoc.text.charAt")(lambda s, i: ord(s[i]) if s and 0 <= i < len(s) else None)
_reg("apoc.text.code")(lambda i: chr(i))
_reg("apoc.text.repeat")(lambda s, n: (s or "") * int(n))
_reg("apoc.text.slug")(lambda s, sep="-": re.sub(r"[\W_]+", sep, (s or "").strip()).strip(sep).lower())


def _expand_ranges(spec: str) -> str:
    out = []
    i = 0
    while i < len(spec):
        if i + 2 < len(spec) and spec[i + 1] == "-":
            out.extend(chr(c) for c in range(ord(spec[i]), ord(spec[i + 2]) + 1))
            i += 3
        else:
            out.append(spec[i])
            i += 1
    return "".join(out)


def _camel(s, upper_first):
    if s is None:
        return None
    parts = re.split(r"[\s_-]+", s)
    parts = [p for p in parts if p]
    if not parts:
        return ""
    first = parts[0].capitalize() if upper_first else parts[0].lower()
    return first + "".join(p.capitalize() for p in parts[1:])


def _levenshtein(a: str, b: str) -> int:
    if len(a) < len(b):
        a, b = b, a
    prev = list(range(len(b) + 1))
    for i, ca in enumerate(a, 1):
        cur = [i]
        for j, cb in enumerate(b, 1):
            cur.append(min(prev[j] + 1, cur[-1] + 1, prev[j - 1] + (ca != cb)))
        prev = cur
    return prev[-1]


def _dice(a, b):
    if a == b:
        return 1.0
    ba = {a[i:i + 2] for i in range(len(a) - 1)}
    bb = {b[i:i + 2] for i in range(len(b) - 1)}
    if not ba or not bb:
        return 0.0
    return 2 * len(ba & bb) / (len(ba) + len(bb))


def _jaro_winkler(a, b):
    if a == b:
        return 1.0
    la, lb = len(a), len(b)
    if not la or not lb:
        return 0.0
    window = max(la, lb) // 2 - 1
    ma = [False] * la
    mb = [False] * lb
    matches = 0
    for i in range(la):
        lo, hi = max(0, i - window), min(lb, i + window + 1)
        for j in range(lo, hi):
            if not mb[j] and a[i] == b[j]:
                ma[i] = mb[j] = True
                matches += 1
                break
    if not matches:
        return 0.0
    t = 0
    k = 0
    for i in range(la):
        if ma[i]:
            while not mb[k]:
                k += 1
            if a[i] != b[k]:
                t += 1
            k += 1
    t /= 2
    jaro = (matches / la + matches / lb + (matches - t) / matches) / 3
    prefix = 0
    for x, y in zip(a, b):
        if x == y and prefix < 4:
            prefix += 1
        else:
            break
    return jaro + prefix * 0.1 * (1 - jaro)


# ============================== apoc.math / number ==============================
_reg("apoc.math.round")(lambda v, precision=0: round(v, int(precision)) if v is not None else None)
_reg("apoc.math.maxLong")(lambda: 2 ** 63 - 1)
_reg("apoc.math.minLong")(lambda: -2 ** 63)
_reg("apoc.math.maxDouble")(lambda: 1.7976931348623157e308)
_reg("apoc.math.sigmoid")(lambda x: 1.0 / (1.0 + math.exp(-x)) if x is not None else None)
_reg("apoc.math.tanh")(lambda x: math.tanh(x) if x is not None else None)
_reg("apoc.math.cosh")(lambda x: math.cosh(x) if x is not None else None)
_reg("apoc.math.sinh")(lambda x: math.sinh(x) if x is not None else None)
_reg("apoc.number.format")(lambda v, pattern=None: f"{v:,}" if v is not None else None)
_reg("apoc.number.parseInt")(lambda s: int(re.sub(r"[^\d-]", "", s)) if s else None)
_reg("apoc.number.parseFloat")(lambda s: float(s) if s else None)


# ============================== apoc.date / temporal ==============================
_DATE_FMT_MAP = {"yyyy": "%Y", "MM": "%m", "dd": "%d", "HH": "%H",
                 "mm": "%M", "ss": "%S", "SSS": "%f"}


def _java_fmt(fmt: str) -> str:
    for j, p in _DATE_FMT_MAP.items():
        fmt = fmt.replace(j, p)
    return fmt


@_reg("apoc.date.format")
def _date_format(epoch, unit="ms", fmt="yyyy-MM-dd HH:mm:ss"):
    if epoch is None:
        return None
    secs = epoch / 1000.0 if unit == "ms" else float(epoch)
    return _dt.datetime.utcfromtimestamp(secs).strftime(_java_fmt(fmt))


@_reg("apoc.date.parse")
def _date_parse(s, unit="ms", fmt="yyyy-MM-dd HH:mm:ss"):
    if s is None:
        return None
    dt = _dt.datetime.strptime(s, _java_fmt(fmt))
    epoch = (dt - _dt.datetime(1970, 1, 1)).total_seconds()
    return int(epoch * 1000) if unit == "ms" else int(epoch)


_reg("apoc.date.currentTimestamp")(lambda: int(_dt.datetime.now().timestamp() * 1000))
_reg("apoc.date.add")(lambda t, unit, value, add_unit: t + _unit_ms(add_unit) * value)
_reg("apoc.date.convert")(lambda t, from_u, to_u: int(t * _unit_ms(from_u) / _unit_ms(to_u)))


def _unit_ms(u):
    return {"ms": 1, "s": 1000, "m": 60000, "h": 3600000, "d": 86400000}[u]


# ============================== apoc.convert / json ==============================
_reg("apoc.convert.toJson")(lambda v: json.dumps(_plain(v), default=str))
_reg("apoc.convert.fromJsonMap")(lambda s: json.loads(s) if s else None)
_reg("apoc.convert.fromJsonList")(lambda s: json.loads(s) if s else None)
_reg("apoc.convert.toList")(lambda v: list(v) if v is not None else [])
_reg("apoc.convert.toString")(lambda v: str(v) if v is not None else None)
_reg("apoc.convert.toBoolean")(lambda v: bool(v) if not isinstance(v, str) else v.lower() in ("true", "1", "yes"))
_reg("apoc.convert.toInteger")(lambda v: int(float(v)) if v is not None else None)
_reg("apoc.convert.toFloat")(lambda v: float(v) if v is not None else None)
_reg("apoc.json.path")(lambda m, path: _json_path(m, path))


def _plain(v):
    if isinstance(v, Node):
        return {"id": v.id, "labels": v.labels, "properties": v.properties}
    if isinstance(v, Edge):
        return {"id": v.id, "type": v.type, "start": v.start_node,
                "end": v.end_node, "properties": v.properties}
    if isinstance(v, list):
        return [_plain(x) for x in v]
    if isinstance(v, dict):
        return {k: _plain(x) for k, x in v.items()}
    return v


def _json_path(m, path):
    cur = m
    for part in path.lstrip("$").lstrip(".").split("."):
        if not part:
            continue
        mm = re.match(r"(\w+)(\[(\d+)\])?", part)
        cur = (cur or {}).get(mm.group(1))
        if mm.group(3) is not None and isinstance(cur, list):
            idx = int(mm.group(3))
            cur = cur[idx] if idx < len(cur) else None
    return cur


# ============================== apoc.hashing / util ==============================
_reg("apoc.util.md5")(lambda vals: hashlib.md5("".join(map(str, vals if isinstance(vals, list) else [vals])).encode()).hexdigest())
_reg("apoc.util.sha1")(lambda vals: hashlib.sha1("".join(map(str, vals if isinstance(vals, list) else [vals])).encode()).hexdigest())
_reg("apoc.util.sha256")(lambda vals: hashlib.sha256("".join(map(str, vals if isinstance(vals, list) else [vals])).encode()).hexdigest())
_reg("apoc.util.sha512")(lambda vals: hashlib.sha512("".join(map(str, vals if isinstance(vals, list) else [vals])).encode()).hexdigest())
_reg("apoc.hashing.fingerprint")(lambda v: hashlib.md5(json.dumps(_plain(v), sort_keys=True, default=str).encode()).hexdigest())
_reg("apoc.util.compress")(lambda s: zlib.compress((s or "").encode()))
_reg("apoc.util.decompress")(lambda b: zlib.decompress(b).decode())
_reg("apoc.util.validatePredicate")(lambda pred, msg, params=None: _validate(pred, msg))


def _validate(pred, msg):
    if pred:
        raise ValueError(msg)
    return True


# ============================== apoc.label / node ==============================
_reg("apoc.label.exists")(lambda n, lb: lb in n.labels if isinstance(n, Node) else False)
_reg("apoc.node.degree")(lambda n: None)  # engine-bound; overridden per-executor
_reg("apoc.meta.cypher.type")(lambda v: _cypher_type(v))


def _cypher_type(v):
    if v is None:
        return "NULL"
    if isinstance(v, bool):
        return "BOOLEAN"
    if isinstance(v, int):
        return "INTEGER"
    if isinstance(v, float):
        return "FLOAT"
    if isinstance(v, str):
        return "STRING"
    if isinstance(v, list):
        return "LIST"
    if isinstance(v, dict):
        return "MAP"
    if isinstance(v, Node):
        return "NODE"
    if isinstance(v, Edge):
        return "RELATIONSHIP"
    return type(v).__name__.upper()


def function_count() -> int:
    return len([k for k in FUNCTIONS if k.startswith("apoc.")])
