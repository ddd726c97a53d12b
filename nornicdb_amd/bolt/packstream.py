"""PackStream binary serialization (Bolt wire format).

Parity: reference pkg/bolt/packstream.go (1.3K LoC). Implements PackStream
v1/v2 markers plus the Bolt graph structures (Node 'N', Relationship 'R',
UnboundRelationship 'r', Path 'P').
"""

from __future__ import annotations

import struct
from typing import Any, Dict, List, Tuple


class Structure:
    def __init__(self, tag: int, fields: List[Any]):
        self.tag = tag
        self.fields = fields

    def __repr__(self):
        return f"Structure(0x{self.tag:02x}, {self.fields!r})"

    def __eq__(self, o):
        return isinstance(o, Structure) and o.tag == self.tag and o.fields == self.fields


class PackStreamError(Exception):
    pass


try:  # native codec (csrc/packstream.cpp); python fallback below
    from nornicdb_amd import _C as _native
    _HAS_NATIVE_PS = hasattr(_native, "ps_pack")
except ImportError:  # pragma: no cover
    _native = None
    _HAS_NATIVE_PS = False


# ---------------------------------------------------------------- packing
def pack(value: Any) -> bytes:
    if _HAS_NATIVE_PS:
        return _native.ps_pack(value)
    out = bytearray()
    _pack_into(out, value)
    return bytes(out)


def pack_py(value: Any) -> bytes:
    """Pure-python packer (kept as the oracle for codec tests)."""
    out = bytearray()
    _pack_into(out, value)
    return bytes(out)


def _pack_into(out: bytearray, v: Any) -> None:
    if v is None:
        out.append(0xC0)
    elif v is True:
        out.append(0xC3)
    elif v is False:
        out.append(0xC2)
    elif isinstance(v, int):
        _pack_int(out, v)
    elif isinstance(v, float):
        out.append(0xC1)
        out += struct.pack(">d", v)
    elif isinstance(v, str):
        b = v.encode("utf-8")
        n = len(b)
        if n < 0x10:
            out.append(0x80 + n)
        elif n < 0x100:
            out += bytes((0xD0, n))
        elif n < 0x10000:
            out.append(0xD1)
            out += struct.pack(">H", n)
        else:
            out.append(0xD2)
            out += struct.pack(">I", n)
        out += b
    elif isinstance(v, (bytes, bytearray)):
        n = len(v)
        if n < 0x100:
            out += bytes((0xCC, n))
        elif n < 0x10000:
            out.append(0xCD)
            out += struct.pack(">H", n)
        else:
            out.append(0xCE)
            out += struct.pack(">I", n)
        out += v
    elif isinstance(v, (list, tuple)):
        n = len(v)
        if n < 0x10:
            out.append(0x90 + n)
        elif n < 0x100:
            out += bytes((0xD4, n))
        elif n < 0x10000:
            out.append(0xD5)
            out += struct.pack(">H", n)
        else:
            out.append(0xD6)
            out += struct.pack(">I", n)
        for item in v:
            _pack_into(out, item)
    elif isinstance(v, dict):
        n = len(v)
        if n < 0x10:
            out.append(0xA0 + n)
        elif n < 0x100:
            out += bytes((0xD8, n))
        elif n < 0x10000:
            out.append(0xD9)
            out += struct.pack(">H", n)
        else:
            out.append(0xDA)
            out += struct.pack(">I", n)
        for k, item in v.items():
            _pack_into(out, str(k))
            _pack_into(out, item)
    elif isinstance(v, Structure):
        n = len(v.fields)
        if n < 0x10:
            out.append(0xB0 + n)
        else:
            raise PackStreamError("struct too large")
        out.append(v.tag)
        for f in v.fields:
            _pack_into(out, f)
    else:
        raise PackStreamError(f"cannot pack {type(v).__name__}")


def _pack_int(out: bytearray, v: int) -> None:
    if -16 <= v < 128:
        out += struct.pack(">b", v)
    elif -128 <= v < 128:
        out.append(0xC8)
        out += struct.pack(">b", v)
    elif -32768 <= v < 32768:
        out.append(0xC9)
        out += struct.pack(">h", v)
    elif -2147483648 <= v < 2147483648:
        out.append(0xCA)
        out += struct.pack(">i", v)
    else:
        out.append(0xCB)
        out += struct.pack(">q", v)


# -------------------------------------------------------------- unpacking
class Unpacker:
    def __init__(self, data: bytes, offset: int = 0):
        self.data = data
        self.i = offset

    def _take(self, n: int) -> bytes:
        if self.i + n > len(self.data):
            raise PackStreamError("truncated data")
        b = self.data[self.i:self.i + n]
        self.i += n
        return b

    def unpack(self) -> Any:
        m = self._take(1)[0]
        if m <= 0x7F:
            return m
        if m >= 0xF0:
            return m - 0x100
        if 0x80 <= m <= 0x8F:
            return self._take(m & 0x0F).decode("utf-8")
        if 0x90 <= m <= 0x9F:
            return [self.unpack() for _ in range(m & 0x0F)]
        if 0xA0 <= m <= 0xAF:
            return {self.unpack(): self.unpack() for _ in range(m & 0x0F)}
        if 0xB0 <= m <= 0xBF:
            n = m & 0x0F
            tag = self._take(1)[0]
            return Structure(tag, [self.unpack() for _ in range(n)])
        if m == 0xC0:
            return None
        if m == 0xC1:
            return struct.unpack(">d", self._take(8))[0]
        if m == 0xC2:
            return False
        if m == 0xC3:
            return True
        if m == 0xC8:
            return struct.unpack(">b", self._take(1))[0]
        if m == 0xC9:
            return struct.unpack(">h", self._take(2))[0]
        if m == 0xCA:
            return struct.unpack(">i", self._take(4))[0]
        if m == 0xCB:
            return struct.unpack(">q", self._take(8))[0]
        if m == 0xCC:
            return bytes(self._take(self._take(1)[0]))
        if m == 0xCD:
            return bytes(self._take(struct.unpack(">H", self._take(2))[0]))
        if m == 0xCE:
            return bytes(self._take(struct.unpack(">I", self._take(4))[0]))
        if m == 0xD0:
            return self._take(self._take(1)[0]).decode("utf-8")
        if m == 0xD1:
            return self._take(struct.unpack(">H", self._take(2))[0]).decode("utf-8")
        if m == 0xD2:
            return self._take(struct.unpack(">I", self._take(4))[0]).decode("utf-8")
        if m == 0xD4:
            return [self.unpack() for _ in range(self._take(1)[0])]
        if m == 0xD5:
            return [self.unpack() for _ in range(struct.unpack(">H", self._take(2))[0])]
        if m == 0xD6:
            return [self.unpack() for _ in range(struct.unpack(">I", self._take(4))[0])]
        if m == 0xD8:
            return {self.unpack(): self.unpack() for _ in range(self._take(1)[0])}
        if m == 0xD9:
            return {self.unpack(): self.unpack()
                    for _ in range(struct.unpack(">H", self._take(2))[0])}
        if m == 0xDA:
            return {self.unpack(): self.unpack()
                    for _ in range(struct.unpack(">I", self._take(4))[0])}
        if m in (0xDC, 0xDD):
            n = self._take(1)[0] if m == 0xDC else struct.unpack(">H", self._take(2))[0]
            tag = self._take(1)[0]
            return Structure(tag, [self.unpack() for _ in range(n)])
        raise PackStreamError(f"unknown marker 0x{m:02x}")


def unpack(data: bytes) -> Any:
    # note: the native decoder exists (_C.ps_unpack) but loses to the
    # python one on typical small Bolt messages because every nested
    # Structure requires a callback into python; encode is native.
    return Unpacker(data).unpack()


def unpack_py(data: bytes) -> Any:
    """Pure-python unpacker (codec-test oracle)."""
    return Unpacker(data).unpack()


# ----------------------------------------------------- Bolt graph structs
NODE_TAG = 0x4E
REL_TAG = 0x52
UNBOUND_REL_TAG = 0x72
PATH_TAG = 0x50


class IdMap:
    """Stable string-id <-> int-id mapping for Bolt's integer entity ids."""

    def __init__(self):
        self._s2i: Dict[str, int] = {}
        self._i2s: Dict[int, str] = {}

    def to_int(self, s: str) -> int:
        i = self._s2i.get(s)
        if i is None:
            i = len(self._s2i) + 1
            self._s2i[s] = i
            self._i2s[i] = s
        return i

    def to_str(self, i: int) -> str:
        return self._i2s.get(i, str(i))


# ---- temporal structures (Bolt 4.4/5.x PackStream spec) ----
DATE_TAG = 0x44            # 'D' days since epoch
TIME_TAG = 0x54            # 'T' nanos-of-day + tz offset seconds
LOCAL_TIME_TAG = 0x74      # 't'
DATETIME_TAG = 0x49        # 'I' UTC epoch seconds + nanos + offset (Bolt 5)
DATETIME_LEGACY_TAG = 0x46  # 'F' local-epoch seconds variant (Bolt 4)
LOCAL_DATETIME_TAG = 0x64  # 'd'
DURATION_TAG = 0x45        # 'E' months, days, seconds, nanoseconds


def temporal_from_struct(s: "Structure"):
    """Inverse of temporal_struct: Bolt temporal structures (as sent in
    driver parameters) -> cypher temporal values. Returns None if the
    tag is not temporal."""
    import datetime as _dt

    from ..cypher import temporal as _tp

    if s.tag == DURATION_TAG:
        months, days, secs, nanos = s.fields
        return _tp.CypherDuration(months, days, secs, nanos)
    if s.tag == DATE_TAG:
        return _tp.CypherDate(_dt.date(1970, 1, 1) +
                              _dt.timedelta(days=s.fields[0]))
    if s.tag == LOCAL_DATETIME_TAG:
        secs, nanos = s.fields
        return _tp.CypherDateTime(
            _dt.datetime.utcfromtimestamp(secs) +
            _dt.timedelta(microseconds=nanos // 1000))
    if s.tag == DATETIME_TAG:
        secs, nanos, off = s.fields
        tz = _dt.timezone(_dt.timedelta(seconds=off))
        return _tp.CypherDateTime(_dt.datetime.fromtimestamp(
            secs + nanos / 1e9, tz))
    if s.tag == DATETIME_LEGACY_TAG:
        secs, nanos, off = s.fields
        tz = _dt.timezone(_dt.timedelta(seconds=off))
        return _tp.CypherDateTime(_dt.datetime.fromtimestamp(
            secs - off + nanos / 1e9, tz))
    if s.tag == LOCAL_TIME_TAG:
        nanos = s.fields[0]
        us = nanos // 1000
        return _tp.make_time(_dt.time((us // 3600000000) % 24,
                                      (us // 60000000) % 60,
                                      (us // 1000000) % 60,
                                      us % 1000000), local=True)
    if s.tag == TIME_TAG:
        nanos, off = s.fields
        us = nanos // 1000
        t = _dt.time((us // 3600000000) % 24, (us // 60000000) % 60,
                     (us // 1000000) % 60, us % 1000000,
                     tzinfo=_dt.timezone(_dt.timedelta(seconds=off)))
        return _tp.CypherTime(t)
    return None


POINT2D_TAG = 0x58   # 'X'
POINT3D_TAG = 0x59   # 'Y'


def point_struct(v):
    """CypherPoint -> Bolt Point2D/Point3D structure (srid, x, y[, z]).
    SRIDs per Neo4j: 7203/9157 cartesian 2D/3D, 4326/4979 wgs-84."""
    from ..cypher.functions import CypherPoint
    if not isinstance(v, CypherPoint):
        return None
    wgs = v.crs == "wgs-84"
    if v.z is None:
        return Structure(POINT2D_TAG, [4326 if wgs else 7203,
                                       float(v.x), float(v.y)])
    return Structure(POINT3D_TAG, [4979 if wgs else 9157,
                                   float(v.x), float(v.y), float(v.z)])


def point_from_struct(st):
    if not isinstance(st, Structure) or st.tag not in (POINT2D_TAG,
                                                       POINT3D_TAG):
        return None
    from ..cypher.functions import CypherPoint
    srid = st.fields[0]
    wgs = srid in (4326, 4979)
    if wgs:
        src = {"longitude": st.fields[1], "latitude": st.fields[2]}
    else:
        src = {"x": st.fields[1], "y": st.fields[2]}
    if st.tag == POINT3D_TAG:
        src["z"] = st.fields[3]
    return CypherPoint(src)


def temporal_struct(v, bolt5: bool = False):
    """Convert a cypher temporal value to its Bolt structure (or None)."""
    import datetime as _dt

    from ..cypher import temporal as _tp

    if isinstance(v, _tp.CypherDuration):
        return Structure(DURATION_TAG,
                         [v.months, v.days, v.seconds, v.nanoseconds])
    if isinstance(v, _tp.CypherDate):
        days = (v.date - _dt.date(1970, 1, 1)).days
        return Structure(DATE_TAG, [days])
    if isinstance(v, _tp.CypherDateTime):
        dt = v._v
        if dt.tzinfo is None:
            epoch = int(dt.replace(tzinfo=_dt.timezone.utc).timestamp())
            return Structure(LOCAL_DATETIME_TAG,
                             [epoch, dt.microsecond * 1000])
        off = int(dt.utcoffset().total_seconds())
        secs = int(dt.timestamp())
        if bolt5:
            return Structure(DATETIME_TAG, [secs, dt.microsecond * 1000, off])
        return Structure(DATETIME_LEGACY_TAG,
                         [secs + off, dt.microsecond * 1000, off])
    if isinstance(v, _tp.CypherTime):
        dt = v._v
        nanos = ((dt.hour * 3600 + dt.minute * 60 + dt.second) * 1_000_000
                 + dt.microsecond) * 1000
        if dt.tzinfo is None:
            return Structure(LOCAL_TIME_TAG, [nanos])
        return Structure(TIME_TAG,
                         [nanos, int(dt.utcoffset().total_seconds())])
    return None


def node_struct(node, ids: IdMap, bolt5: bool = False) -> Structure:
    fields = [ids.to_int(node.id), list(node.labels), dict(node.properties)]
    if bolt5:
        fields.append(node.id)
    return Structure(NODE_TAG, fields)


def rel_struct(edge, ids: IdMap, bolt5: bool = False) -> Structure:
    fields = [ids.to_int(edge.id), ids.to_int(edge.start_node),
              ids.to_int(edge.end_node), edge.type, dict(edge.properties)]
    if bolt5:
        fields += [edge.id, edge.start_node, edge.end_node]
    return Structure(REL_TAG, fields)


def path_struct(path, ids: IdMap, bolt5: bool = False) -> Structure:
    nodes = [node_struct(n, ids, bolt5) for n in path.nodes]
    rels = []
    for e in path.edges:
        f = [ids.to_int(e.id), e.type, dict(e.properties)]
        if bolt5:
            f.append(e.id)
        rels.append(Structure(UNBOUND_REL_TAG, f))
    # sequence: alternating rel index (1-based, negative=reversed), node index
    seq = []
    for i, e in enumerate(path.edges):
        prev = path.nodes[i]
        sign = 1 if e.start_node == prev.id else -1
        seq.append(sign * (i + 1))
        seq.append(i + 1)
    return Structure(PATH_TAG, [nodes, rels, seq])
