"""Msgpack codec hooks for non-primitive property values (temporal
types). Used by the WAL and snapshots so `CREATE (n {d: date(...)})`
persists and replays (Neo4j stores temporals natively; reference
pkg/storage serializes them through its property codec)."""

from __future__ import annotations


def default(obj):
    """msgpack `default=` hook: temporal values -> tagged maps."""
    from ..cypher import temporal as tp
    if isinstance(obj, tp.CypherDate):
        return {"__t__": "date", "v": str(obj)}
    if isinstance(obj, tp.CypherDateTime):
        return {"__t__": "datetime", "v": obj._v.isoformat()}
    if isinstance(obj, tp.CypherTime):
        return {"__t__": "time", "v": str(obj)}
    if isinstance(obj, tp.CypherDuration):
        return {"__t__": "duration", "v": str(obj)}
    from ..cypher.functions import CypherPoint
    if isinstance(obj, CypherPoint):
        d = {"__t__": "point", "x": obj.x, "y": obj.y, "crs": obj.crs}
        if getattr(obj, "z", None) is not None:
            d["z"] = obj.z
        return d
    raise TypeError(f"cannot serialize {type(obj).__name__}")


def object_hook(m):
    """msgpack `object_hook`: tagged maps -> temporal values."""
    tag = m.get("__t__") if isinstance(m, dict) else None
    if tag is None:
        return m
    if tag == "point":
        from ..cypher.functions import CypherPoint
        src = {"x": m["x"], "y": m["y"]}
        if m.get("crs") == "wgs-84":
            src = {"longitude": m["x"], "latitude": m["y"]}
        if m.get("z") is not None:
            src["z"] = m["z"]
        src["crs"] = m.get("crs", "cartesian")
        return CypherPoint(src)
    from ..cypher import temporal as tp
    v = m["v"]
    if tag == "date":
        return tp.make_date(v)
    if tag == "datetime":
        return tp.make_datetime(v)
    if tag == "time":
        return tp.make_time(v)
    if tag == "duration":
        return tp.make_duration(v)
    return m
