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
    raise TypeError(f"cannot serialize {type(obj).__name__}")


def object_hook(m):
    """msgpack `object_hook`: tagged maps -> temporal values."""
    tag = m.get("__t__") if isinstance(m, dict) else None
    if tag is None:
        return m
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
