"""Qdrant-compatible REST API (collections + points subset).

Parity: reference pkg/qdrantgrpc implements the official Qdrant v1.16
gRPC contract (collections/points/snapshots) backed by vector spaces.
Here the same surface is exposed over Qdrant's REST dialect (same JSON
shapes the Qdrant SDKs use for HTTP), backed by EmbeddingIndex shards.
Embedding ownership rule (reference COMPAT.md:12-14): points bring their
own vectors; NornicDB does not auto-embed through this API.
"""

from __future__ import annotations

import threading
import time
from typing import Any, Dict, List, Optional

from fastapi import APIRouter, HTTPException

from ..search.embedding_index import EmbeddingIndex
from ..search.vectorspace import COSINE, GLOBAL, VectorSpace


class _Collection:
    def __init__(self, name: str, size: int, distance: str):
        self.name = name
        self.size = size
        self.distance = distance
        self.index = EmbeddingIndex(size, device="cpu")
        self.payloads: Dict[str, dict] = {}
        self.vectors: Dict[str, list] = {}
        self.created = time.time()


class QdrantRegistry:
    def __init__(self, db_name: str = "neo4j"):
        self._lock = threading.Lock()
        self.collections: Dict[str, _Collection] = {}
        self.db_name = db_name

    def create(self, name, size, distance):
        with self._lock:
            if name in self.collections:
                raise KeyError(name)
            self.collections[name] = _Collection(name, size, distance)
            GLOBAL.register(VectorSpace(self.db_name, "qdrant", name, size,
                                        distance.lower()))

    def get(self, name) -> _Collection:
        c = self.collections.get(name)
        if c is None:
            raise KeyError(name)
        return c


def qdrant_router(registry: QdrantRegistry = None) -> APIRouter:
    reg = registry or QdrantRegistry()
    r = APIRouter()

    def ok(result):
        return {"result": result, "status": "ok", "time": 0.0}

    @r.get("/collections")
    def list_collections():
        return ok({"collections": [{"name": n} for n in sorted(reg.collections)]})

    @r.put("/collections/{name}")
    def create_collection(name: str, body: Dict[str, Any]):
        vectors = body.get("vectors", {})
        size = vectors.get("size")
        distance = vectors.get("distance", "Cosine")
        if not size:
            raise HTTPException(400, "vectors.size required")
        try:
            reg.create(name, int(size), distance)
        except KeyError:
            raise HTTPException(409, f"collection {name} exists")
        return ok(True)

    @r.get("/collections/{name}")
    def get_collection(name: str):
        try:
            c = reg.get(name)
        except KeyError:
            raise HTTPException(404, "not found")
        return ok({
            "status": "green",
            "vectors_count": len(c.index),
            "points_count": len(c.index),
            "config": {"params": {"vectors": {"size": c.size,
                                              "distance": c.distance}}},
        })

    @r.delete("/collections/{name}")
    def delete_collection(name: str):
        reg.collections.pop(name, None)
        return ok(True)

    @r.put("/collections/{name}/points")
    def upsert_points(name: str, body: Dict[str, Any]):
        try:
            c = reg.get(name)
        except KeyError:
            raise HTTPException(404, "not found")
        pts = body.get("points", [])
        ids, vecs = [], []
        for p in pts:
            pid = str(p["id"])
            vec = p.get("vector")
            if vec is None or len(vec) != c.size:
                raise HTTPException(400, f"point {pid}: vector of size {c.size} required")
            ids.append(pid)
            vecs.append(vec)
            c.payloads[pid] = p.get("payload") or {}
            c.vectors[pid] = list(vec)
        if ids:
            c.index.add_batch(ids, vecs)
        return ok({"operation_id": 0, "status": "completed"})

    @r.post("/collections/{name}/points/search")
    def search_points(name: str, body: Dict[str, Any]):
        try:
            c = reg.get(name)
        except KeyError:
            raise HTTPException(404, "not found")
        vec = body.get("vector")
        if vec is None:
            raise HTTPException(400, "vector required")
        limit = int(body.get("limit", 10))
        with_payload = body.get("with_payload", True)
        hits = c.index.search(vec, limit)
        out = []
        for pid, score in hits:
            item = {"id": _maybe_int(pid), "version": 0, "score": score}
            if with_payload:
                item["payload"] = c.payloads.get(pid, {})
            if body.get("with_vector"):
                item["vector"] = c.vectors.get(pid)
            out.append(item)
        return ok(out)

    @r.post("/collections/{name}/points/scroll")
    def scroll_points(name: str, body: Dict[str, Any] = None):
        body = body or {}
        try:
            c = reg.get(name)
        except KeyError:
            raise HTTPException(404, "not found")
        limit = int(body.get("limit", 10))
        offset = body.get("offset")
        ids = sorted(c.payloads.keys())
        if offset is not None:
            offset = str(offset)
            try:
                start = ids.index(offset)
            except ValueError:
                start = 0
        else:
            start = 0
        page = ids[start:start + limit]
        nxt = ids[start + limit] if start + limit < len(ids) else None
        return ok({
            "points": [{"id": _maybe_int(i), "payload": c.payloads[i]}
                       for i in page],
            "next_page_offset": _maybe_int(nxt) if nxt else None,
        })

    @r.post("/collections/{name}/points/delete")
    def delete_points(name: str, body: Dict[str, Any]):
        try:
            c = reg.get(name)
        except KeyError:
            raise HTTPException(404, "not found")
        for pid in body.get("points", []):
            pid = str(pid)
            c.index.remove(pid)
            c.payloads.pop(pid, None)
            c.vectors.pop(pid, None)
        return ok({"operation_id": 0, "status": "completed"})

    @r.post("/collections/{name}/points")
    def get_points(name: str, body: Dict[str, Any]):
        try:
            c = reg.get(name)
        except KeyError:
            raise HTTPException(404, "not found")
        out = []
        for pid in body.get("ids", []):
            pid = str(pid)
            if pid in c.payloads:
                out.append({"id": _maybe_int(pid),
                            "payload": c.payloads.get(pid, {}),
                            "vector": c.vectors.get(pid)})
        return ok(out)


    # ---- payload ops (reference qdrantgrpc SetPayload/OverwritePayload/
    # DeletePayload/ClearPayload) ----
    def _coll(name):
        try:
            return reg.get(name)
        except KeyError:
            raise HTTPException(404, "not found")

    def _sel_ids(c, body):
        if "points" in body:
            return [str(p) for p in body["points"]]
        flt = body.get("filter")
        if flt:
            return [pid for pid in c.payloads if _matches(c.payloads[pid], flt)]
        return list(c.payloads)

    @r.post("/collections/{name}/points/payload")
    def set_payload(name: str, body: Dict[str, Any]):
        c = _coll(name)
        for pid in _sel_ids(c, body):
            if pid in c.payloads:
                c.payloads[pid].update(body.get("payload", {}))
        return ok({"operation_id": 0, "status": "completed"})

    @r.put("/collections/{name}/points/payload")
    def overwrite_payload(name: str, body: Dict[str, Any]):
        c = _coll(name)
        for pid in _sel_ids(c, body):
            if pid in c.payloads:
                c.payloads[pid] = dict(body.get("payload", {}))
        return ok({"operation_id": 0, "status": "completed"})

    @r.post("/collections/{name}/points/payload/delete")
    def delete_payload(name: str, body: Dict[str, Any]):
        c = _coll(name)
        keys = body.get("keys", [])
        for pid in _sel_ids(c, body):
            for k in keys:
                c.payloads.get(pid, {}).pop(k, None)
        return ok({"operation_id": 0, "status": "completed"})

    @r.post("/collections/{name}/points/payload/clear")
    def clear_payload(name: str, body: Dict[str, Any] = None):
        c = _coll(name)
        for pid in _sel_ids(c, body or {}):
            c.payloads[pid] = {}
        return ok({"operation_id": 0, "status": "completed"})

    # ---- count / exists / query (qdrant >=1.10 universal query) ----
    @r.post("/collections/{name}/points/count")
    def count_points(name: str, body: Dict[str, Any] = None):
        c = _coll(name)
        flt = (body or {}).get("filter")
        if flt:
            n = sum(1 for pid in c.payloads
                    if _matches(c.payloads[pid], flt))
        else:
            n = len(c.payloads)
        return ok({"count": n})

    @r.get("/collections/{name}/exists")
    def collection_exists(name: str):
        return ok({"exists": name in reg.collections})

    @r.post("/collections/{name}/points/query")
    def query_points(name: str, body: Dict[str, Any]):
        c = _coll(name)
        q = body.get("query")
        limit = int(body.get("limit", 10))
        if isinstance(q, dict) and "nearest" in q:
            q = q["nearest"]
        if isinstance(q, list):            # vector query
            hits = c.index.search(q, limit * 4)
        elif q is not None and str(q) in c.vectors:  # by point id
            hits = c.index.search(c.vectors[str(q)], limit * 4 + 1)
            hits = [(p, s) for p, s in hits if p != str(q)]
        else:
            hits = [(pid, 1.0) for pid in sorted(c.payloads)][:limit * 4]
        flt = body.get("filter")
        out = []
        for pid, score in hits:
            if flt and not _matches(c.payloads.get(pid, {}), flt):
                continue
            item = {"id": _maybe_int(pid), "score": score}
            if body.get("with_payload", True):
                item["payload"] = c.payloads.get(pid, {})
            out.append(item)
            if len(out) >= limit:
                break
        return ok({"points": out})

    # ---- vector update/delete ----
    @r.put("/collections/{name}/points/vectors")
    def update_vectors(name: str, body: Dict[str, Any]):
        c = _coll(name)
        for p in body.get("points", []):
            pid = str(p["id"])
            vec = p.get("vector")
            if vec is not None and pid in c.payloads:
                c.vectors[pid] = list(vec)
                c.index.remove(pid)
                c.index.add_batch([pid], [vec])
        return ok({"operation_id": 0, "status": "completed"})

    @r.post("/collections/{name}/points/vectors/delete")
    def delete_vectors(name: str, body: Dict[str, Any]):
        c = _coll(name)
        for pid in body.get("points", []):
            c.index.remove(str(pid))
        return ok({"operation_id": 0, "status": "completed"})

    return r


def _matches(payload: Dict[str, Any], flt: Dict[str, Any]) -> bool:
    """Qdrant filter subset: must / must_not / should with match/range."""
    def cond_ok(cond):
        key = cond.get("key")
        if "match" in cond:
            want = cond["match"].get("value", cond["match"].get("text"))
            return payload.get(key) == want
        if "range" in cond:
            v = payload.get(key)
            if not isinstance(v, (int, float)):
                return False
            rg = cond["range"]
            return all([
                v >= rg["gte"] if "gte" in rg else True,
                v > rg["gt"] if "gt" in rg else True,
                v <= rg["lte"] if "lte" in rg else True,
                v < rg["lt"] if "lt" in rg else True])
        return True

    for cond in flt.get("must", []):
        if not cond_ok(cond):
            return False
    for cond in flt.get("must_not", []):
        if cond_ok(cond):
            return False
    should = flt.get("should", [])
    if should and not any(cond_ok(c) for c in should):
        return False
    return True


def _maybe_int(pid):
    try:
        return int(pid)
    except (ValueError, TypeError):
        return pid
