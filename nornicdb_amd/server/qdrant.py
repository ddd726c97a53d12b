"""Qdrant-compatible REST API (collections + points subset).

Parity: reference pkg/qdrantgrpc implements the official Qdrant v1.16
gRPC contract (collections/points/snapshots) backed by vector spaces.
Here the same surface is exposed over Qdrant's REST dialect (same JSON
shapes the Qdrant SDKs use for HTTP), backed by EmbeddingIndex shards.
Embedding ownership rule (reference COMPAT.md:12-14): points bring their
own vectors; NornicDB does not auto-embed through this API.
"""

from __future__ import annotations

import os
import tempfile
import threading
import time

import msgpack
from typing import Any, Dict, List, Optional

from fastapi import APIRouter, HTTPException

from ..search.embedding_index import EmbeddingIndex
from ..search.vectorspace import COSINE, GLOBAL, VectorSpace


class _Collection:
    def __init__(self, name: str, size: int, distance: str, eng=None):
        self.name = name
        self.size = size
        self.distance = distance
        self.index = EmbeddingIndex(size, device="cpu")
        self.payloads: Dict[str, dict] = {}
        self.vectors: Dict[str, list] = {}
        self.created = time.time()
        self.eng = eng   # storage engine for durability (reference
                         # registry.go _QdrantCollection/QdrantPoint nodes)

    # ---- durability (write-through to the storage engine) ----
    def _pt_id(self, pid: str) -> str:
        return f"_qdr:{self.name}:{pid}"

    def persist_point(self, pid: str):
        if self.eng is None:
            return
        from ..storage.types import Node
        n = Node(id=self._pt_id(pid),
                 labels=[f"_QdrantPoint_{self.name}"],
                 properties={"payload": self.payloads.get(pid, {}),
                             "vector": self.vectors.get(pid)})
        try:
            self.eng.update_node(n)
        except Exception:
            try:
                self.eng.create_node(n)
            except Exception:
                pass

    def unpersist_point(self, pid: str):
        if self.eng is None:
            return
        try:
            self.eng.delete_node(self._pt_id(pid))
        except Exception:
            pass


class QdrantRegistry:
    def __init__(self, db_name: str = "neo4j", engine=None):
        self._lock = threading.Lock()
        self.collections: Dict[str, _Collection] = {}
        self.db_name = db_name
        self.engine = engine
        if engine is not None:
            self._load()

    def _load(self):
        """Rebuild collections + points from the storage engine
        (reference pkg/qdrantgrpc/registry.go loads _QdrantCollection
        metadata nodes + per-point nodes on boot)."""
        try:
            metas = self.engine.get_nodes_by_label("_QdrantCollection")
        except Exception:
            return
        for m in metas:
            props = m.properties
            name = props.get("name")
            if not name or name in self.collections:
                continue
            c = _Collection(name, int(props.get("size", 0)),
                            props.get("distance", "Cosine"),
                            eng=self.engine)
            self.collections[name] = c
            try:
                GLOBAL.register(VectorSpace(self.db_name, "qdrant", name,
                                            c.size, c.distance.lower()))
            except Exception:
                pass   # already registered by a previous open in-process
            try:
                pts = self.engine.get_nodes_by_label(
                    f"_QdrantPoint_{name}")
            except Exception:
                pts = []
            ids, vecs = [], []
            for pn in pts:
                pid = pn.id.split(":", 2)[2] if pn.id.count(":") >= 2 \
                    else pn.id
                vec = pn.properties.get("vector")
                c.payloads[pid] = pn.properties.get("payload") or {}
                if vec is not None:
                    c.vectors[pid] = list(vec)
                    ids.append(pid)
                    vecs.append(vec)
            if ids:
                c.index.add_batch(ids, vecs)

    def create(self, name, size, distance):
        with self._lock:
            if name in self.collections:
                raise KeyError(name)
            self.collections[name] = _Collection(name, size, distance,
                                                 eng=self.engine)
            GLOBAL.register(VectorSpace(self.db_name, "qdrant", name, size,
                                        distance.lower()))
            if self.engine is not None:
                from ..storage.types import Node
                try:
                    self.engine.create_node(Node(
                        id=f"_qdrcol:{name}", labels=["_QdrantCollection"],
                        properties={"name": name, "size": int(size),
                                    "distance": distance}))
                except Exception:
                    pass

    def drop(self, name: str):
        """Remove a collection and its persisted nodes."""
        with self._lock:
            c = self.collections.pop(name, None)
        if c is not None and self.engine is not None:
            try:
                self.engine.delete_node(f"_qdrcol:{name}")
            except Exception:
                pass
            for pid in list(c.payloads):
                c.unpersist_point(pid)

    def get(self, name) -> _Collection:
        c = self.collections.get(name)
        if c is None:
            raise KeyError(name)
        return c

    # ---- snapshots (reference pkg/qdrantgrpc/snapshots_service.go) ----
    @property
    def snapshot_dir(self) -> str:
        d = getattr(self, "_snapshot_dir", None)
        if d is None:
            d = os.environ.get("NORNICDB_QDRANT_SNAPSHOT_DIR",
                               os.path.join(tempfile.gettempdir(),
                                            "nornicdb-qdrant-snapshots"))
            self._snapshot_dir = d
        return d

    def set_snapshot_dir(self, d: str):
        self._snapshot_dir = d

    def _snap_coll_dir(self, name: str) -> str:
        return os.path.join(self.snapshot_dir, "collections", name)

    def snapshot_create(self, name: str) -> Dict[str, Any]:
        c = self.get(name)
        d = self._snap_coll_dir(name)
        os.makedirs(d, exist_ok=True)
        ts = time.time()
        snap_name = f"{name}-{int(ts * 1e9)}.snapshot"
        blob = msgpack.packb({
            "version": "qdrant-compat-1.0",
            "collection": name,
            "config": {"size": c.size, "distance": c.distance},
            "points": {pid: {"vector": c.vectors.get(pid),
                             "payload": c.payloads.get(pid, {})}
                       for pid in c.payloads},
            "timestamp": ts,
        }, use_bin_type=True)
        path = os.path.join(d, snap_name)
        with open(path, "wb") as f:
            f.write(blob)
        return {"name": snap_name, "creation_time": ts,
                "size": os.path.getsize(path)}

    def snapshot_list(self, name: str) -> List[Dict[str, Any]]:
        self.get(name)
        d = self._snap_coll_dir(name)
        out = []
        if os.path.isdir(d):
            for fn in sorted(os.listdir(d)):
                if fn.endswith(".snapshot"):
                    p = os.path.join(d, fn)
                    st = os.stat(p)
                    out.append({"name": fn, "creation_time": st.st_mtime,
                                "size": st.st_size})
        return out

    def snapshot_path(self, name: str, snap: str) -> str:
        p = os.path.join(self._snap_coll_dir(name), os.path.basename(snap))
        if not os.path.exists(p):
            raise KeyError(snap)
        return p

    def snapshot_delete(self, name: str, snap: str):
        os.remove(self.snapshot_path(name, snap))

    def snapshot_recover(self, name: str, location: str):
        """Rebuild collection `name` from a snapshot file (file:// URI or
        local path; reference recover semantics: replace contents)."""
        path = location[7:] if location.startswith("file://") else location
        with open(path, "rb") as f:
            data = msgpack.unpackb(f.read(), raw=False)
        cfg = data["config"]
        self.drop(name)
        self.create(name, int(cfg["size"]), cfg["distance"])
        c = self.get(name)
        for pid, rec in data["points"].items():
            vec = rec.get("vector")
            if vec is not None:
                c.index.add(pid, vec)
                c.vectors[pid] = list(vec)
            c.payloads[pid] = rec.get("payload") or {}
            c.persist_point(pid)


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
        reg.drop(name)
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
        for pid in ids:
            c.persist_point(pid)
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
            c.unpersist_point(pid)
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
                c.persist_point(pid)
        return ok({"operation_id": 0, "status": "completed"})

    @r.put("/collections/{name}/points/payload")
    def overwrite_payload(name: str, body: Dict[str, Any]):
        c = _coll(name)
        for pid in _sel_ids(c, body):
            if pid in c.payloads:
                c.payloads[pid] = dict(body.get("payload", {}))
                c.persist_point(pid)
        return ok({"operation_id": 0, "status": "completed"})

    @r.post("/collections/{name}/points/payload/delete")
    def delete_payload(name: str, body: Dict[str, Any]):
        c = _coll(name)
        keys = body.get("keys", [])
        for pid in _sel_ids(c, body):
            for k in keys:
                c.payloads.get(pid, {}).pop(k, None)
            c.persist_point(pid)
        return ok({"operation_id": 0, "status": "completed"})

    @r.post("/collections/{name}/points/payload/clear")
    def clear_payload(name: str, body: Dict[str, Any] = None):
        c = _coll(name)
        for pid in _sel_ids(c, body or {}):
            c.payloads[pid] = {}
            c.persist_point(pid)
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


    # ---- snapshots (reference snapshots_service.go; Qdrant REST parity) ----
    @r.post("/collections/{name}/snapshots")
    @r.put("/collections/{name}/snapshots")
    def create_snapshot(name: str):
        try:
            return ok(reg.snapshot_create(name))
        except KeyError:
            raise HTTPException(404, f"collection {name} not found")

    @r.get("/collections/{name}/snapshots")
    def list_snapshots(name: str):
        try:
            return ok(reg.snapshot_list(name))
        except KeyError:
            raise HTTPException(404, f"collection {name} not found")

    @r.delete("/collections/{name}/snapshots/{snap}")
    def delete_snapshot(name: str, snap: str):
        try:
            reg.snapshot_delete(name, snap)
            return ok(True)
        except KeyError:
            raise HTTPException(404, "snapshot not found")

    @r.get("/collections/{name}/snapshots/{snap}")
    def download_snapshot(name: str, snap: str):
        from fastapi.responses import FileResponse
        try:
            return FileResponse(reg.snapshot_path(name, snap),
                                media_type="application/octet-stream",
                                filename=snap)
        except KeyError:
            raise HTTPException(404, "snapshot not found")

    @r.put("/collections/{name}/snapshots/recover")
    def recover_snapshot(name: str, body: Dict[str, Any]):
        loc = (body or {}).get("location", "")
        try:
            reg.snapshot_recover(name, loc)
            return ok(True)
        except (KeyError, OSError) as e:
            raise HTTPException(404, f"recover failed: {e}")

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
