"""Qdrant-compatible gRPC endpoint.

Parity: reference pkg/qdrantgrpc (collections_service.go,
points_service.go, server.go) which serves the official
`qdrant.Collections` / `qdrant.Points` gRPC services over NornicDB
storage. protoc is unavailable offline, so the required subset of the
public Qdrant schema (collections.proto, points.proto,
json_with_int.proto — service `qdrant.Collections`, `qdrant.Points`)
is reconstructed as dynamic protobuf descriptors with the upstream
field numbers; the same `QdrantRegistry` that backs the REST compat
layer (server/qdrant.py) backs this endpoint, so REST and gRPC views
of a collection are identical.

Implemented RPCs (the reference's COMPAT.md core set):
  Collections: Create, Get, List, Delete, CollectionExists
  Snapshots: Create, List, Delete (REST adds download + recover)
  Points: Upsert, Get, Delete, Count, Search, SearchBatch, Scroll,
          SetPayload, OverwritePayload, DeletePayload, ClearPayload
"""

from __future__ import annotations

import time
from typing import Dict, List

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

from .qdrant import QdrantRegistry, _matches, _maybe_int

_T = descriptor_pb2.FieldDescriptorProto
_PKG = "qdrant"


def _f(name, number, ftype, label=_T.LABEL_OPTIONAL, type_name=None,
       oneof=None, opt=False):
    f = _T(name=name, number=number, type=ftype, label=label)
    if type_name:
        f.type_name = type_name
    if oneof is not None:
        f.oneof_index = oneof
    if opt:
        f.proto3_optional = True
    return f


def _msg(fd, name, fields, oneofs=()):
    m = fd.message_type.add()
    m.name = name
    for o in oneofs:
        m.oneof_decl.add().name = o
    for f in fields:
        m.field.append(f)
    return m


def _build_pool():
    pool = descriptor_pool.DescriptorPool()
    fd = descriptor_pb2.FileDescriptorProto(
        name="qdrant_subset.proto", package=_PKG, syntax="proto3")

    # ---- json_with_int.proto: Qdrant's Value (int-preserving JSON) ----
    _msg(fd, "Value", [
        _f("null_value", 1, _T.TYPE_INT32, oneof=0),
        _f("double_value", 2, _T.TYPE_DOUBLE, oneof=0),
        _f("integer_value", 3, _T.TYPE_INT64, oneof=0),
        _f("string_value", 4, _T.TYPE_STRING, oneof=0),
        _f("bool_value", 5, _T.TYPE_BOOL, oneof=0),
        _f("struct_value", 6, _T.TYPE_MESSAGE, type_name=".qdrant.Struct",
           oneof=0),
        _f("list_value", 7, _T.TYPE_MESSAGE, type_name=".qdrant.ListValue",
           oneof=0),
    ], oneofs=("kind",))
    st = _msg(fd, "Struct", [])
    entry = st.nested_type.add()
    entry.name = "FieldsEntry"
    entry.options.map_entry = True
    entry.field.append(_f("key", 1, _T.TYPE_STRING))
    entry.field.append(_f("value", 2, _T.TYPE_MESSAGE,
                          type_name=".qdrant.Value"))
    st.field.append(_f("fields", 1, _T.TYPE_MESSAGE, label=_T.LABEL_REPEATED,
                       type_name=".qdrant.Struct.FieldsEntry"))
    _msg(fd, "ListValue", [
        _f("values", 1, _T.TYPE_MESSAGE, label=_T.LABEL_REPEATED,
           type_name=".qdrant.Value")])

    def payload_map(parent, number=3, name="payload"):
        e = parent.nested_type.add()
        e.name = "PayloadEntry"
        e.options.map_entry = True
        e.field.append(_f("key", 1, _T.TYPE_STRING))
        e.field.append(_f("value", 2, _T.TYPE_MESSAGE,
                          type_name=".qdrant.Value"))
        parent.field.append(_f(name, number, _T.TYPE_MESSAGE,
                               label=_T.LABEL_REPEATED,
                               type_name=f".qdrant.{parent.name}.PayloadEntry"))

    # ---- collections.proto subset ----
    _msg(fd, "VectorParams", [
        _f("size", 1, _T.TYPE_UINT64),
        _f("distance", 2, _T.TYPE_INT32),  # enum Distance: Cosine=1,Euclid=2,Dot=3
    ])
    _msg(fd, "VectorsConfig", [
        _f("params", 1, _T.TYPE_MESSAGE, type_name=".qdrant.VectorParams",
           oneof=0)], oneofs=("config",))
    _msg(fd, "CreateCollection", [
        _f("collection_name", 1, _T.TYPE_STRING),
        _f("vectors_config", 10, _T.TYPE_MESSAGE,
           type_name=".qdrant.VectorsConfig"),
    ])
    _msg(fd, "CollectionOperationResponse", [
        _f("result", 1, _T.TYPE_BOOL), _f("time", 2, _T.TYPE_DOUBLE)])
    _msg(fd, "GetCollectionInfoRequest", [
        _f("collection_name", 1, _T.TYPE_STRING)])
    _msg(fd, "CollectionParams", [
        _f("vectors_config", 5, _T.TYPE_MESSAGE,
           type_name=".qdrant.VectorsConfig")])
    _msg(fd, "CollectionConfig", [
        _f("params", 1, _T.TYPE_MESSAGE, type_name=".qdrant.CollectionParams")])
    _msg(fd, "CollectionInfo", [
        _f("status", 1, _T.TYPE_INT32),          # CollectionStatus: Green=1
        _f("vectors_count", 3, _T.TYPE_UINT64, opt=True, oneof=0),
        _f("segments_count", 4, _T.TYPE_UINT64),
        _f("config", 7, _T.TYPE_MESSAGE, type_name=".qdrant.CollectionConfig"),
        _f("points_count", 9, _T.TYPE_UINT64, opt=True, oneof=1),
    ], oneofs=("_vectors_count", "_points_count"))
    _msg(fd, "GetCollectionInfoResponse", [
        _f("result", 1, _T.TYPE_MESSAGE, type_name=".qdrant.CollectionInfo"),
        _f("time", 2, _T.TYPE_DOUBLE)])
    _msg(fd, "ListCollectionsRequest", [])
    _msg(fd, "CollectionDescription", [_f("name", 1, _T.TYPE_STRING)])
    _msg(fd, "ListCollectionsResponse", [
        _f("collections", 1, _T.TYPE_MESSAGE, label=_T.LABEL_REPEATED,
           type_name=".qdrant.CollectionDescription"),
        _f("time", 2, _T.TYPE_DOUBLE)])
    _msg(fd, "DeleteCollection", [_f("collection_name", 1, _T.TYPE_STRING)])
    _msg(fd, "CollectionExistsRequest", [
        _f("collection_name", 1, _T.TYPE_STRING)])
    _msg(fd, "CollectionExists", [_f("exists", 1, _T.TYPE_BOOL)])
    _msg(fd, "CollectionExistsResponse", [
        _f("result", 1, _T.TYPE_MESSAGE, type_name=".qdrant.CollectionExists"),
        _f("time", 2, _T.TYPE_DOUBLE)])

    # ---- points.proto subset ----
    _msg(fd, "PointId", [
        _f("num", 1, _T.TYPE_UINT64, oneof=0),
        _f("uuid", 2, _T.TYPE_STRING, oneof=0)], oneofs=("point_id_options",))
    _msg(fd, "Vector", [
        _f("data", 1, _T.TYPE_FLOAT, label=_T.LABEL_REPEATED)])
    _msg(fd, "Vectors", [
        _f("vector", 1, _T.TYPE_MESSAGE, type_name=".qdrant.Vector",
           oneof=0)], oneofs=("vectors_options",))
    ps = _msg(fd, "PointStruct", [
        _f("id", 1, _T.TYPE_MESSAGE, type_name=".qdrant.PointId"),
        _f("vectors", 4, _T.TYPE_MESSAGE, type_name=".qdrant.Vectors")])
    payload_map(ps)
    _msg(fd, "UpsertPoints", [
        _f("collection_name", 1, _T.TYPE_STRING),
        _f("wait", 2, _T.TYPE_BOOL, opt=True, oneof=0),
        _f("points", 3, _T.TYPE_MESSAGE, label=_T.LABEL_REPEATED,
           type_name=".qdrant.PointStruct")], oneofs=("_wait",))
    _msg(fd, "UpdateResult", [
        _f("operation_id", 1, _T.TYPE_UINT64, opt=True, oneof=0),
        _f("status", 2, _T.TYPE_INT32)],  # UpdateStatus: Completed=2
         oneofs=("_operation_id",))
    _msg(fd, "PointsOperationResponse", [
        _f("result", 1, _T.TYPE_MESSAGE, type_name=".qdrant.UpdateResult"),
        _f("time", 2, _T.TYPE_DOUBLE)])
    _msg(fd, "WithPayloadSelector", [
        _f("enable", 1, _T.TYPE_BOOL, oneof=0)],
         oneofs=("selector_options",))
    _msg(fd, "WithVectorsSelector", [
        _f("enable", 1, _T.TYPE_BOOL, oneof=0)],
         oneofs=("selector_options",))
    _msg(fd, "GetPoints", [
        _f("collection_name", 1, _T.TYPE_STRING),
        _f("ids", 2, _T.TYPE_MESSAGE, label=_T.LABEL_REPEATED,
           type_name=".qdrant.PointId"),
        _f("with_payload", 4, _T.TYPE_MESSAGE,
           type_name=".qdrant.WithPayloadSelector"),
        _f("with_vectors", 5, _T.TYPE_MESSAGE,
           type_name=".qdrant.WithVectorsSelector")])
    rp = _msg(fd, "RetrievedPoint", [
        _f("id", 1, _T.TYPE_MESSAGE, type_name=".qdrant.PointId"),
        _f("vectors", 4, _T.TYPE_MESSAGE, type_name=".qdrant.Vectors")])
    payload_map(rp, number=2)
    _msg(fd, "GetResponse", [
        _f("result", 1, _T.TYPE_MESSAGE, label=_T.LABEL_REPEATED,
           type_name=".qdrant.RetrievedPoint"),
        _f("time", 2, _T.TYPE_DOUBLE)])
    _msg(fd, "PointsIdsList", [
        _f("ids", 1, _T.TYPE_MESSAGE, label=_T.LABEL_REPEATED,
           type_name=".qdrant.PointId")])
    # Filter subset: same JSON shape as the REST layer, carried as a
    # qdrant.Struct for must/must_not/should FieldCondition matches.
    _msg(fd, "Filter", [
        _f("conditions", 1, _T.TYPE_MESSAGE, type_name=".qdrant.Struct")])
    _msg(fd, "PointsSelector", [
        _f("points", 1, _T.TYPE_MESSAGE, type_name=".qdrant.PointsIdsList",
           oneof=0),
        _f("filter", 2, _T.TYPE_MESSAGE, type_name=".qdrant.Filter",
           oneof=0)], oneofs=("points_selector_one_of",))
    _msg(fd, "DeletePoints", [
        _f("collection_name", 1, _T.TYPE_STRING),
        _f("points", 3, _T.TYPE_MESSAGE, type_name=".qdrant.PointsSelector")])
    _msg(fd, "CountPoints", [
        _f("collection_name", 1, _T.TYPE_STRING),
        _f("filter", 2, _T.TYPE_MESSAGE, type_name=".qdrant.Filter")])
    _msg(fd, "CountResult", [_f("count", 1, _T.TYPE_UINT64)])
    _msg(fd, "CountResponse", [
        _f("result", 1, _T.TYPE_MESSAGE, type_name=".qdrant.CountResult"),
        _f("time", 2, _T.TYPE_DOUBLE)])
    _msg(fd, "SearchPoints", [
        _f("collection_name", 1, _T.TYPE_STRING),
        _f("vector", 2, _T.TYPE_FLOAT, label=_T.LABEL_REPEATED),
        _f("filter", 3, _T.TYPE_MESSAGE, type_name=".qdrant.Filter"),
        _f("limit", 4, _T.TYPE_UINT64),
        _f("with_payload", 6, _T.TYPE_MESSAGE,
           type_name=".qdrant.WithPayloadSelector"),
        _f("score_threshold", 8, _T.TYPE_FLOAT, opt=True, oneof=0),
        _f("with_vectors", 10, _T.TYPE_MESSAGE,
           type_name=".qdrant.WithVectorsSelector")],
         oneofs=("_score_threshold",))
    sp = _msg(fd, "ScoredPoint", [
        _f("id", 1, _T.TYPE_MESSAGE, type_name=".qdrant.PointId"),
        _f("score", 3, _T.TYPE_FLOAT),
        _f("version", 5, _T.TYPE_UINT64),
        _f("vectors", 6, _T.TYPE_MESSAGE, type_name=".qdrant.Vectors")])
    payload_map(sp, number=2)
    _msg(fd, "SearchResponse", [
        _f("result", 1, _T.TYPE_MESSAGE, label=_T.LABEL_REPEATED,
           type_name=".qdrant.ScoredPoint"),
        _f("time", 2, _T.TYPE_DOUBLE)])
    _msg(fd, "SearchBatchPoints", [
        _f("collection_name", 1, _T.TYPE_STRING),
        _f("search_points", 2, _T.TYPE_MESSAGE, label=_T.LABEL_REPEATED,
           type_name=".qdrant.SearchPoints")])
    _msg(fd, "BatchResult", [
        _f("result", 1, _T.TYPE_MESSAGE, label=_T.LABEL_REPEATED,
           type_name=".qdrant.ScoredPoint")])
    _msg(fd, "SearchBatchResponse", [
        _f("result", 1, _T.TYPE_MESSAGE, label=_T.LABEL_REPEATED,
           type_name=".qdrant.BatchResult"),
        _f("time", 2, _T.TYPE_DOUBLE)])
    _msg(fd, "ScrollPoints", [
        _f("collection_name", 1, _T.TYPE_STRING),
        _f("filter", 2, _T.TYPE_MESSAGE, type_name=".qdrant.Filter"),
        _f("offset", 3, _T.TYPE_MESSAGE, type_name=".qdrant.PointId"),
        _f("limit", 4, _T.TYPE_UINT32, opt=True, oneof=0),
        _f("with_payload", 6, _T.TYPE_MESSAGE,
           type_name=".qdrant.WithPayloadSelector"),
        _f("with_vectors", 7, _T.TYPE_MESSAGE,
           type_name=".qdrant.WithVectorsSelector")], oneofs=("_limit",))
    _msg(fd, "ScrollResponse", [
        _f("next_page_offset", 1, _T.TYPE_MESSAGE,
           type_name=".qdrant.PointId"),
        _f("result", 2, _T.TYPE_MESSAGE, label=_T.LABEL_REPEATED,
           type_name=".qdrant.RetrievedPoint"),
        _f("time", 3, _T.TYPE_DOUBLE)])
    spp = _msg(fd, "SetPayloadPoints", [
        _f("collection_name", 1, _T.TYPE_STRING),
        _f("points_selector", 5, _T.TYPE_MESSAGE,
           type_name=".qdrant.PointsSelector")])
    payload_map(spp)
    _msg(fd, "DeletePayloadPoints", [
        _f("collection_name", 1, _T.TYPE_STRING),
        _f("keys", 3, _T.TYPE_STRING, label=_T.LABEL_REPEATED),
        _f("points_selector", 5, _T.TYPE_MESSAGE,
           type_name=".qdrant.PointsSelector")])
    _msg(fd, "ClearPayloadPoints", [
        _f("collection_name", 1, _T.TYPE_STRING),
        _f("points", 3, _T.TYPE_MESSAGE, type_name=".qdrant.PointsSelector")])

    # ---- snapshots.proto subset (official field numbers) ----
    from google.protobuf import timestamp_pb2
    ts_fd = descriptor_pb2.FileDescriptorProto()
    timestamp_pb2.DESCRIPTOR.CopyToProto(ts_fd)
    pool.Add(ts_fd)
    fd.dependency.append("google/protobuf/timestamp.proto")
    _msg(fd, "CreateSnapshotRequest", [
        _f("collection_name", 1, _T.TYPE_STRING)])
    _msg(fd, "ListSnapshotsRequest", [
        _f("collection_name", 1, _T.TYPE_STRING)])
    _msg(fd, "DeleteSnapshotRequest", [
        _f("collection_name", 1, _T.TYPE_STRING),
        _f("snapshot_name", 2, _T.TYPE_STRING)])
    _msg(fd, "SnapshotDescription", [
        _f("name", 1, _T.TYPE_STRING),
        _f("creation_time", 2, _T.TYPE_MESSAGE,
           type_name=".google.protobuf.Timestamp"),
        _f("size", 3, _T.TYPE_INT64),
        _f("checksum", 4, _T.TYPE_STRING)])
    _msg(fd, "CreateSnapshotResponse", [
        _f("snapshot_description", 1, _T.TYPE_MESSAGE,
           type_name=".qdrant.SnapshotDescription"),
        _f("time", 2, _T.TYPE_DOUBLE)])
    _msg(fd, "ListSnapshotsResponse", [
        _f("snapshot_descriptions", 1, _T.TYPE_MESSAGE,
           label=_T.LABEL_REPEATED,
           type_name=".qdrant.SnapshotDescription"),
        _f("time", 2, _T.TYPE_DOUBLE)])
    _msg(fd, "DeleteSnapshotResponse", [
        _f("time", 1, _T.TYPE_DOUBLE)])

    pool.Add(fd)
    return pool


_POOL = _build_pool()


def _cls(name):
    return message_factory.GetMessageClass(
        _POOL.FindMessageTypeByName(f"{_PKG}.{name}"))


M = {n: _cls(n) for n in (
    "Value", "Struct", "ListValue", "VectorParams", "VectorsConfig",
    "CreateCollection", "CollectionOperationResponse",
    "GetCollectionInfoRequest", "GetCollectionInfoResponse",
    "ListCollectionsRequest", "ListCollectionsResponse", "DeleteCollection",
    "CollectionExistsRequest", "CollectionExistsResponse", "PointId",
    "Vector", "Vectors", "PointStruct", "UpsertPoints",
    "PointsOperationResponse", "GetPoints", "GetResponse", "DeletePoints",
    "CountPoints", "CountResponse", "SearchPoints", "SearchResponse",
    "SearchBatchPoints", "SearchBatchResponse", "ScrollPoints",
    "ScrollResponse", "SetPayloadPoints", "DeletePayloadPoints",
    "ClearPayloadPoints", "CreateSnapshotRequest", "ListSnapshotsRequest",
    "DeleteSnapshotRequest", "SnapshotDescription",
    "CreateSnapshotResponse", "ListSnapshotsResponse",
    "DeleteSnapshotResponse")}

_DISTANCES = {0: "cosine", 1: "cosine", 2: "euclid", 3: "dot", 4: "manhattan"}
_DIST_NUM = {"cosine": 1, "euclid": 2, "dot": 3, "manhattan": 4}


# ---- Value <-> python ----
def to_value(v) -> "M['Value']":
    m = M["Value"]()
    if v is None:
        m.null_value = 0
    elif isinstance(v, bool):
        m.bool_value = v
    elif isinstance(v, int):
        m.integer_value = v
    elif isinstance(v, float):
        m.double_value = v
    elif isinstance(v, str):
        m.string_value = v
    elif isinstance(v, (list, tuple)):
        m.list_value.values.extend(to_value(x) for x in v)
    elif isinstance(v, dict):
        for k, x in v.items():
            m.struct_value.fields[str(k)].CopyFrom(to_value(x))
    else:
        m.string_value = str(v)
    return m


def from_value(m) -> object:
    kind = m.WhichOneof("kind")
    if kind is None or kind == "null_value":
        return None
    if kind == "list_value":
        return [from_value(x) for x in m.list_value.values]
    if kind == "struct_value":
        return {k: from_value(x) for k, x in m.struct_value.fields.items()}
    return getattr(m, kind)


def _payload_to_py(pmap) -> dict:
    return {k: from_value(v) for k, v in pmap.items()}


def _payload_from_py(pmap, d: dict):
    for k, v in (d or {}).items():
        pmap[str(k)].CopyFrom(to_value(v))


def _pid_str(pid) -> str:
    return pid.uuid if pid.WhichOneof("point_id_options") == "uuid" \
        else str(pid.num)


def _pid_msg(s: str):
    m = M["PointId"]()
    v = _maybe_int(s)
    if isinstance(v, int):
        m.num = v
    else:
        m.uuid = s
    return m


def _filter_to_json(f) -> dict:
    if not f.ByteSize():
        return {}
    return _payload_to_py(f.conditions.fields)


class QdrantGrpc:
    """Both Qdrant services over one registry (reference server.go)."""

    def __init__(self, registry: QdrantRegistry = None):
        self.reg = registry or QdrantRegistry()

    # ---- helpers ----
    def _coll(self, name, context):
        try:
            return self.reg.get(name)
        except KeyError:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"collection {name} not found")

    def _selected(self, c, sel) -> List[str]:
        which = sel.WhichOneof("points_selector_one_of")
        if which == "points":
            return [_pid_str(i) for i in sel.points.ids]
        if which == "filter":
            flt = _filter_to_json(sel.filter)
            return [pid for pid, pl in c.payloads.items()
                    if _matches(pl, flt)]
        return list(c.payloads)

    def _op_ok(self):
        r = M["PointsOperationResponse"]()
        r.result.status = 2  # Completed
        r.time = 0.0
        return r

    # ---- Collections ----
    def Create(self, req, context):
        params = req.vectors_config.params
        try:
            self.reg.create(req.collection_name, int(params.size),
                            _DISTANCES.get(params.distance, "cosine"))
        except KeyError:
            context.abort(grpc.StatusCode.ALREADY_EXISTS,
                          "collection exists")
        return M["CollectionOperationResponse"](result=True)

    def CollectionGet(self, req, context):
        c = self._coll(req.collection_name, context)
        r = M["GetCollectionInfoResponse"]()
        r.result.status = 1  # Green
        r.result.segments_count = 1
        r.result.vectors_count = len(c.payloads)
        r.result.points_count = len(c.payloads)
        p = r.result.config.params.vectors_config.params
        p.size = c.size
        p.distance = _DIST_NUM.get(c.distance.lower(), 1)
        return r

    def List(self, req, context):
        r = M["ListCollectionsResponse"]()
        for name in sorted(self.reg.collections):
            r.collections.add().name = name
        return r

    def Delete(self, req, context):
        existed = req.collection_name in self.reg.collections
        self.reg.drop(req.collection_name)
        return M["CollectionOperationResponse"](result=existed)

    def CollectionExists(self, req, context):
        r = M["CollectionExistsResponse"]()
        r.result.exists = req.collection_name in self.reg.collections
        return r

    # ---- Points ----
    def Upsert(self, req, context):
        c = self._coll(req.collection_name, context)
        ids, vecs = [], []
        for p in req.points:
            pid = _pid_str(p.id)
            vec = list(p.vectors.vector.data)
            if len(vec) != c.size:
                context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                              f"point {pid}: vector of size {c.size} required")
            ids.append(pid)
            vecs.append(vec)
            c.payloads[pid] = _payload_to_py(p.payload)
            c.vectors[pid] = vec
        if ids:
            c.index.add_batch(ids, vecs)
        for pid in ids:
            c.persist_point(pid)
        return self._op_ok()

    def _points_get(self, req, context):
        c = self._coll(req.collection_name, context)
        r = M["GetResponse"]()
        want_vec = req.with_vectors.enable
        for pid_m in req.ids:
            pid = _pid_str(pid_m)
            if pid not in c.payloads:
                continue
            pt = r.result.add()
            pt.id.CopyFrom(_pid_msg(pid))
            _payload_from_py(pt.payload, c.payloads[pid])
            if want_vec:
                pt.vectors.vector.data.extend(c.vectors.get(pid, []))
        return r

    def DeletePoints(self, req, context):
        c = self._coll(req.collection_name, context)
        for pid in self._selected(c, req.points):
            c.payloads.pop(pid, None)
            c.vectors.pop(pid, None)
            c.index.remove(pid)
            c.unpersist_point(pid)
        return self._op_ok()

    def Count(self, req, context):
        c = self._coll(req.collection_name, context)
        flt = _filter_to_json(req.filter)
        n = sum(1 for pl in c.payloads.values() if _matches(pl, flt)) \
            if flt else len(c.payloads)
        r = M["CountResponse"]()
        r.result.count = n
        return r

    def _search_one(self, c, req):
        limit = int(req.limit) or 10
        flt = _filter_to_json(req.filter)
        hits = c.index.search(list(req.vector), limit * 4 if flt else limit)
        out = []
        for pid, score in hits:
            if flt and not _matches(c.payloads.get(pid, {}), flt):
                continue
            if req.HasField("score_threshold") and score < req.score_threshold:
                continue
            out.append((pid, score))
            if len(out) >= limit:
                break
        return out

    def Search(self, req, context):
        c = self._coll(req.collection_name, context)
        r = M["SearchResponse"]()
        for pid, score in self._search_one(c, req):
            sp = r.result.add()
            sp.id.CopyFrom(_pid_msg(pid))
            sp.score = score
            if not req.HasField("with_payload") or req.with_payload.enable:
                _payload_from_py(sp.payload, c.payloads.get(pid, {}))
            if req.with_vectors.enable:
                sp.vectors.vector.data.extend(c.vectors.get(pid, []))
        return r

    def SearchBatch(self, req, context):
        c = self._coll(req.collection_name, context)
        r = M["SearchBatchResponse"]()
        for sub in req.search_points:
            batch = r.result.add()
            for pid, score in self._search_one(c, sub):
                sp = batch.result.add()
                sp.id.CopyFrom(_pid_msg(pid))
                sp.score = score
                _payload_from_py(sp.payload, c.payloads.get(pid, {}))
        return r

    def Scroll(self, req, context):
        c = self._coll(req.collection_name, context)
        flt = _filter_to_json(req.filter)
        limit = req.limit if req.HasField("limit") else 10
        ids = sorted(pid for pid, pl in c.payloads.items()
                     if not flt or _matches(pl, flt))
        start = 0
        if req.offset.ByteSize():
            off = _pid_str(req.offset)
            start = next((i for i, x in enumerate(ids) if x >= off), len(ids))
        page = ids[start:start + limit]
        r = M["ScrollResponse"]()
        for pid in page:
            pt = r.result.add()
            pt.id.CopyFrom(_pid_msg(pid))
            _payload_from_py(pt.payload, c.payloads[pid])
            if req.with_vectors.enable:
                pt.vectors.vector.data.extend(c.vectors.get(pid, []))
        nxt = start + len(page)
        if nxt < len(ids):
            r.next_page_offset.CopyFrom(_pid_msg(ids[nxt]))
        return r

    def SetPayload(self, req, context, overwrite=False):
        c = self._coll(req.collection_name, context)
        patch = _payload_to_py(req.payload)
        for pid in self._selected(c, req.points_selector):
            if pid in c.payloads:
                if overwrite:
                    c.payloads[pid] = dict(patch)
                else:
                    c.payloads[pid].update(patch)
                c.persist_point(pid)
        return self._op_ok()

    def OverwritePayload(self, req, context):
        return self.SetPayload(req, context, overwrite=True)

    def DeletePayload(self, req, context):
        c = self._coll(req.collection_name, context)
        for pid in self._selected(c, req.points_selector):
            pl = c.payloads.get(pid)
            if pl:
                for k in req.keys:
                    pl.pop(k, None)
                c.persist_point(pid)
        return self._op_ok()

    def ClearPayload(self, req, context):
        c = self._coll(req.collection_name, context)
        for pid in self._selected(c, req.points):
            if pid in c.payloads:
                c.payloads[pid] = {}
                c.persist_point(pid)
        return self._op_ok()

    # ---- qdrant.Snapshots (reference snapshots_service.go) ----
    def SnapshotCreate(self, req, context):
        import time as _time
        t0 = _time.time()
        try:
            d = self.reg.snapshot_create(req.collection_name)
        except KeyError:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"collection {req.collection_name!r} not found")
        r = M["CreateSnapshotResponse"]()
        r.snapshot_description.name = d["name"]
        r.snapshot_description.size = d["size"]
        r.snapshot_description.creation_time.FromSeconds(
            int(d["creation_time"]))
        r.time = _time.time() - t0
        return r

    def SnapshotList(self, req, context):
        import time as _time
        t0 = _time.time()
        try:
            snaps = self.reg.snapshot_list(req.collection_name)
        except KeyError:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"collection {req.collection_name!r} not found")
        r = M["ListSnapshotsResponse"]()
        for d in snaps:
            sd = r.snapshot_descriptions.add()
            sd.name = d["name"]
            sd.size = d["size"]
            sd.creation_time.FromSeconds(int(d["creation_time"]))
        r.time = _time.time() - t0
        return r

    def SnapshotDelete(self, req, context):
        import time as _time
        t0 = _time.time()
        try:
            self.reg.snapshot_delete(req.collection_name,
                                          req.snapshot_name)
        except KeyError:
            context.abort(grpc.StatusCode.NOT_FOUND, "snapshot not found")
        r = M["DeleteSnapshotResponse"]()
        r.time = _time.time() - t0
        return r


def _handler(fn, req_cls):
    def call(request_bytes, context):
        return fn(req_cls.FromString(request_bytes),
                  context).SerializeToString()
    return grpc.unary_unary_rpc_method_handler(call)


def _service_handlers(svc: QdrantGrpc):
    collections = {
        "Create": _handler(svc.Create, M["CreateCollection"]),
        "Get": _handler(svc.CollectionGet, M["GetCollectionInfoRequest"]),
        "List": _handler(svc.List, M["ListCollectionsRequest"]),
        "Delete": _handler(svc.Delete, M["DeleteCollection"]),
        "CollectionExists": _handler(svc.CollectionExists,
                                     M["CollectionExistsRequest"]),
    }
    points = {
        "Upsert": _handler(svc.Upsert, M["UpsertPoints"]),
        "Get": _handler(svc._points_get, M["GetPoints"]),
        "Delete": _handler(svc.DeletePoints, M["DeletePoints"]),
        "Count": _handler(svc.Count, M["CountPoints"]),
        "Search": _handler(svc.Search, M["SearchPoints"]),
        "SearchBatch": _handler(svc.SearchBatch, M["SearchBatchPoints"]),
        "Scroll": _handler(svc.Scroll, M["ScrollPoints"]),
        "SetPayload": _handler(svc.SetPayload, M["SetPayloadPoints"]),
        "OverwritePayload": _handler(svc.OverwritePayload,
                                     M["SetPayloadPoints"]),
        "DeletePayload": _handler(svc.DeletePayload,
                                  M["DeletePayloadPoints"]),
        "ClearPayload": _handler(svc.ClearPayload, M["ClearPayloadPoints"]),
    }
    snapshots = {
        "Create": _handler(svc.SnapshotCreate, M["CreateSnapshotRequest"]),
        "List": _handler(svc.SnapshotList, M["ListSnapshotsRequest"]),
        "Delete": _handler(svc.SnapshotDelete, M["DeleteSnapshotRequest"]),
    }
    return (grpc.method_handlers_generic_handler("qdrant.Collections",
                                                 collections),
            grpc.method_handlers_generic_handler("qdrant.Points", points),
            grpc.method_handlers_generic_handler("qdrant.Snapshots",
                                                 snapshots))


def serve(registry: QdrantRegistry = None, host: str = "127.0.0.1",
          port: int = 6334, max_workers: int = 8):
    """Start the Qdrant-compatible gRPC endpoint (default port 6334,
    Qdrant's standard gRPC port); returns (server, bound_port, service)."""
    from concurrent import futures
    svc = QdrantGrpc(registry)
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    server.add_generic_rpc_handlers(_service_handlers(svc))
    bound = server.add_insecure_port(f"{host}:{port}")
    server.start()
    return server, bound, svc


def stub(channel, service: str, method: str, req_cls, resp_cls):
    """Generic typed unary-unary callable (client/test helper)."""
    return channel.unary_unary(f"/qdrant.{service}/{method}",
                               request_serializer=req_cls.SerializeToString,
                               response_deserializer=resp_cls.FromString)
