"""NornicDB-native gRPC search API.

Parity: reference pkg/nornicgrpc (proto/nornicdb_search.proto +
search_service.go) — one RPC, `NornicSearch.SearchText`, performing
hybrid (vector + BM25) search with server-side query embedding and
BM25-only fallback when embeddings are unavailable.

protoc is not available offline, so the messages are built at import
time from a hand-written FileDescriptorProto mirroring the reference
schema exactly (same package, field numbers and types); wire
compatibility with clients generated from the reference .proto is
preserved. The service is registered through grpc generic handlers.
"""

from __future__ import annotations

import time
from typing import Callable, List, Optional

import grpc
from google.protobuf import (descriptor_pb2, descriptor_pool,
                             message_factory, struct_pb2)

_PACKAGE = "nornicdb.grpc.v1"
_SERVICE = f"{_PACKAGE}.NornicSearch"

_T = descriptor_pb2.FieldDescriptorProto


def _field(name, number, ftype, label=_T.LABEL_OPTIONAL, type_name=None,
           proto3_optional=False):
    f = _T(name=name, number=number, type=ftype, label=label)
    if type_name:
        f.type_name = type_name
    if proto3_optional:
        f.proto3_optional = True
        f.oneof_index = 0
    return f


def _build_pool():
    pool = descriptor_pool.DescriptorPool()
    # struct.proto must exist in the pool for SearchHit.properties
    pool.Add(descriptor_pb2.FileDescriptorProto.FromString(
        struct_pb2.DESCRIPTOR.serialized_pb))

    fd = descriptor_pb2.FileDescriptorProto(
        name="nornicdb_search.proto", package=_PACKAGE, syntax="proto3",
        dependency=["google/protobuf/struct.proto"])

    req = fd.message_type.add()
    req.name = "SearchTextRequest"
    req.field.append(_field("database", 1, _T.TYPE_STRING))
    req.field.append(_field("query", 2, _T.TYPE_STRING))
    req.field.append(_field("limit", 3, _T.TYPE_UINT32))
    req.field.append(_field("labels", 4, _T.TYPE_STRING,
                            label=_T.LABEL_REPEATED))
    req.field.append(_field("min_similarity", 5, _T.TYPE_FLOAT,
                            proto3_optional=True))
    req.oneof_decl.add().name = "_min_similarity"

    hit = fd.message_type.add()
    hit.name = "SearchHit"
    hit.field.append(_field("node_id", 1, _T.TYPE_STRING))
    hit.field.append(_field("labels", 2, _T.TYPE_STRING,
                            label=_T.LABEL_REPEATED))
    hit.field.append(_field("properties", 3, _T.TYPE_MESSAGE,
                            type_name=".google.protobuf.Struct"))
    hit.field.append(_field("score", 4, _T.TYPE_FLOAT))
    hit.field.append(_field("rrf_score", 5, _T.TYPE_FLOAT))
    hit.field.append(_field("vector_rank", 6, _T.TYPE_INT32))
    hit.field.append(_field("bm25_rank", 7, _T.TYPE_INT32))

    resp = fd.message_type.add()
    resp.name = "SearchTextResponse"
    resp.field.append(_field("search_method", 1, _T.TYPE_STRING))
    resp.field.append(_field("hits", 2, _T.TYPE_MESSAGE,
                             label=_T.LABEL_REPEATED,
                             type_name=f".{_PACKAGE}.SearchHit"))
    resp.field.append(_field("fallback_triggered", 3, _T.TYPE_BOOL))
    resp.field.append(_field("message", 4, _T.TYPE_STRING))
    resp.field.append(_field("time_seconds", 5, _T.TYPE_DOUBLE))

    pool.Add(fd)
    return pool


_POOL = _build_pool()
SearchTextRequest = message_factory.GetMessageClass(
    _POOL.FindMessageTypeByName(f"{_PACKAGE}.SearchTextRequest"))
SearchHit = message_factory.GetMessageClass(
    _POOL.FindMessageTypeByName(f"{_PACKAGE}.SearchHit"))
SearchTextResponse = message_factory.GetMessageClass(
    _POOL.FindMessageTypeByName(f"{_PACKAGE}.SearchTextResponse"))


def _jsonable(v):
    if isinstance(v, (str, bool, int, float)) or v is None:
        return v
    if isinstance(v, (list, tuple)):
        return [_jsonable(x) for x in v]
    if isinstance(v, dict):
        return {str(k): _jsonable(x) for k, x in v.items()}
    return str(v)


class NornicSearchService:
    """SearchText over a DatabaseManager (reference search_service.go:55).

    embed_query returning None signals "embeddings unavailable" and
    triggers the BM25-only fallback, mirroring EmbedQueryFunc.
    """

    def __init__(self, manager, default_database: str = "neo4j",
                 max_limit: int = 1000,
                 embed_query: Optional[Callable] = None):
        self.manager = manager
        self.default_database = default_database
        self.max_limit = max_limit
        self._embed_query = embed_query

    def _db(self, name):
        return self.manager.get(name or self.default_database)

    def search_text(self, request: "SearchTextRequest") -> "SearchTextResponse":
        start = time.monotonic()
        if not request.query:
            raise _RpcError(grpc.StatusCode.INVALID_ARGUMENT,
                            "query is required")
        limit = int(request.limit) or 10
        limit = min(limit, self.max_limit)
        try:
            db = self._db(request.database)
        except Exception as e:
            raise _RpcError(grpc.StatusCode.NOT_FOUND, str(e))

        labels = list(request.labels) or None
        qv = None
        fallback = False
        embedder = self._embed_query or (
            db.embedder.embed_query if getattr(db, "embedder", None) else None)
        if embedder is not None:
            try:
                qv = embedder(request.query)
            except Exception:
                qv = None
        if qv is None:
            fallback = embedder is not None
            results = db.search.text_search(request.query, limit,
                                            labels=labels)
            method = "bm25"
        else:
            results = db.search.search(query=request.query, query_vec=qv,
                                       k=limit, labels=labels)
            method = "hybrid"
        if request.HasField("min_similarity") and method == "hybrid":
            results = [r for r in results
                       if r.score >= request.min_similarity]

        resp = SearchTextResponse(search_method=method,
                                  fallback_triggered=fallback,
                                  message="",
                                  time_seconds=time.monotonic() - start)
        for rank, r in enumerate(results, 1):
            hit = resp.hits.add()
            hit.node_id = str(r.id)
            if r.node is not None:
                hit.labels.extend(r.node.labels)
                hit.properties.update(_jsonable(dict(r.node.properties)))
            hit.score = float(r.score)
            hit.rrf_score = float(r.score)
            if r.source in ("vector", "hybrid"):
                hit.vector_rank = rank
            if r.source in ("fulltext", "hybrid"):
                hit.bm25_rank = rank
        return resp


class _RpcError(Exception):
    def __init__(self, code, details):
        self.code = code
        self.details = details


def _make_handler(service: NornicSearchService):
    def search_text(request_bytes, context):
        request = SearchTextRequest.FromString(request_bytes)
        try:
            return service.search_text(request).SerializeToString()
        except _RpcError as e:
            context.abort(e.code, e.details)

    method_handlers = {
        "SearchText": grpc.unary_unary_rpc_method_handler(
            search_text,
            request_deserializer=None,   # raw bytes in, bytes out
            response_serializer=None),
    }
    return grpc.method_handlers_generic_handler(_SERVICE, method_handlers)


def serve(manager, host: str = "127.0.0.1", port: int = 50052,
          embed_query: Optional[Callable] = None, max_workers: int = 8):
    """Start the native gRPC search server; returns (server, bound_port)."""
    from concurrent import futures
    service = NornicSearchService(manager, embed_query=embed_query)
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    server.add_generic_rpc_handlers((_make_handler(service),))
    bound = server.add_insecure_port(f"{host}:{port}")
    server.start()
    return server, bound


def client_stub(channel):
    """Typed SearchText callable over a grpc channel (test/client helper)."""
    return channel.unary_unary(
        f"/{_SERVICE}/SearchText",
        request_serializer=SearchTextRequest.SerializeToString,
        response_deserializer=SearchTextResponse.FromString)
