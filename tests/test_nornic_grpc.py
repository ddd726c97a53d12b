"""Native gRPC search API (reference pkg/nornicgrpc).

Wire-compatible SearchText RPC: dynamic protobuf messages mirroring
proto/nornicdb_search.proto, served via grpc generic handlers.
"""

import grpc
import pytest

from nornicdb_amd.db import DatabaseManager
from nornicdb_amd.storage.memory import MemoryEngine
from nornicdb_amd.server import nornic_grpc as ng


@pytest.fixture
def served():
    mgr = DatabaseManager(MemoryEngine())
    db = mgr.get()
    db.cypher("CREATE (:Doc {title: 'graph databases', "
              "content: 'vector search and cypher queries'})")
    db.cypher("CREATE (:Doc {title: 'cooking', content: 'boil pasta'})")
    db.cypher("CREATE (:Note {title: 'vector note', content: 'vectors'})")
    server, port = ng.serve(mgr, port=0)
    ch = grpc.insecure_channel(f"127.0.0.1:{port}")
    yield ng.client_stub(ch), db
    ch.close()
    server.stop(0)


class TestSearchText:
    def test_hybrid_search(self, served):
        stub, _ = served
        resp = stub(ng.SearchTextRequest(query="vector search", limit=5))
        assert resp.search_method in ("hybrid", "bm25")
        assert len(resp.hits) >= 1
        top = resp.hits[0]
        assert top.node_id
        assert top.score > 0
        assert dict(top.properties)  # Struct round-trip

    def test_label_filter(self, served):
        stub, _ = served
        resp = stub(ng.SearchTextRequest(query="vector", limit=5,
                                         labels=["Note"]))
        assert all(list(h.labels) == ["Note"] for h in resp.hits)

    def test_limit_clamp(self, served):
        stub, _ = served
        resp = stub(ng.SearchTextRequest(query="vector", limit=1))
        assert len(resp.hits) <= 1

    def test_empty_query_rejected(self, served):
        stub, _ = served
        with pytest.raises(grpc.RpcError) as ei:
            stub(ng.SearchTextRequest(query=""))
        assert ei.value.code() == grpc.StatusCode.INVALID_ARGUMENT

    def test_wire_compat_field_numbers(self):
        """Serialized bytes must match the reference schema's numbering
        (proto/nornicdb_search.proto fields 1-5)."""
        raw = ng.SearchTextRequest(database="db", query="q",
                                   limit=3).SerializeToString()
        # field 1 (database): tag 0x0A; field 2 (query): 0x12;
        # field 3 (limit, varint): 0x18
        assert b"\x0a\x02db" in raw
        assert b"\x12\x01q" in raw
        assert b"\x18\x03" in raw

    def test_bm25_fallback_when_embedder_fails(self):
        mgr = DatabaseManager(MemoryEngine())
        db = mgr.get()
        db.cypher("CREATE (:Doc {title: 'fallback doc', "
                  "content: 'only bm25 finds this'})")

        def broken_embedder(q):
            raise RuntimeError("no embeddings")

        server, port = ng.serve(mgr, port=0, embed_query=broken_embedder)
        ch = grpc.insecure_channel(f"127.0.0.1:{port}")
        try:
            resp = ng.client_stub(ch)(
                ng.SearchTextRequest(query="bm25", limit=5))
            assert resp.search_method == "bm25"
            assert resp.fallback_triggered is True
            assert len(resp.hits) == 1
        finally:
            ch.close()
            server.stop(0)
