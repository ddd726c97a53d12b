"""Qdrant-compatible gRPC endpoint (reference pkg/qdrantgrpc).

Collections + Points services reconstructed as dynamic protobuf
descriptors (upstream field numbers) over the shared QdrantRegistry.
"""

import grpc
import pytest

from nornicdb_amd.server import qdrant_grpc as qg

M = qg.M


@pytest.fixture
def served():
    server, port, svc = qg.serve(port=0)
    ch = grpc.insecure_channel(f"127.0.0.1:{port}")

    def call(service, method, req, resp_name):
        return qg.stub(ch, service, method, type(req), M[resp_name])(req)

    req = M["CreateCollection"](collection_name="c")
    req.vectors_config.params.size = 4
    req.vectors_config.params.distance = 1
    call("Collections", "Create", req, "CollectionOperationResponse")
    up = M["UpsertPoints"](collection_name="c")
    for i, vec in enumerate([[1, 0, 0, 0], [0, 1, 0, 0], [0.9, 0.1, 0, 0]], 1):
        p = up.points.add()
        p.id.num = i
        p.vectors.vector.data.extend(vec)
        p.payload["city"].string_value = "berlin" if i < 3 else "paris"
        p.payload["rank"].integer_value = i
    call("Points", "Upsert", up, "PointsOperationResponse")
    yield call, svc
    ch.close()
    server.stop(0)


class TestCollections:
    def test_lifecycle(self, served):
        call, _ = served
        r = call("Collections", "List", M["ListCollectionsRequest"](),
                 "ListCollectionsResponse")
        assert [c.name for c in r.collections] == ["c"]
        r = call("Collections", "Get",
                 M["GetCollectionInfoRequest"](collection_name="c"),
                 "GetCollectionInfoResponse")
        assert r.result.points_count == 3
        assert r.result.config.params.vectors_config.params.size == 4
        r = call("Collections", "CollectionExists",
                 M["CollectionExistsRequest"](collection_name="nope"),
                 "CollectionExistsResponse")
        assert r.result.exists is False
        call("Collections", "Delete",
             M["DeleteCollection"](collection_name="c"),
             "CollectionOperationResponse")
        r = call("Collections", "CollectionExists",
                 M["CollectionExistsRequest"](collection_name="c"),
                 "CollectionExistsResponse")
        assert r.result.exists is False

    def test_missing_collection_not_found(self, served):
        call, _ = served
        with pytest.raises(grpc.RpcError) as ei:
            call("Points", "Count", M["CountPoints"](collection_name="x"),
                 "CountResponse")
        assert ei.value.code() == grpc.StatusCode.NOT_FOUND


class TestPoints:
    def test_search_scores_and_payload(self, served):
        call, _ = served
        sr = M["SearchPoints"](collection_name="c", limit=2)
        sr.vector.extend([1, 0, 0, 0])
        r = call("Points", "Search", sr, "SearchResponse")
        assert [h.id.num for h in r.result] == [1, 3]
        assert r.result[0].score == pytest.approx(1.0)
        assert qg.from_value(r.result[0].payload["city"]) == "berlin"

    def test_search_batch(self, served):
        call, _ = served
        sb = M["SearchBatchPoints"](collection_name="c")
        for vec in ([1, 0, 0, 0], [0, 1, 0, 0]):
            sub = sb.search_points.add()
            sub.collection_name = "c"
            sub.limit = 1
            sub.vector.extend(vec)
        r = call("Points", "SearchBatch", sb, "SearchBatchResponse")
        assert [b.result[0].id.num for b in r.result] == [1, 2]

    def test_get_with_vectors(self, served):
        call, _ = served
        gp = M["GetPoints"](collection_name="c")
        gp.ids.add().num = 2
        gp.with_vectors.enable = True
        r = call("Points", "Get", gp, "GetResponse")
        assert list(r.result[0].vectors.vector.data) == [0.0, 1.0, 0.0, 0.0]

    def test_scroll_pagination(self, served):
        call, _ = served
        sc = M["ScrollPoints"](collection_name="c")
        sc.limit = 2
        r = call("Points", "Scroll", sc, "ScrollResponse")
        assert [p.id.num for p in r.result] == [1, 2]
        sc2 = M["ScrollPoints"](collection_name="c")
        sc2.limit = 2
        sc2.offset.CopyFrom(r.next_page_offset)
        r2 = call("Points", "Scroll", sc2, "ScrollResponse")
        assert [p.id.num for p in r2.result] == [3]
        assert not r2.next_page_offset.ByteSize()

    def test_payload_ops(self, served):
        call, _ = served
        sp = M["SetPayloadPoints"](collection_name="c")
        sp.payload["tag"].string_value = "x"
        sp.points_selector.points.ids.add().num = 1
        call("Points", "SetPayload", sp, "PointsOperationResponse")
        dp = M["DeletePayloadPoints"](collection_name="c")
        dp.keys.append("city")
        dp.points_selector.points.ids.add().num = 1
        call("Points", "DeletePayload", dp, "PointsOperationResponse")
        gp = M["GetPoints"](collection_name="c")
        gp.ids.add().num = 1
        r = call("Points", "Get", gp, "GetResponse")
        pl = qg._payload_to_py(r.result[0].payload)
        assert pl == {"tag": "x", "rank": 1}
        cp = M["ClearPayloadPoints"](collection_name="c")
        cp.points.points.ids.add().num = 1
        call("Points", "ClearPayload", cp, "PointsOperationResponse")
        r = call("Points", "Get", gp, "GetResponse")
        assert qg._payload_to_py(r.result[0].payload) == {}

    def test_delete_points(self, served):
        call, _ = served
        dp = M["DeletePoints"](collection_name="c")
        dp.points.points.ids.add().num = 3
        call("Points", "Delete", dp, "PointsOperationResponse")
        r = call("Points", "Count", M["CountPoints"](collection_name="c"),
                 "CountResponse")
        assert r.result.count == 2

    def test_value_roundtrip(self):
        v = {"a": 1, "b": 2.5, "c": "s", "d": [1, "x", None],
             "e": {"nested": True}}
        assert qg.from_value(qg.to_value(v)) == v

    def test_rest_grpc_shared_registry(self, served):
        """gRPC writes are visible through the REST layer's registry."""
        _, svc = served
        c = svc.reg.get("c")
        assert set(c.payloads) == {"1", "2", "3"}
