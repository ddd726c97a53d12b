"""Qdrant REST compat + GraphQL endpoint tests."""

import pytest
from fastapi.testclient import TestClient

from nornicdb_amd.db import open_db
from nornicdb_amd.embed import MockEmbedder
from nornicdb_amd.server import create_app


@pytest.fixture
def client():
    mgr = open_db(embedder=MockEmbedder(16), dims=16)
    with TestClient(create_app(mgr)) as c:
        yield c
    mgr.close()


class TestQdrant:
    def test_collection_lifecycle(self, client):
        r = client.put("/collections/docs", json={
            "vectors": {"size": 4, "distance": "Cosine"}})
        assert r.json()["status"] == "ok"
        assert client.put("/collections/docs", json={
            "vectors": {"size": 4}}).status_code == 409
        got = client.get("/collections/docs").json()["result"]
        assert got["config"]["params"]["vectors"]["size"] == 4
        names = [c["name"] for c in
                 client.get("/collections").json()["result"]["collections"]]
        assert "docs" in names

    def test_upsert_search_delete(self, client):
        client.put("/collections/c1", json={"vectors": {"size": 4,
                                                        "distance": "Cosine"}})
        pts = [{"id": i, "vector": v, "payload": {"tag": f"t{i}"}}
               for i, v in [(1, [1, 0, 0, 0]), (2, [0, 1, 0, 0]),
                            (3, [0.9, 0.1, 0, 0])]]
        r = client.put("/collections/c1/points", json={"points": pts})
        assert r.json()["result"]["status"] == "completed"
        res = client.post("/collections/c1/points/search",
                          json={"vector": [1, 0, 0, 0], "limit": 2}).json()["result"]
        assert res[0]["id"] == 1 and res[1]["id"] == 3
        assert res[0]["payload"]["tag"] == "t1"
        client.post("/collections/c1/points/delete", json={"points": [1]})
        res = client.post("/collections/c1/points/search",
                          json={"vector": [1, 0, 0, 0], "limit": 2}).json()["result"]
        assert res[0]["id"] == 3

    def test_scroll(self, client):
        client.put("/collections/c2", json={"vectors": {"size": 2}})
        client.put("/collections/c2/points", json={"points": [
            {"id": i, "vector": [i, 0], "payload": {}} for i in range(5)]})
        page = client.post("/collections/c2/points/scroll",
                           json={"limit": 3}).json()["result"]
        assert len(page["points"]) == 3
        assert page["next_page_offset"] is not None


class TestGraphQL:
    def _gql(self, client, q):
        return client.post("/graphql", json={"query": q}).json()

    def test_nodes_query(self, client):
        mgr = client.app.state.manager
        mgr.get().cypher("CREATE (:Person {name: 'Ada'}), (:Person {name: 'Bob'})")
        r = self._gql(client, '{ nodes(label: "Person") { id properties } }')
        assert len(r["data"]["nodes"]) == 2
        names = {n["properties"]["name"] for n in r["data"]["nodes"]}
        assert names == {"Ada", "Bob"}

    def test_node_with_relationships(self, client):
        db = client.app.state.manager.get()
        db.cypher("CREATE (a:G {name:'a'})-[:KNOWS]->(b:G {name:'b'})")
        nid = db.cypher("MATCH (a:G {name:'a'}) RETURN a").rows[0][0].id
        r = self._gql(client,
                      '{ node(id: "%s") { id relationships { type } } }' % nid)
        assert r["data"]["node"]["relationships"][0]["type"] == "KNOWS"

    def test_mutation_create(self, client):
        r = self._gql(client,
                      'mutation { createNode(labels: ["M"], '
                      'properties: "{\\"v\\": 7}") { id properties } }')
        assert r["data"]["createNode"]["properties"]["v"] == 7

    def test_cypher_field(self, client):
        r = self._gql(client, '{ cypher(query: "RETURN 1 + 1") { columns rows } }')
        assert r["data"]["cypher"]["rows"] == [[2]]

    def test_error_reported(self, client):
        r = self._gql(client, '{ nosuch }')
        assert "errors" in r


class TestGraphQLExtended:
    """Full resolver surface (reference pkg/graphql/resolvers/*_impl.go)."""

    @pytest.fixture
    def gq(self):
        from nornicdb_amd.db import NornicDB
        from nornicdb_amd.server.graphql import GraphQLExecutor
        from nornicdb_amd.storage.memory import MemoryEngine
        db = NornicDB(MemoryEngine(), auto_embed=False)
        return GraphQLExecutor(db)

    def test_input_objects_and_variables(self, gq):
        r = gq.execute(
            'mutation($p: JSON) { createNode(input: {labels: ["P"], '
            'properties: $p}) { id properties } }', {"p": {"name": "ann"}})
        assert r["data"]["createNode"]["properties"] == {"name": "ann"}

    def test_crud_and_counts(self, gq):
        a = gq.execute('mutation { createNode(input: {labels: ["P"], '
                       'properties: {name: "a"}}) { id } }')["data"]["createNode"]["id"]
        b = gq.execute('mutation { createNode(input: {labels: ["P"], '
                       'properties: {name: "b"}}) { id } }')["data"]["createNode"]["id"]
        rel = gq.execute('mutation { createRelationship(input: {from: "%s", '
                         'to: "%s", type: "K"}) { id } }' % (a, b))
        assert "errors" not in rel
        r = gq.execute('{ nodeCount relationshipCount '
                       'nodesByLabel(label: "P") { id } }')["data"]
        assert r["nodeCount"] == 2 and r["relationshipCount"] == 1
        assert len(r["nodesByLabel"]) == 2
        r = gq.execute('{ relationshipsBetween(from: "%s", to: "%s") '
                       '{ type } }' % (a, b))["data"]
        assert r["relationshipsBetween"][0]["type"] == "K"
        upd = gq.execute('mutation { updateNode(input: {id: "%s", '
                         'properties: {age: 3}}) { properties } }' % a)
        assert upd["data"]["updateNode"]["properties"]["age"] == 3
        assert gq.execute('mutation { deleteRelationship(id: "%s") }'
                          % rel["data"]["createRelationship"]["id"]
                          )["data"]["deleteRelationship"] is True

    def test_bulk_and_merge(self, gq):
        r = gq.execute('mutation { bulkCreateNodes(input: {nodes: ['
                       '{labels: ["B"], properties: {i: 1}}, '
                       '{labels: ["B"], properties: {i: 2}}]}) { count } }')
        assert r["data"]["bulkCreateNodes"]["count"] == 2
        m1 = gq.execute('mutation { mergeNode(input: {labels: ["B"], '
                        'properties: {i: 1}, mergeKey: "i"}) { id } }')
        m2 = gq.execute('mutation { mergeNode(input: {labels: ["B"], '
                        'properties: {i: 1}, mergeKey: "i"}) { id } }')
        assert m1["data"]["mergeNode"]["id"] == m2["data"]["mergeNode"]["id"]
        assert gq.execute('{ nodeCount(label: "B") }')["data"]["nodeCount"] == 2

    def test_neighborhood_and_nested(self, gq):
        a = gq.execute('mutation { createNode(input: {labels: ["N"]}) '
                       '{ id } }')["data"]["createNode"]["id"]
        b = gq.execute('mutation { createNode(input: {labels: ["N"]}) '
                       '{ id } }')["data"]["createNode"]["id"]
        gq.execute('mutation { createRelationship(input: {from: "%s", '
                   'to: "%s", type: "L"}) { id } }' % (a, b))
        r = gq.execute('{ node(id: "%s") { id outgoing { type endNode } '
                       'neighbors { id } } }' % a)["data"]["node"]
        assert r["outgoing"][0]["endNode"] == b
        assert r["neighbors"][0]["id"] == b
        nb = gq.execute('{ neighborhood(id: "%s", depth: 1) '
                        '{ nodes { id } } }' % a)["data"]["neighborhood"]
        assert {n["id"] for n in nb["nodes"]} == {a, b}

    def test_event_broker(self, gq):
        sub = gq.broker.subscribe(["EV"])
        gq.execute('mutation { createNode(input: {labels: ["EV"], '
                   'properties: {x: 1}}) { id } }')
        ev = sub.get_nowait()
        assert ev["event"] == "nodeCreated"
        assert ev["data"]["properties"] == {"x": 1}
        # label filter: other labels don't reach this subscriber
        gq.execute('mutation { createNode(input: {labels: ["OTHER"]}) { id } }')
        import queue as q
        with pytest.raises(q.Empty):
            sub.get_nowait()

    def test_clear_all(self, gq):
        gq.execute('mutation { createNode(input: {labels: ["X"]}) { id } }')
        assert gq.execute('mutation { clearAll }')["data"]["clearAll"] is True
        assert gq.execute('{ nodeCount }')["data"]["nodeCount"] == 0


class TestQdrantExtended:
    """Payload ops, count, filters, universal query (reference
    pkg/qdrantgrpc surface)."""

    @pytest.fixture
    def qc(self):
        from starlette.testclient import TestClient

        from nornicdb_amd.db import DatabaseManager
        from nornicdb_amd.server.http import create_app
        from nornicdb_amd.storage.memory import MemoryEngine
        app = create_app(DatabaseManager(MemoryEngine()))
        c = TestClient(app)
        c.put("/collections/t", json={"vectors": {"size": 4,
                                                  "distance": "Cosine"}})
        c.put("/collections/t/points", json={"points": [
            {"id": 1, "vector": [1, 0, 0, 0], "payload": {"city": "oslo", "n": 1}},
            {"id": 2, "vector": [0, 1, 0, 0], "payload": {"city": "bergen", "n": 2}},
            {"id": 3, "vector": [0, 0, 1, 0], "payload": {"city": "oslo", "n": 3}},
        ]})
        return c

    def test_count_and_exists(self, qc):
        assert qc.post("/collections/t/points/count", json={}
                       ).json()["result"]["count"] == 3
        assert qc.post("/collections/t/points/count", json={
            "filter": {"must": [{"key": "city", "match": {"value": "oslo"}}]}
        }).json()["result"]["count"] == 2
        assert qc.get("/collections/t/exists").json()["result"]["exists"]
        assert not qc.get("/collections/nope/exists").json()["result"]["exists"]

    def test_payload_ops(self, qc):
        qc.post("/collections/t/points/payload", json={
            "points": [1], "payload": {"extra": True}})
        pts = qc.post("/collections/t/points", json={"ids": [1]}
                      ).json()["result"]
        assert pts[0]["payload"]["extra"] is True
        qc.put("/collections/t/points/payload", json={
            "points": [1], "payload": {"only": 1}})
        pts = qc.post("/collections/t/points", json={"ids": [1]}
                      ).json()["result"]
        assert pts[0]["payload"] == {"only": 1}
        qc.post("/collections/t/points/payload/delete", json={
            "points": [2], "keys": ["n"]})
        pts = qc.post("/collections/t/points", json={"ids": [2]}
                      ).json()["result"]
        assert "n" not in pts[0]["payload"]
        qc.post("/collections/t/points/payload/clear", json={"points": [3]})
        pts = qc.post("/collections/t/points", json={"ids": [3]}
                      ).json()["result"]
        assert pts[0]["payload"] == {}

    def test_query_with_filter(self, qc):
        r = qc.post("/collections/t/points/query", json={
            "query": [1, 0, 0, 0], "limit": 2,
            "filter": {"must": [{"key": "city", "match": {"value": "oslo"}}]}
        }).json()["result"]["points"]
        assert r[0]["id"] == 1
        assert all(p["payload"]["city"] == "oslo" for p in r)

    def test_query_by_id_and_range(self, qc):
        r = qc.post("/collections/t/points/query", json={
            "query": 1, "limit": 2}).json()["result"]["points"]
        assert all(p["id"] != 1 for p in r)  # excludes the anchor
        r = qc.post("/collections/t/points/count", json={
            "filter": {"must": [{"key": "n", "range": {"gte": 2}}]}
        }).json()["result"]["count"]
        assert r == 2


class TestGraphQLPaths:
    """shortestPath/allPaths/neighborhood with the reference schema's
    argument names (schema.graphql: startNodeId/endNodeId/maxDepth)."""

    def _gq(self):
        from nornicdb_amd.db import DatabaseManager
        from nornicdb_amd.storage.memory import MemoryEngine
        from nornicdb_amd.server.graphql import GraphQLExecutor
        db = DatabaseManager(MemoryEngine()).get()
        db.cypher("CREATE (a:P {name:'a'})-[:R]->(b:P {name:'b'})"
                  "-[:R]->(c:P {name:'c'})")
        db.cypher("MATCH (a:P {name:'a'}), (c:P {name:'c'}) "
                  "CREATE (a)-[:R]->(c)")
        ids = {r[1]: r[0] for r in
               db.cypher("MATCH (n:P) RETURN id(n), n.name").rows}
        return GraphQLExecutor(db), ids

    def test_shortest_path(self):
        gq, ids = self._gq()
        r = gq.execute('query($a: ID!, $b: ID!) { shortestPath('
                       'startNodeId: $a, endNodeId: $b) { nodes { id } } }',
                       {"a": ids["a"], "b": ids["c"]})
        assert len(r["data"]["shortestPath"]["nodes"]) == 2

    def test_all_paths(self):
        gq, ids = self._gq()
        r = gq.execute('query($a: ID!, $b: ID!) { allPaths(startNodeId: $a, '
                       'endNodeId: $b, maxDepth: 4) { id } }',
                       {"a": ids["a"], "b": ids["c"]})
        assert len(r["data"]["allPaths"]) == 2  # direct + via b

    def test_neighborhood_node_id_arg(self):
        gq, ids = self._gq()
        r = gq.execute('query($n: ID!) { neighborhood(nodeId: $n, depth: 1) '
                       '{ nodes { id } } }', {"n": ids["a"]})
        assert len(r["data"]["neighborhood"]["nodes"]) == 3


def test_qdrant_collections_survive_restart(tmp_path):
    """Collections/points persist as _QdrantCollection/_QdrantPoint_*
    nodes in the storage engine and reload on boot (reference
    pkg/qdrantgrpc/registry.go) — create, upsert, restart, search,
    delete, restart again."""
    from fastapi.testclient import TestClient

    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder
    from nornicdb_amd.server import create_app

    d = str(tmp_path / "store")

    def client(mgr):
        app = create_app(mgr, auth=None)
        return TestClient(app.app if hasattr(app, "app") else app)

    mgr = open_db(d, embedder=MockEmbedder(8), dims=8)
    c = client(mgr)
    assert c.put("/collections/docs",
                 json={"vectors": {"size": 4, "distance": "Cosine"}}
                 ).json()["status"] == "ok"
    assert c.put("/collections/docs/points", json={"points": [
        {"id": 1, "vector": [1, 0, 0, 0], "payload": {"t": "a"}},
        {"id": 2, "vector": [0, 1, 0, 0], "payload": {"t": "b"}}]}
        ).json()["status"] == "ok"
    mgr.close()

    mgr2 = open_db(d, embedder=MockEmbedder(8), dims=8)
    c2 = client(mgr2)
    names = [x["name"] for x in
             c2.get("/collections").json()["result"]["collections"]]
    assert "docs" in names
    hits = c2.post("/collections/docs/points/search",
                   json={"vector": [1, 0, 0, 0], "limit": 1}).json()["result"]
    assert str(hits[0]["id"]) == "1"
    pl = c2.post("/collections/docs/points",
                 json={"ids": [2]}).json()["result"]
    assert pl[0]["payload"] == {"t": "b"}
    c2.post("/collections/docs/points/delete", json={"points": [1]})
    mgr2.close()

    mgr3 = open_db(d, embedder=MockEmbedder(8), dims=8)
    c3 = client(mgr3)
    assert c3.post("/collections/docs/points/count",
                   json={}).json()["result"]["count"] == 1
    # collection drop removes persisted state too
    c3.delete("/collections/docs")
    mgr3.close()
    mgr4 = open_db(d, embedder=MockEmbedder(8), dims=8)
    c4 = client(mgr4)
    names = [x["name"] for x in
             c4.get("/collections").json()["result"]["collections"]]
    assert "docs" not in names
    mgr4.close()
