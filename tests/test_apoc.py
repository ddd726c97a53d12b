"""APOC library tests (functions in expressions + CALL procedures)."""

import pytest

from nornicdb_amd.db import open_db
from nornicdb_amd.embed import MockEmbedder


@pytest.fixture
def db():
    mgr = open_db(embedder=MockEmbedder(16), dims=16)
    yield mgr.get()
    mgr.close()


class TestCollFunctions:
    def test_coll_basics(self, db):
        r = db.cypher(
            "RETURN apoc.coll.sum([1,2,3]), apoc.coll.max([1,5,2]), "
            "apoc.coll.sort([3,1,2]), apoc.coll.toSet([1,1,2]), "
            "apoc.coll.flatten([[1,2],[3]]), apoc.coll.contains([1,2], 2)")
        assert r.rows == [[6, 5, [1, 2, 3], [1, 2], [1, 2, 3], True]]

    def test_coll_sets(self, db):
        r = db.cypher(
            "RETURN apoc.coll.union([1,2],[2,3]), "
            "apoc.coll.intersection([1,2,3],[2,3,4]), "
            "apoc.coll.subtract([1,2,3],[2])")
        assert r.rows == [[[1, 2, 3], [2, 3], [1, 3]]]

    def test_coll_partition_pairs(self, db):
        r = db.cypher("RETURN apoc.coll.partition([1,2,3,4,5], 2), "
                      "apoc.coll.pairsMin([1,2,3])")
        assert r.rows == [[[[1, 2], [3, 4], [5]], [[1, 2], [2, 3]]]]


class TestMapText:
    def test_map(self, db):
        r = db.cypher(
            "RETURN apoc.map.merge({a:1},{b:2}), apoc.map.fromPairs([['x',1]]), "
            "apoc.map.removeKey({a:1,b:2}, 'a'), apoc.map.flatten({a:{b:1}})")
        assert r.rows == [[{"a": 1, "b": 2}, {"x": 1}, {"b": 2}, {"a.b": 1}]]

    def test_text(self, db):
        r = db.cypher(
            "RETURN apoc.text.join(['a','b'], '-'), apoc.text.capitalize('ada'), "
            "apoc.text.camelCase('hello world'), "
            "apoc.text.levenshteinDistance('kitten','sitting'), "
            "apoc.text.slug('Hello, World!')")
        assert r.rows == [["a-b", "Ada", "helloWorld", 3, "hello-world"]]

    def test_hashing(self, db):
        r = db.cypher("RETURN apoc.util.md5(['abc'])")
        assert r.rows[0][0] == "900150983cd24fb0d6963f7d28e17f72"

    def test_json(self, db):
        r = db.cypher("RETURN apoc.convert.fromJsonMap('{\"k\": 5}'), "
                      "apoc.json.path({a: {b: [1,2]}}, '$.a.b[1]')")
        assert r.rows == [[{"k": 5}, 2]]

    def test_date(self, db):
        r = db.cypher("RETURN apoc.date.format(0, 'ms', 'yyyy-MM-dd')")
        assert r.rows == [["1970-01-01"]]
        r = db.cypher("RETURN apoc.date.parse('1970-01-02', 'ms', 'yyyy-MM-dd')")
        assert r.rows == [[86400000]]


class TestAlgoProcedures:
    def _make_graph(self, db):
        db.cypher("CREATE (a:V {name:'a'})-[:R]->(b:V {name:'b'})-[:R]->(c:V {name:'c'})")
        db.cypher("MATCH (a:V {name:'a'}), (c:V {name:'c'}) CREATE (c)-[:R]->(a)")

    def test_pagerank(self, db):
        self._make_graph(db)
        r = db.cypher("CALL apoc.algo.pageRank() YIELD node, score "
                      "RETURN node.name, score")
        assert len(r.rows) == 3
        assert abs(sum(row[1] for row in r.rows) - 1.0) < 1e-2

    def test_dijkstra(self, db):
        db.cypher("CREATE (a:W {name:'a'})-[:L {weight: 2}]->(b:W {name:'b'})"
                  "-[:L {weight: 3}]->(c:W {name:'c'})")
        r = db.cypher(
            "MATCH (a:W {name:'a'}), (c:W {name:'c'}) "
            "CALL apoc.algo.dijkstra(a, c, 'L', 'weight') YIELD path, weight "
            "RETURN weight")
        assert r.rows == [[5.0]]

    def test_community(self, db):
        self._make_graph(db)
        r = db.cypher("CALL apoc.community.wcc() YIELD node, component "
                      "RETURN count(DISTINCT component)")
        assert r.rows == [[1]]


class TestCreateRefactor:
    def test_create_node_rel(self, db):
        r = db.cypher("CALL apoc.create.node(['X'], {v: 1}) YIELD node RETURN node.v")
        assert r.rows == [[1]]
        db.cypher("CALL apoc.create.node(['Y'], {v: 2}) YIELD node RETURN node")
        r = db.cypher(
            "MATCH (x:X), (y:Y) "
            "CALL apoc.create.relationship(x, 'REL', {w: 1}, y) YIELD rel "
            "RETURN type(rel)")
        assert r.rows == [["REL"]]

    def test_merge_node_idempotent(self, db):
        for _ in range(2):
            db.cypher("CALL apoc.merge.node(['M'], {k: 1}, {c: true}, {m: true}) "
                      "YIELD node RETURN node")
        r = db.cypher("MATCH (n:M) RETURN count(n), n.c, n.m")
        assert r.rows == [[1, True, True]]

    def test_refactor_rename_label(self, db):
        db.cypher("CREATE (:Old {v:1}), (:Old {v:2})")
        r = db.cypher("CALL apoc.refactor.rename.label('Old', 'New') "
                      "YIELD count RETURN count")
        assert r.rows == [[2]]
        assert db.cypher("MATCH (n:New) RETURN count(n)").rows == [[2]]

    def test_merge_nodes(self, db):
        db.cypher("CREATE (a:MN {x:1})-[:R]->(b:T), (c:MN {y:2})")
        r = db.cypher("MATCH (a:MN {x:1}), (c:MN {y:2}) "
                      "CALL apoc.refactor.mergeNodes([a, c]) YIELD node "
                      "RETURN node.x, node.y")
        assert r.rows == [[1, 2]]
        assert db.cypher("MATCH (n:MN) RETURN count(n)").rows == [[1]]


class TestPeriodicMeta:
    def test_periodic_iterate(self, db):
        db.cypher("UNWIND range(1, 25) AS i CREATE (:PI {v: i})")
        r = db.cypher(
            "CALL apoc.periodic.iterate("
            "'MATCH (n:PI) RETURN n', 'SET n.doubled = n.v * 2', "
            "{batchSize: 10}) YIELD batches, total RETURN batches, total")
        assert r.rows == [[3, 25]]
        assert db.cypher("MATCH (n:PI {v: 5}) RETURN n.doubled").rows == [[10]]

    def test_meta_stats(self, db):
        db.cypher("CREATE (:A)-[:R1]->(:B)")
        r = db.cypher("CALL apoc.meta.stats() YIELD nodeCount, relCount, labels "
                      "RETURN nodeCount, relCount, labels")
        assert r.rows[0][0] == 2 and r.rows[0][1] == 1
        assert r.rows[0][2] == {"A": 1, "B": 1}

    def test_export_json(self, db, tmp_path):
        db.cypher("CREATE (:E {v: 1})")
        f = str(tmp_path / "out.json")
        r = db.cypher(f"CALL apoc.export.json.all('{f}') YIELD nodes RETURN nodes")
        assert r.rows == [[1]]
        import json
        data = json.load(open(f))
        assert data["nodes"][0]["properties"]["v"] == 1


class TestPathAtomicTrigger:
    def test_path_expand(self, db):
        db.cypher("CREATE (a:PE {n:'a'})-[:R]->(b:PE {n:'b'})-[:R]->(c:PE {n:'c'})")
        r = db.cypher(
            "MATCH (a:PE {n:'a'}) CALL apoc.path.expand(a, 'R>', null, 1, 2) "
            "YIELD nodes RETURN size(nodes)")
        assert sorted(x[0] for x in r.rows) == [2, 3]

    def test_subgraph_nodes(self, db):
        db.cypher("CREATE (a:SG {n:1})-[:R]->(:SG {n:2})-[:R]->(:SG {n:3})")
        r = db.cypher("MATCH (a:SG {n:1}) "
                      "CALL apoc.path.subgraphNodes(a, 2) YIELD node "
                      "RETURN count(node)")
        assert r.rows == [[3]]

    def test_atomic_add(self, db):
        db.cypher("CREATE (:AT {c: 10})")
        r = db.cypher("MATCH (n:AT) CALL apoc.atomic.add(n, 'c', 5) "
                      "YIELD value RETURN value")
        assert r.rows == [[15]]
        assert db.cypher("MATCH (n:AT) RETURN n.c").rows == [[15]]

    def test_trigger_fires_on_write(self, db):
        db.cypher("CALL apoc.trigger.add('audit', "
                  "'MERGE (c:TriggerCounter) ON CREATE SET c.n = 1 "
                  "ON MATCH SET c.n = c.n + 1')")
        db.cypher("CREATE (:TG)")
        db.cypher("CREATE (:TG)")
        r = db.cypher("MATCH (c:TriggerCounter) RETURN c.n")
        assert r.rows[0][0] >= 2
        db.cypher("CALL apoc.trigger.remove('audit')")

    def test_load_export_csv(self, db, tmp_path):
        db.cypher("CREATE (:CSV {name: 'x', v: 1}), (:CSV {name: 'y', v: 2})")
        f = str(tmp_path / "out.csv")
        r = db.cypher(f"CALL apoc.export.csv.all('{f}') YIELD nodes RETURN nodes")
        assert r.rows == [[2]]
        r = db.cypher(f"CALL apoc.load.csv('{f}') YIELD map RETURN map['name'] "
                      "ORDER BY map['name']")
        assert [x[0] for x in r.rows] == ["x", "y"]
