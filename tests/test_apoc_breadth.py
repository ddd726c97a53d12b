"""APOC breadth categories (bitwise/math/number/stats/scoring/spatial/
hashing/util/json/temporal/label/meta/diff/xml/graph/agg + engine-backed
procedure categories).

Parity: reference apoc/ packages (SURVEY.md §2 "APOC library", ~950 fns).
"""

import pytest

from nornicdb_amd.db import NornicDB
from nornicdb_amd.storage.memory import MemoryEngine


@pytest.fixture
def db():
    return NornicDB(MemoryEngine(), auto_embed=False)


def one(db, q, params=None):
    return db.cypher(q, params).rows[0][0]


class TestPureFunctions:
    def test_bitwise(self, db):
        assert one(db, "RETURN apoc.bitwise.op(60, '&', 13)") == 12
        assert one(db, "RETURN apoc.bitwise.setBit(0, 3)") == 8
        assert one(db, "RETURN apoc.bitwise.countBits(255)") == 8

    def test_math(self, db):
        assert one(db, "RETURN apoc.math.gcd(12, 18)") == 6
        assert one(db, "RETURN apoc.math.lcm(4, 6)") == 12
        assert one(db, "RETURN apoc.math.isPrime(97)") is True
        assert one(db, "RETURN apoc.math.nextPrime(14)") == 17
        assert one(db, "RETURN apoc.math.factorial(5)") == 120
        assert one(db, "RETURN apoc.math.fibonacci(10)") == 55
        assert abs(one(db, "RETURN apoc.math.sigmoid(0)") - 0.5) < 1e-9

    def test_number(self, db):
        assert one(db, "RETURN apoc.number.romanize(1987)") == "MCMLXXXVII"
        assert one(db, "RETURN apoc.number.arabize('XIV')") == 14
        assert one(db, "RETURN apoc.number.toHex(255)") == "ff"
        assert one(db, "RETURN apoc.number.fromBinary('1010')") == 10
        assert one(db, "RETURN apoc.number.isEven(4)") is True

    def test_stats(self, db):
        assert one(db, "RETURN apoc.stats.median([1,2,3,4,5])") == 3
        assert one(db, "RETURN apoc.stats.iqr([1,2,3,4,5,6,7,8])") == 3.5
        assert one(db, "RETURN apoc.stats.correlation([1,2,3],[2,4,6])") == pytest.approx(1.0)
        assert one(db, "RETURN apoc.stats.outliers([1,2,3,2,3,1,2,100])") == [100]

    def test_scoring(self, db):
        assert one(db, "RETURN apoc.scoring.cosine([1,0],[0,1])") == pytest.approx(0.0)
        assert one(db, "RETURN apoc.scoring.jaccard([1,2,3],[2,3,4])") == pytest.approx(0.5)
        sm = one(db, "RETURN apoc.scoring.softmax([1.0,1.0])")
        assert sm == pytest.approx([0.5, 0.5])

    def test_spatial(self, db):
        # Paris -> London great-circle ~343 km
        d = one(db, "RETURN apoc.spatial.haversineDistance(48.8566, 2.3522, 51.5074, -0.1278)")
        assert 330e3 < d < 350e3
        gh = one(db, "RETURN apoc.spatial.encodeGeohash(48.8583, 2.2945, 7)")
        back = one(db, f"RETURN apoc.spatial.decodeGeohash('{gh}')")
        assert abs(back["latitude"] - 48.8583) < 0.01

    def test_hashing(self, db):
        assert one(db, "RETURN apoc.hashing.sha256('abc')").startswith("ba7816bf")
        assert 0 <= one(db, "RETURN apoc.hashing.jumpHash(12345, 10)") < 10
        a = one(db, "RETURN apoc.hashing.fingerprint({a: 1, b: 2})")
        b = one(db, "RETURN apoc.hashing.fingerprint({b: 2, a: 1})")
        assert a == b  # order-insensitive

    def test_util(self, db):
        assert one(db, "RETURN apoc.util.decodeBase64(apoc.util.encodeBase64('hi'))") == "hi"
        assert one(db, "RETURN apoc.util.partition([1,2,3,4,5], 2)") == [[1, 2], [3, 4], [5]]
        assert one(db, "RETURN apoc.util.when(true, 'a', 'b')") == "a"

    def test_json(self, db):
        assert one(db, "RETURN apoc.json.path('{\"a\": {\"b\": [1,2]}}', '$.a.b[1]')") == 2
        flat = one(db, "RETURN apoc.json.flatten({a: {b: 1}})")
        assert flat == {"a.b": 1}
        assert one(db, "RETURN apoc.json.unflatten({`a.b`: 1})") == {"a": {"b": 1}}

    def test_temporal(self, db):
        assert one(db, "RETURN apoc.temporal.isLeapYear(2024)") is True
        assert one(db, "RETURN apoc.temporal.quarter(datetime('2026-09-12T00:00:00Z'))") == 3
        assert one(db, "RETURN toString(apoc.temporal.add("
                       "datetime('2026-01-01T00:00:00Z'), 'P1D'))").startswith("2026-01-02")

    def test_meta_and_diff(self, db):
        assert one(db, "RETURN apoc.meta.cypherType(1.5)") == "FLOAT"
        assert one(db, "RETURN apoc.meta.cypherType(date('2026-01-01'))") == "DATE"
        d = one(db, "RETURN apoc.diff.maps({a:1, b:2}, {b:3, c:4})")
        assert set(d["leftOnly"]) == {"a"} and set(d["different"]) == {"b"}

    def test_xml(self, db):
        m = one(db, "RETURN apoc.xml.parse('<a x=\"1\"><b>hello</b></a>')")
        assert m["_type"] == "a" and m["@x"] == "1"
        assert m["_children"][0]["_text"] == "hello"

    def test_agg(self, db):
        r = db.cypher("UNWIND [5,1,3] AS x RETURN apoc.agg.first(x), "
                      "apoc.agg.last(x), apoc.agg.median(x), apoc.agg.product(x)")
        assert r.rows == [[5, 3, 3, 15]]
        r = db.cypher("UNWIND [1,1,2] AS x RETURN apoc.agg.mode(x)")
        assert r.rows == [[1]]


class TestEngineProcedures:
    @pytest.fixture
    def g(self, db):
        db.cypher("CREATE (a:P {name:'alpha'})-[:K]->(b:P {name:'beta'})"
                  "-[:K]->(c:P {name:'gamma'})")
        return db

    def test_node_degree_and_neighbors(self, g):
        assert one(g, "MATCH (b:P {name:'beta'}) CALL apoc.node.degree(b) "
                      "YIELD value RETURN value") == 2
        r = g.cypher("MATCH (a:P {name:'alpha'}) "
                     "CALL apoc.neighbors.tohop(a, 'K', 2) YIELD node "
                     "RETURN node.name ORDER BY node.name")
        assert [x[0] for x in r.rows] == ["beta", "gamma"]

    def test_cypher_run(self, g):
        r = g.cypher("CALL apoc.cypher.runFirstColumnSingle("
                     "'MATCH (n:P) RETURN count(n)', {}) YIELD value RETURN value")
        assert r.rows == [[3]]

    def test_search(self, g):
        r = g.cypher("CALL apoc.search.node({P: 'name'}, 'contains', 'amm') "
                     "YIELD node RETURN node.name")
        assert r.rows == [["gamma"]]

    def test_export_import_roundtrip(self, g):
        data = one(g, "CALL apoc.export.json.query("
                      "'MATCH (n:P) RETURN n.name AS name', {}) "
                      "YIELD data RETURN data")
        assert data.count("\n") == 2  # 3 rows
        db2 = NornicDB(MemoryEngine(), auto_embed=False)
        r = db2.cypher(
            "CALL apoc.import.json($d) YIELD nodes RETURN nodes",
            {"d": '{"id": "x1", "labels": ["Q"], "properties": {"v": 1}}'})
        assert r.rows == [[1]]
        assert db2.cypher("MATCH (n:Q) RETURN n.v").rows == [[1]]

    def test_refactor_invert(self, g):
        g.cypher("MATCH (:P {name:'alpha'})-[r:K]->(:P {name:'beta'}) "
                 "CALL apoc.refactor.invertRelationship(r) YIELD rel "
                 "RETURN rel")
        r = g.cypher("MATCH (b:P {name:'beta'})-[:K]->(a:P {name:'alpha'}) "
                     "RETURN count(*)")
        assert r.rows == [[1]]

    def test_schema_procs(self, g):
        g.cypher("CALL apoc.schema.createUniqueConstraint('P', 'name') "
                 "YIELD name RETURN name")
        assert one(g, "CALL apoc.schema.nodeConstraintExists('P', ['name']) "
                      "YIELD value RETURN value") is True

    def test_merge_relationship_idempotent(self, g):
        for _ in range(2):
            g.cypher("MATCH (a:P {name:'alpha'}), (c:P {name:'gamma'}) "
                     "CALL apoc.merge.relationship(a, 'LINKS', {}, {}, c) "
                     "YIELD rel RETURN rel")
        assert g.cypher("MATCH (:P {name:'alpha'})-[r:LINKS]->() "
                        "RETURN count(r)").rows == [[1]]

    def test_meta_data(self, g):
        rows = g.cypher("CALL apoc.meta.data() YIELD label, property "
                        "RETURN label, property").rows
        assert ["P", "name"] in rows

    def test_function_count(self):
        from nornicdb_amd.cypher.functions import FUNCTIONS
        assert len([k for k in FUNCTIONS if k.startswith("apoc.")]) >= 450


class TestApocBatch3:
    """atomic extras, periodic scheduler, node/rel mutator procs,
    cypher.parallel, text phonetics, coll/map extras, convert.toTree."""

    @pytest.fixture
    def db(self):
        return NornicDB(MemoryEngine(), auto_embed=False)

    def test_atomic_cas_and_concat(self, db):
        db.cypher("CREATE (:A {v: 0, s: 'x'})")
        r = db.cypher("MATCH (a:A) CALL apoc.atomic.compareAndSwap(a, 'v', 0, 5) "
                      "YIELD swapped RETURN swapped")
        assert r.rows == [[True]]
        r = db.cypher("MATCH (a:A) CALL apoc.atomic.compareAndSwap(a, 'v', 0, 9) "
                      "YIELD swapped RETURN swapped")
        assert r.rows == [[False]]
        db.cypher("MATCH (a:A) CALL apoc.atomic.concat(a, 's', 'y') "
                  "YIELD value RETURN value")
        assert db.cypher("MATCH (a:A) RETURN a.s").rows == [["xy"]]

    def test_periodic_submit_and_list(self, db):
        import time
        db.cypher("CALL apoc.periodic.submit('t1', 'CREATE (:Done)') "
                  "YIELD name RETURN name")
        # poll: the job runs on a background thread (slow machines need
        # longer than a fixed sleep)
        for _ in range(100):
            if db.cypher("MATCH (d:Done) RETURN count(d)").rows == [[1]]:
                break
            time.sleep(0.1)
        assert db.cypher("MATCH (d:Done) RETURN count(d)").rows == [[1]]
        rows = db.cypher("CALL apoc.periodic.list() YIELD name, done "
                         "RETURN name, done").rows
        assert ["t1", True] in rows

    def test_node_mutator_procs(self, db):
        db.cypher("CREATE (:M {a: 1})")
        db.cypher("MATCH (m:M) CALL apoc.node.addLabel(m, 'Extra') "
                  "YIELD node RETURN node")
        assert db.cypher("MATCH (m:Extra) RETURN count(m)").rows == [[1]]
        db.cypher("MATCH (m:M) CALL apoc.node.removeProperty(m, 'a') "
                  "YIELD node RETURN node")
        assert db.cypher("MATCH (m:M) RETURN m.a").rows == [[None]]

    def test_cypher_parallel(self, db):
        r = db.cypher("CALL apoc.cypher.parallel('RETURN $x * 2 AS y', "
                      "{x: [1,2,3]}, 'x') YIELD value RETURN value.y "
                      "ORDER BY value.y")
        assert [x[0] for x in r.rows] == [2, 4, 6]

    def test_text_phonetics(self, db):
        r = db.cypher("RETURN apoc.text.phonetic('Robert'), "
                      "apoc.text.fuzzyMatch('color', 'colour')")
        assert r.rows[0] == ["R163", True]

    def test_coll_map_extras(self, db):
        assert one(db, "RETURN apoc.coll.insertAll([1,4], 1, [2,3])") == [1, 2, 3, 4]
        assert one(db, "RETURN apoc.coll.dropDuplicateNeighbors([1,1,2,2,1])") == [1, 2, 1]
        assert one(db, "RETURN apoc.map.fromValues(['a', 1, 'b', 2])") == {"a": 1, "b": 2}
        assert one(db, "RETURN apoc.map.mget({a: 1}, ['a', 'z'], [0, 0])") == [1, 0]

    def test_convert_to_tree(self, db):
        db.cypher("CREATE (a:T {name:'root'})-[:HAS]->(b:T {name:'leaf'})")
        r = db.cypher("MATCH p = (a:T)-[:HAS]->(b) "
                      "RETURN apoc.convert.toTree(collect(p))")
        tree = r.rows[0][0]
        assert tree and tree[0]["name"] == "root"
        assert tree[0]["has"][0]["name"] == "leaf"


class TestGraphML:
    def test_roundtrip(self):
        a = NornicDB(MemoryEngine(), auto_embed=False)
        a.cypher("CREATE (x:G {name:'x'})-[:L {w: '2'}]->(y:H)")
        data = a.cypher("CALL apoc.export.graphml.all() YIELD data "
                        "RETURN data").rows[0][0]
        assert "<graphml" in data
        b = NornicDB(MemoryEngine(), auto_embed=False)
        r = b.cypher("CALL apoc.import.graphml($d) YIELD nodes, relationships "
                     "RETURN nodes, relationships", {"d": data})
        assert r.rows == [[2, 1]]
        assert b.cypher("MATCH (g:G)-[l:L]->(h:H) RETURN g.name, l.w"
                        ).rows == [["x", "2"]]


class TestApocBatch5:
    @pytest.fixture
    def g(self):
        db = NornicDB(MemoryEngine(), auto_embed=False)
        db.cypher("CREATE (a:P {v:1})-[:K]->(b:P {v:2})-[:K]->(c:P {v:3}), (d:Q)")
        return db

    def test_community_metrics(self, g):
        assert one(g, "CALL apoc.community.numComponents() YIELD count "
                      "RETURN count") == 2
        assert 0 < one(g, "CALL apoc.community.density() YIELD density "
                          "RETURN density") < 1
        assert one(g, "CALL apoc.community.stronglyConnectedComponents() "
                      "YIELD component RETURN count(DISTINCT component)") == 4

    def test_trigger_lifecycle(self, g):
        g.cypher("CALL apoc.trigger.add('t', 'RETURN 1', null) "
                 "YIELD name RETURN name")
        assert one(g, "CALL apoc.trigger.isEnabled('t') YIELD enabled "
                      "RETURN enabled") is True
        g.cypher("CALL apoc.trigger.disable('t') YIELD name RETURN name")
        assert one(g, "CALL apoc.trigger.isEnabled('t') YIELD enabled "
                      "RETURN enabled") is False
        data = one(g, "CALL apoc.trigger.export() YIELD data RETURN data")
        g.cypher("CALL apoc.trigger.removeAll() YIELD removed RETURN removed")
        assert one(g, "CALL apoc.trigger.count() YIELD count RETURN count") == 0
        g.cypher("CALL apoc.trigger.import($d) YIELD imported RETURN imported",
                 {"d": data})
        assert one(g, "CALL apoc.trigger.count() YIELD count RETURN count") == 1

    def test_paths_utilities(self, g):
        r = g.cypher("MATCH p = (a:P {v:1})-[:K*2]->(c) "
                     "RETURN length(apoc.paths.reverse(p)), "
                     "apoc.paths.simple(p)")
        assert r.rows == [[2, True]]

    def test_merge_map_utils(self, g):
        assert one(g, "RETURN apoc.merge.properties({a:1}, {b:2})") == \
            {"a": 1, "b": 2}
        assert one(g, "RETURN apoc.merge.conflict({a:1, c:3}, {a:2, b:2})") == ["a"]

    def test_offline_loaders_raise(self, g):
        import pytest as _pt
        with _pt.raises(Exception, match="offline|network"):
            g.cypher("CALL apoc.load.jdbc('x', 'y') YIELD value RETURN value")

    def test_algo_aliases(self, g):
        assert one(g, "CALL apoc.algo.betweennessCentrality() YIELD node "
                      "RETURN count(node)") == 4


class TestRegistryCompletion:
    """Final 20 names from the reference registry (apoc/apoc.go):
    allPairs/cover, cloneSubgraph, merge.*, hamiltonian/eulerian,
    normalize/denormalize/redirect/categorize, search index mgmt."""

    def _db(self):
        from nornicdb_amd.db import DatabaseManager
        from nornicdb_amd.storage.memory import MemoryEngine
        db = DatabaseManager(MemoryEngine()).get()
        db.cypher("CREATE (a:N {name:'A'})-[:R {weight:1.0}]->"
                  "(b:N {name:'B'})-[:R {weight:1.0}]->(c:N {name:'C'})")
        return db

    def test_all_pairs_and_cover(self):
        db = self._db()
        r = db.cypher("MATCH (n:N) WITH collect(n) AS ns "
                      "CALL apoc.algo.allPairs(ns, 'weight') "
                      "YIELD source, target, path RETURN count(*)")
        assert r.rows == [[3]]  # A->B, A->C, B->C (directed)
        r = db.cypher("MATCH (n:N) WITH collect(n) AS ns "
                      "CALL apoc.algo.cover(ns) YIELD node RETURN node.name")
        assert r.rows == [["B"]]  # B covers both edges

    def test_merge_node_and_relationship(self):
        db = self._db()
        r1 = db.cypher("CALL apoc.merge.mergeNode(['M'], {k: 1}, "
                       "{created: true}, {seen: true}) YIELD node "
                       "RETURN node.created, node.seen")
        assert r1.rows == [[True, None]]
        r2 = db.cypher("CALL apoc.merge.mergeNode(['M'], {k: 1}, "
                       "{created: true}, {seen: true}) YIELD node "
                       "RETURN node.created, node.seen")
        assert r2.rows == [[True, True]]  # matched: on_match applied
        assert db.cypher("MATCH (m:M) RETURN count(m)").rows == [[1]]
        r = db.cypher("MATCH (a:M), (c:N {name:'C'}) CALL "
                      "apoc.merge.mergeRelationship(a, 'L', {w: 1}, "
                      "{new: true}, c) YIELD rel RETURN rel.new")
        assert r.rows == [[True]]

    def test_clone_subgraph(self):
        db = self._db()
        r = db.cypher("MATCH (n:N) WITH collect(n) AS ns "
                      "CALL apoc.refactor.cloneSubgraph(ns) "
                      "YIELD input, output RETURN count(*)")
        assert r.rows == [[3]]
        assert db.cypher("MATCH (n:N) RETURN count(n)").rows == [[6]]
        # clones carry the edges between cloned nodes
        assert db.cypher("MATCH (:N)-[r:R]->(:N) RETURN count(r)"
                         ).rows == [[4]]

    def test_normalize_denormalize(self):
        db = self._db()
        r = db.cypher("MATCH (a:N {name:'A'}) CALL apoc.refactor.normalize("
                      "a, 'name', 'Val', 'HAS') YIELD node, relationship "
                      "RETURN node.value")
        assert r.rows == [["A"]]
        r = db.cypher("MATCH (a:N)-[:HAS]->(:Val) CALL "
                      "apoc.refactor.denormalize(a, 'HAS', 'name') "
                      "YIELD node RETURN node.name")
        assert r.rows == [["A"]]

    def test_hamiltonian_path(self):
        db = self._db()
        r = db.cypher("MATCH (n:N) WITH collect(n) AS ns "
                      "MATCH (a:N {name:'A'}), (c:N {name:'C'}) "
                      "CALL apoc.paths.hamiltonian(ns, a, c) YIELD path "
                      "RETURN length(path)")
        assert r.rows == [[2]]  # A-B-C uses both edges

    def test_eulerian_path(self):
        db = self._db()
        r = db.cypher("MATCH (a:N {name:'A'}), (c:N {name:'C'}) "
                      "CALL apoc.paths.eulerian(a, c) YIELD path "
                      "RETURN length(path)")
        assert r.rows == [[2]]

    def test_search_not_in_and_index_mgmt(self):
        db = self._db()
        r = db.cypher("CALL apoc.search.notIn('N', 'name', ['B']) "
                      "YIELD node RETURN count(node)")
        assert r.rows == [[2]]
        assert db.cypher("CALL apoc.search.index('N', ['name'])"
                         ).rows == [["ok"]]
        assert db.cypher("CALL apoc.search.reindex('N')").rows == [["ok"]]
        assert db.cypher("CALL apoc.search.dropIndex('N', ['name'])"
                         ).rows == [["ok"]]

    def test_json_schema(self):
        db = self._db()
        r = db.cypher("CALL apoc.load.jsonSchema($j) YIELD value "
                      "RETURN value", {"j": '{"a": 1, "b": ["x"]}'})
        assert r.rows == [[{"a": "int", "b": ["str"]}]]

    def test_full_reference_registry_covered(self):
        """Every name register()ed in the reference's apoc/apoc.go
        resolves to a procedure, function, or aggregate here."""
        import re as _re
        src = open("/root/reference/apoc/apoc.go").read()
        refnames = {m.group(1).lower() for m in
                    _re.finditer(r'register\("(apoc\.[a-zA-Z0-9_.]+)"', src)}
        from nornicdb_amd.cypher.functions import (AGGREGATES,
                                                   AGG_FINALIZERS, FUNCTIONS)
        db = self._db()
        mine = set(db.executor.procedures) | set(FUNCTIONS) | \
            set(AGGREGATES) | set(AGG_FINALIZERS)
        missing = sorted(refnames - mine)
        assert not missing, f"uncovered reference APOC names: {missing}"
