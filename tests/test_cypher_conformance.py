"""Extended Cypher conformance + regression tests.

Modeled on the reference's behavioral test corpus shape (pkg/cypher's 186
files include count-consistency regressions, OPTIONAL MATCH edges, MERGE
semantics, aggregation nulls — *_count_bug_test.go etc.).
"""

import pytest

import nornicdb_amd.apoc  # noqa: F401  (registers apoc.* functions)
from nornicdb_amd.cypher import Executor
from nornicdb_amd.storage import MemoryEngine


@pytest.fixture
def ex():
    return Executor(MemoryEngine())


class TestCountConsistency:
    """Mirrors the reference's *_count_bug_test.go suites."""

    def test_count_after_create_delete_cycle(self, ex):
        for i in range(10):
            ex.execute("CREATE (:C {v: $v})", {"v": i})
        ex.execute("MATCH (n:C) WHERE n.v < 5 DETACH DELETE n")
        assert ex.execute("MATCH (n:C) RETURN count(n)").rows == [[5]]
        ex.execute("UNWIND range(0, 4) AS i CREATE (:C {v: i})")
        assert ex.execute("MATCH (n:C) RETURN count(n)").rows == [[10]]

    def test_count_distinct_nodes_vs_rows(self, ex):
        ex.execute("CREATE (a:P)-[:R]->(:Q), (a2:P)-[:R]->(:Q)")
        # wait: 'a' pattern var reuse in one CREATE makes two distinct
        r = ex.execute("MATCH (p:P)-[:R]->(q:Q) RETURN count(p), count(DISTINCT p)")
        assert r.rows == [[2, 2]]

    def test_relationship_count_after_detach(self, ex):
        ex.execute("CREATE (a:X)-[:R]->(b:X)-[:R]->(c:X)")
        ex.execute("MATCH (n:X) DETACH DELETE n")
        assert ex.execute("MATCH ()-[r]->() RETURN count(r)").rows == [[0]]
        assert ex.execute("MATCH (n) RETURN count(n)").rows == [[0]]

    def test_double_delete_is_idempotent_per_row(self, ex):
        ex.execute("CREATE (a:D)-[:R]->(b:D)")
        # both rows reference the same relationship via both directions
        ex.execute("MATCH (a:D)-[r]-(b:D) DELETE r")
        assert ex.execute("MATCH ()-[r]->() RETURN count(r)").rows == [[0]]


class TestMatchSemantics:
    def test_relationship_uniqueness_within_pattern(self, ex):
        """Cypher: one relationship cannot be traversed twice in a MATCH."""
        ex.execute("CREATE (a:U {n:'a'})-[:R]->(b:U {n:'b'})")
        r = ex.execute("MATCH (x)-[r1]-(y)-[r2]-(z) RETURN count(*)")
        # only path would reuse the single edge; uniqueness forbids it
        assert r.rows == [[0]]

    def test_cartesian_product_multi_match(self, ex):
        ex.execute("CREATE (:A1), (:A1), (:B1), (:B1), (:B1)")
        r = ex.execute("MATCH (a:A1) MATCH (b:B1) RETURN count(*)")
        assert r.rows == [[6]]

    def test_self_loop(self, ex):
        ex.execute("CREATE (a:S {n: 1}) ")
        ex.execute("MATCH (a:S) CREATE (a)-[:SELF]->(a)")
        r = ex.execute("MATCH (a:S)-[:SELF]->(a2:S) RETURN a.n, a2.n")
        assert r.rows == [[1, 1]]

    def test_bidirectional_counts_both_rows(self, ex):
        ex.execute("CREATE (a:BD {n:'a'})-[:R]->(b:BD {n:'b'})")
        r = ex.execute("MATCH (x:BD)-[:R]-(y:BD) RETURN count(*)")
        assert r.rows == [[2]]  # once from each endpoint

    def test_var_length_zero_hops(self, ex):
        ex.execute("CREATE (a:Z {n:1})-[:R]->(b:Z {n:2})")
        r = ex.execute("MATCH (a:Z {n:1})-[:R*0..1]->(x) RETURN x.n ORDER BY x.n")
        assert r.rows == [[1], [2]]

    def test_var_length_exact(self, ex):
        ex.execute("CREATE (:V2 {n:1})-[:R]->(:V2 {n:2})-[:R]->(:V2 {n:3})")
        r = ex.execute("MATCH (a:V2 {n:1})-[:R*2]->(x) RETURN x.n")
        assert r.rows == [[3]]

    def test_multiple_rel_types(self, ex):
        ex.execute("CREATE (a:MT {n:1})-[:X]->(:MT {n:2})")
        ex.execute("MATCH (a:MT {n:1}) CREATE (a)-[:Y]->(:MT {n:3})")
        r = ex.execute("MATCH (a:MT {n:1})-[:X|Y]->(b) RETURN b.n ORDER BY b.n")
        assert r.rows == [[2], [3]]

    def test_optional_match_preserves_row(self, ex):
        ex.execute("CREATE (:OM {n: 1}), (:OM {n: 2})")
        ex.execute("MATCH (a:OM {n:1}) CREATE (a)-[:L]->(:Leaf)")
        r = ex.execute(
            "MATCH (a:OM) OPTIONAL MATCH (a)-[:L]->(l) "
            "RETURN a.n, l IS NULL AS missing ORDER BY a.n")
        assert r.rows == [[1, False], [2, True]]

    def test_optional_match_with_where(self, ex):
        ex.execute("CREATE (:OW {n: 1})-[:R {w: 5}]->(:OW {n: 2})")
        r = ex.execute(
            "MATCH (a:OW {n:1}) OPTIONAL MATCH (a)-[r:R]->(b) WHERE r.w > 10 "
            "RETURN a.n, b")
        assert r.rows == [[1, None]]


class TestMergeSemantics:
    def test_merge_with_bound_endpoint(self, ex):
        ex.execute("CREATE (:MB {k: 'a'}), (:MB {k: 'b'})")
        for _ in range(3):
            ex.execute("MATCH (a:MB {k:'a'}), (b:MB {k:'b'}) "
                       "MERGE (a)-[:REL]->(b)")
        assert ex.execute("MATCH ()-[r:REL]->() RETURN count(r)").rows == [[1]]

    def test_merge_whole_pattern_semantics(self, ex):
        """MERGE of a full path creates everything when no full match."""
        ex.execute("MERGE (a:MP {k:1})-[:R]->(b:MP {k:2})")
        ex.execute("MERGE (a:MP {k:1})-[:R]->(b:MP {k:2})")
        assert ex.execute("MATCH (n:MP) RETURN count(n)").rows == [[2]]
        assert ex.execute("MATCH (:MP)-[r:R]->(:MP) RETURN count(r)").rows == [[1]]

    def test_merge_creates_when_partial_match_only(self, ex):
        ex.execute("CREATE (:PM {k: 1})")
        ex.execute("MERGE (a:PM {k:1})-[:R]->(b:PM2)")
        # full pattern had no match -> whole new pattern created (Neo4j rule)
        counts = ex.execute(
            "MATCH (n:PM) WITH count(n) AS pm MATCH (m:PM2) RETURN pm, count(m)")
        assert counts.rows == [[2, 1]]


class TestAggregationEdges:
    def test_aggregates_skip_nulls(self, ex):
        ex.execute("CREATE (:AN {v: 1}), (:AN {v: 3}), (:AN)")
        r = ex.execute("MATCH (n:AN) RETURN count(n.v), avg(n.v), collect(n.v)")
        assert r.rows[0][0] == 2 and r.rows[0][1] == 2.0
        assert sorted(r.rows[0][2]) == [1, 3]

    def test_min_max_strings(self, ex):
        ex.execute("UNWIND ['b', 'a', 'c'] AS s CREATE (:MS {s: s})")
        r = ex.execute("MATCH (n:MS) RETURN min(n.s), max(n.s)")
        assert r.rows == [["a", "c"]]

    def test_grouping_key_null(self, ex):
        ex.execute("CREATE (:GN {g: 'x', v: 1}), (:GN {v: 2}), (:GN {v: 3})")
        r = ex.execute("MATCH (n:GN) RETURN n.g, sum(n.v) ORDER BY n.g")
        # null group collects together and sorts last
        assert r.rows == [["x", 1], [None, 5]]

    def test_stdev_and_percentile(self, ex):
        ex.execute("UNWIND [1.0, 2.0, 3.0, 4.0] AS v CREATE (:SD {v: v})")
        r = ex.execute("MATCH (n:SD) RETURN stdev(n.v) > 1.29 AND stdev(n.v) < 1.30, "
                       "percentileCont(n.v, 0.5)")
        assert r.rows == [[True, 2.5]]

    def test_collect_nodes_then_unwind(self, ex):
        ex.execute("UNWIND range(1,3) AS i CREATE (:CU {v: i})")
        r = ex.execute(
            "MATCH (n:CU) WITH collect(n) AS ns UNWIND ns AS m "
            "RETURN m.v ORDER BY m.v")
        assert r.rows == [[1], [2], [3]]


class TestWithChaining:
    def test_with_aggregation_barrier(self, ex):
        ex.execute("UNWIND range(1, 6) AS i CREATE (:WB {v: i, g: i % 2})")
        r = ex.execute(
            "MATCH (n:WB) WITH n.g AS g, count(n) AS c WHERE c > 2 "
            "RETURN g, c ORDER BY g")
        assert r.rows == [[0, 3], [1, 3]]

    def test_with_rename_shadows(self, ex):
        r = ex.execute("WITH 1 AS x WITH x + 1 AS x WITH x * 10 AS x RETURN x")
        assert r.rows == [[20]]

    def test_with_limit_then_match_more(self, ex):
        ex.execute("UNWIND range(1, 5) AS i CREATE (:WL {v: i})")
        ex.execute("CREATE (:Other)")
        r = ex.execute(
            "MATCH (n:WL) WITH n ORDER BY n.v LIMIT 2 MATCH (o:Other) "
            "RETURN count(*)")
        assert r.rows == [[2]]


class TestExpressionsEdge:
    def test_null_propagation_arithmetic(self, ex):
        r = ex.execute("RETURN null * 2, null + 'a', [1, null][1], size(null)")
        assert r.rows == [[None, None, None, None]]

    def test_three_valued_logic(self, ex):
        r = ex.execute(
            "RETURN (null AND false), (null AND true), (null OR true), "
            "(null OR false), (NOT null)")
        assert r.rows == [[False, None, True, None, None]]

    def test_chained_comparison(self, ex):
        r = ex.execute("UNWIND [1, 5, 9] AS x WITH x WHERE 2 < x < 8 RETURN x")
        assert r.rows == [[5]]

    def test_negative_list_index(self, ex):
        r = ex.execute("RETURN [1,2,3][-1], 'hello'[1..3]")
        assert r.rows == [[3, "el"]]

    def test_map_access_and_keys(self, ex):
        r = ex.execute("WITH {a: 1, b: {c: 2}} AS m "
                       "RETURN m.a, m.b.c, m['a'], keys(m)")
        assert r.rows == [[1, 2, 1, ["a", "b"]]]

    def test_escape_sequences(self, ex):
        r = ex.execute(r"RETURN 'a\'b', 'tab\there', 'nl\nend'")
        assert r.rows == [["a'b", "tab\there", "nl\nend"]]

    def test_backtick_identifiers(self, ex):
        ex.execute("CREATE (n:`Weird Label` {`strange prop`: 1})")
        r = ex.execute("MATCH (n:`Weird Label`) RETURN n.`strange prop`")
        assert r.rows == [[1]]

    def test_case_returns_null_without_else(self, ex):
        r = ex.execute("UNWIND [1, 2] AS x "
                       "RETURN CASE x WHEN 1 THEN 'one' END")
        assert [row[0] for row in r.rows] == ["one", None]

    def test_integer_division_and_float(self, ex):
        r = ex.execute("RETURN 7 / 2, 7.0 / 2, -7 / 2")
        assert r.rows == [[3, 3.5, -3]]

    def test_exponent_and_modulo(self, ex):
        r = ex.execute("RETURN 2 ^ 10, 2 ^ 0.5 > 1.41, 10 % 3")
        assert r.rows[0][0] == 1024
        assert r.rows[0][1] is True


class TestPathsAndFunctions:
    def test_path_var_and_functions(self, ex):
        ex.execute("CREATE (:PV {n:1})-[:R {w: 1}]->(:PV {n:2})-[:R {w: 2}]->(:PV {n:3})")
        r = ex.execute(
            "MATCH p = (a:PV {n:1})-[:R*2]->(c) "
            "RETURN length(p), size(nodes(p)), size(relationships(p)), "
            "[x IN nodes(p) | x.n]")
        assert r.rows == [[2, 3, 2, [1, 2, 3]]]

    def test_reduce_like_sum_over_path(self, ex):
        ex.execute("CREATE (:RP {n:1})-[:R {w: 10}]->(:RP {n:2})-[:R {w: 5}]->(:RP {n:3})")
        r = ex.execute(
            "MATCH p = (:RP {n:1})-[:R*2]->(:RP {n:3}) "
            "RETURN apoc.coll.sum([r IN relationships(p) | r.w])")
        assert r.rows == [[15]]

    def test_startnode_endnode(self, ex):
        ex.execute("CREATE (:SE {n:'s'})-[:R]->(:SE {n:'e'})")
        r = ex.execute("MATCH ()-[r:R]->() RETURN startNode(r).n, endNode(r).n")
        assert r.rows == [["s", "e"]]

    def test_exists_property(self, ex):
        ex.execute("CREATE (:EP {a: 1}), (:EP)")
        r = ex.execute("MATCH (n:EP) WHERE n.a IS NOT NULL RETURN count(n)")
        assert r.rows == [[1]]


class TestConformanceBatch3:
    """Probed corners locked in as regressions."""

    @pytest.fixture
    def ex(self):
        from nornicdb_amd.cypher import Executor
        from nornicdb_amd.storage import MemoryEngine
        return Executor(MemoryEngine())

    def test_with_star_passthrough(self, ex):
        ex.execute("CREATE (a:P {v:1})-[:K]->(b:P {v:2})")
        r = ex.execute("MATCH (a:P {v:1})-[:K]->(b) WITH * RETURN a.v, b.v")
        assert r.rows == [[1, 2]]

    def test_nested_list_comprehension(self, ex):
        r = ex.execute("RETURN [x IN [1,2] | [y IN [3,4] | x*y]]")
        assert r.rows == [[[[3, 4], [6, 8]]]]

    def test_order_by_nulls_last(self, ex):
        r = ex.execute("UNWIND [3, null, 1] AS x RETURN x ORDER BY x")
        assert r.rows == [[1], [3], [None]]

    def test_string_concat_null_propagates(self, ex):
        assert ex.execute("RETURN 'a' + null").rows == [[None]]

    def test_set_null_removes_property(self, ex):
        ex.execute("CREATE (:N {a: 1, b: 2})")
        ex.execute("MATCH (n:N) SET n.a = null")
        r = ex.execute("MATCH (n:N) RETURN keys(n)")
        assert r.rows == [[["b"]]]

    def test_merge_full_pattern(self, ex):
        for _ in range(2):
            ex.execute("MERGE (a:MA {k:1})-[r:ML {p: 2}]->(b:MB {k:2})")
        assert ex.execute("MATCH (:MA)-[r:ML]->(:MB) RETURN count(r)"
                          ).rows == [[1]]

    def test_with_aggregate_then_where(self, ex):
        ex.execute("UNWIND range(1,4) AS i CREATE (:W {v: i})")
        r = ex.execute("MATCH (n:W) WITH count(n) AS c WHERE c > 3 RETURN c")
        assert r.rows == [[4]]

    def test_union_three_way_distinct(self, ex):
        r = ex.execute("RETURN 1 AS x UNION RETURN 2 AS x UNION RETURN 1 AS x")
        assert sorted(v[0] for v in r.rows) == [1, 2]

    def test_all_shortest_paths(self, ex):
        ex.execute("CREATE (a:S {n:1})-[:R]->(b:S {n:2})-[:R]->(c:S {n:3})")
        r = ex.execute("MATCH p = allShortestPaths((a:S {n:1})-[*..4]->"
                       "(c:S {n:3})) RETURN length(p)")
        assert r.rows == [[2]]
