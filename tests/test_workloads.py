"""Workload-shaped tests mirroring the reference's published benchmarks
(README.md:208-232: LDBC social network + Northwind query shapes).
Correctness here; throughput in scripts/bench_workload.py."""

import pytest

from nornicdb_amd.db import open_db
from nornicdb_amd.embed import MockEmbedder


@pytest.fixture(scope="module")
def social():
    mgr = open_db(embedder=MockEmbedder(8), dims=8)
    db = mgr.get()
    # people in cities, friendships, messages with tags
    db.cypher("""
        UNWIND range(0, 3) AS i
        CREATE (:City {name: 'city' + toString(i)})
    """)
    db.cypher("UNWIND range(0, 49) AS i CREATE (:Person {pid: i, name: 'p' + toString(i)})")
    db.cypher("""
        MATCH (p:Person), (c:City {name: 'city' + toString(p.pid % 4)})
        WITH p, c LIMIT 1000 CREATE (p)-[:LIVES_IN]->(c)
    """)
    db.cypher("""
        MATCH (a:Person), (b:Person)
        WHERE b.pid = (a.pid + 1) % 50 OR b.pid = (a.pid + 7) % 50
        CREATE (a)-[:KNOWS]->(b)
    """)
    db.cypher("""
        MATCH (p:Person) UNWIND range(0, 2) AS m
        CREATE (p)-[:POSTED]->(:Message {content: 'msg ' + toString(p.pid) +
                '-' + toString(m), ts: p.pid * 10 + m,
                tag: 'tag' + toString(m)})
    """)
    yield db
    mgr.close()


class TestLDBCShapes:
    def test_message_content_lookup(self, social):
        r = social.cypher(
            "MATCH (m:Message {content: 'msg 7-1'}) RETURN m.ts")
        assert r.rows == [[71]]

    def test_recent_messages_of_friends(self, social):
        r = social.cypher(
            "MATCH (p:Person {pid: 0})-[:KNOWS]->(f:Person)-[:POSTED]->(m:Message) "
            "RETURN f.pid, m.content ORDER BY m.ts DESC LIMIT 5")
        assert len(r.rows) == 5
        assert all(row[0] in (1, 7) for row in r.rows)

    def test_avg_friends_per_city(self, social):
        r = social.cypher(
            "MATCH (p:Person)-[:LIVES_IN]->(c:City) "
            "MATCH (p)-[:KNOWS]->(f) "
            "RETURN c.name AS city, count(f) * 1.0 / count(DISTINCT p) AS avg_friends "
            "ORDER BY city")
        assert len(r.rows) >= 4
        for _, avg in r.rows:
            assert avg == 2.0  # everyone has exactly 2 out-friends

    def test_tag_cooccurrence(self, social):
        r = social.cypher(
            "MATCH (p:Person)-[:POSTED]->(m1:Message), (p)-[:POSTED]->(m2:Message) "
            "WHERE m1.tag < m2.tag "
            "RETURN m1.tag, m2.tag, count(*) AS c ORDER BY c DESC LIMIT 3")
        assert r.rows[0][2] == 50  # every person posted each tag pair once


class TestNorthwindShapes:
    @pytest.fixture(scope="class")
    def shop(self):
        mgr = open_db(embedder=MockEmbedder(8), dims=8)
        db = mgr.get()
        db.cypher("UNWIND range(0, 9) AS i CREATE (:Category {cid: i})")
        db.cypher("""
            UNWIND range(0, 99) AS i
            CREATE (:Product {sku: i, stock: i % 7, price: i * 1.5})
        """)
        db.cypher("""
            MATCH (p:Product), (c:Category {cid: p.sku % 10})
            CREATE (p)-[:IN_CATEGORY]->(c)
        """)
        db.cypher("UNWIND range(0, 19) AS i CREATE (:Customer {cust: i})")
        db.cypher("""
            MATCH (cu:Customer), (p:Product)
            WHERE p.sku % 20 = cu.cust
            CREATE (cu)-[:ORDERED {qty: 1 + p.sku % 3}]->(p)
        """)
        yield db
        mgr.close()

    def test_index_lookup(self, shop):
        shop.engine.create_property_index("Product", "sku")
        assert shop.cypher("MATCH (p:Product {sku: 42}) RETURN p.price").rows == [[63.0]]

    def test_count_nodes(self, shop):
        assert shop.cypher("MATCH (p:Product) RETURN count(p)").rows == [[100]]

    def test_orders_by_customer(self, shop):
        r = shop.cypher(
            "MATCH (c:Customer {cust: 3})-[o:ORDERED]->(p:Product) "
            "RETURN p.sku ORDER BY p.sku")
        assert [x[0] for x in r.rows] == [3, 23, 43, 63, 83]

    def test_products_out_of_stock(self, shop):
        r = shop.cypher("MATCH (p:Product) WHERE p.stock = 0 RETURN count(p)")
        assert r.rows == [[15]]  # sku % 7 == 0 for 0..99

    def test_find_category_members(self, shop):
        r = shop.cypher(
            "MATCH (p:Product)-[:IN_CATEGORY]->(c:Category {cid: 4}) "
            "RETURN count(p)")
        assert r.rows == [[10]]

    def test_write_node_and_edge(self, shop):
        r = shop.cypher(
            "CREATE (p:Product {sku: 1000}) "
            "WITH p MATCH (c:Category {cid: 0}) "
            "CREATE (p)-[:IN_CATEGORY]->(c) RETURN p.sku")
        assert r.rows == [[1000]]
        assert r.stats["nodes_created"] == 1
        assert r.stats["edges_created"] == 1
