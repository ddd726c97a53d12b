"""Cypher workload throughput on a 10k-node synthetic graph (CPU path).

BASELINE.json config #1: "Bolt+Cypher MATCH on 10k-node synthetic graph,
CPU-only path". Reference comparison points (README.md:208-224, other
hardware): LDBC message lookup 6389 ops/s, Northwind index lookup 7623,
count nodes 5253, node write 5578, edge write 6626.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from nornicdb_amd.db import open_db
from nornicdb_amd.embed import MockEmbedder


def bench(name, fn, n=2000):
    fn(0)  # warm
    t0 = time.perf_counter()
    for i in range(n):
        fn(i)
    dt = time.perf_counter() - t0
    print(f"{name:<28} {n/dt:>8.0f} ops/s   {dt/n*1e3:6.3f} ms/op", flush=True)
    return n / dt


def main():
    mgr = open_db(embedder=MockEmbedder(8), dims=8)
    db = mgr.get()
    print("building 10k-node graph ...", flush=True)
    for s in range(0, 10000, 1000):
        db.cypher("UNWIND range($s, $e) AS i CREATE "
                  "(:Person {pid: i, name: 'p' + toString(i), age: i % 90})",
                  {"s": s, "e": s + 999})
    db.cypher("MATCH (a:Person) WHERE a.pid < 5000 "
              "MATCH (b:Person {pid: a.pid + 5000}) CREATE (a)-[:KNOWS]->(b)")
    db.engine.create_property_index("Person", "pid")
    db.engine.create_property_index("Person", "name")
    print(f"graph: {db.engine.node_count()} nodes, {db.engine.edge_count()} edges",
          flush=True)

    ex = db.executor  # bypass result cache: measure execution

    bench("indexed property lookup",
          lambda i: ex.execute("MATCH (p:Person {pid: $p}) RETURN p.name",
                               {"p": i % 10000}))
    bench("indexed string lookup",
          lambda i: ex.execute("MATCH (p:Person {name: $n}) RETURN p.pid",
                               {"n": f"p{i % 10000}"}))
    bench("count nodes",
          lambda i: ex.execute("MATCH (p:Person) RETURN count(p)"), n=200)
    bench("1-hop traversal",
          lambda i: ex.execute(
              "MATCH (a:Person {pid: $p})-[:KNOWS]->(b) RETURN b.pid",
              {"p": i % 5000}))
    bench("filtered scan (age)",
          lambda i: ex.execute(
              "MATCH (p:Person) WHERE p.age = $a RETURN count(p)",
              {"a": i % 90}), n=50)
    c = [10000]
    def write_node(i):
        c[0] += 1
        ex.execute("CREATE (:Person {pid: $p, name: 'w', age: 1})", {"p": c[0]})
    bench("write: node", write_node)
    def write_edge(i):
        ex.execute("MATCH (a:Person {pid: $a}), (b:Person {pid: $b}) "
                   "CREATE (a)-[:LINKED]->(b)",
                   {"a": i % 5000, "b": 5000 + i % 5000})
    bench("write: edge", write_edge)
    mgr.close()


if __name__ == "__main__":
    main()
