"""Storage engine tests: CRUD, adjacency, label index, WAL durability,
corruption tolerance, snapshots, transactions, async write-behind,
namespacing.

Models the reference's storage test matrix (pkg/storage/*_test.go:
wal_durability_test, wal_corruption_test, async_engine flush races,
count-consistency suites).
"""

import os
import threading

import pytest

from nornicdb_amd.storage import (AsyncEngine, ConstraintViolation, Edge,
                                  MemoryEngine, NamespacedEngine, Node,
                                  NotFoundError, PersistentEngine, WAL,
                                  WALCorruption)


def mk(id, *labels, **props):
    return Node(id=id, labels=list(labels), properties=props)


class TestMemoryEngine:
    def test_node_crud(self):
        e = MemoryEngine()
        e.create_node(mk("a", "Person", name="Ada"))
        n = e.get_node("a")
        assert n.labels == ["Person"] and n.properties["name"] == "Ada"
        n.properties["name"] = "Ada L"
        e.update_node(n)
        assert e.get_node("a").properties["name"] == "Ada L"
        with pytest.raises(ConstraintViolation):
            e.create_node(mk("a"))
        e.delete_node("a")
        with pytest.raises(NotFoundError):
            e.get_node("a")

    def test_label_index_and_counts(self):
        e = MemoryEngine()
        for i in range(10):
            e.create_node(mk(f"p{i}", "Person", i=i))
        for i in range(5):
            e.create_node(mk(f"c{i}", "City"))
        assert e.node_count() == 15
        assert len(e.get_nodes_by_label("Person")) == 10
        n = e.get_node("p0")
        n.labels = ["Robot"]
        e.update_node(n)
        assert len(e.get_nodes_by_label("Person")) == 9
        assert len(e.get_nodes_by_label("Robot")) == 1

    def test_edges_and_adjacency(self):
        e = MemoryEngine()
        e.create_node(mk("a"))
        e.create_node(mk("b"))
        e.create_node(mk("c"))
        e.create_edge(Edge("e1", "KNOWS", "a", "b"))
        e.create_edge(Edge("e2", "KNOWS", "b", "c"))
        assert e.edge_count() == 2
        assert [x.end_node for x in e.get_out_edges("a")] == ["b"]
        assert [x.start_node for x in e.get_in_edges("c")] == ["b"]
        assert e.neighbors("b") == ["a", "c"]
        # delete with relationships requires detach
        with pytest.raises(ConstraintViolation):
            e.delete_node("b")
        e.detach_delete_node("b")
        assert e.edge_count() == 0

    def test_edge_requires_endpoints(self):
        e = MemoryEngine()
        e.create_node(mk("a"))
        with pytest.raises(NotFoundError):
            e.create_edge(Edge("e1", "KNOWS", "a", "missing"))

    def test_events(self):
        e = MemoryEngine()
        seen = []
        e.register_callback(lambda ev, o: seen.append((ev, o.id)))
        e.create_node(mk("a"))
        e.delete_node("a")
        assert ("node_created", "a") in seen and ("node_deleted", "a") in seen

    def test_property_index(self):
        e = MemoryEngine()
        for i in range(100):
            e.create_node(mk(f"p{i}", "Person", age=i % 10))
        e.create_property_index("Person", "age")
        r = e.lookup_property_index("Person", "age", 3)
        assert len(r) == 10
        assert e.lookup_property_index("Person", "name", "x") is None
        e.create_node(mk("new", "Person", age=3))
        assert len(e.lookup_property_index("Person", "age", 3)) == 11

    def test_pending_embeddings(self):
        e = MemoryEngine()
        e.create_node(mk("a"))
        e.mark_pending_embedding("a")
        assert e.pending_embeddings() == ["a"]
        e.clear_pending_embedding("a")
        assert e.pending_embeddings() == []


class TestWAL:
    def test_roundtrip(self, tmp_path):
        p = str(tmp_path / "w.log")
        w = WAL(p, sync_on_write=True)
        w.append(1, {"x": 1})
        w.append(2, {"y": "abc"})
        w.close()
        recs = list(WAL.replay(p))
        assert recs == [(1, {"x": 1}), (2, {"y": "abc"})]

    def test_torn_tail_tolerated(self, tmp_path):
        p = str(tmp_path / "w.log")
        w = WAL(p, sync_on_write=True)
        w.append(1, {"x": 1})
        w.append(2, {"x": 2})
        w.close()
        # chop off the last 3 bytes (torn write)
        data = open(p, "rb").read()
        open(p, "wb").write(data[:-3])
        recs = list(WAL.replay(p))
        assert recs == [(1, {"x": 1})]

    def test_corruption_detected(self, tmp_path):
        p = str(tmp_path / "w.log")
        w = WAL(p, sync_on_write=True)
        w.append(1, {"x": 1})
        w.append(2, {"x": 2})
        w.close()
        data = bytearray(open(p, "rb").read())
        data[14] ^= 0xFF  # flip a payload byte of record 1
        open(p, "wb").write(bytes(data))
        with pytest.raises(WALCorruption):
            list(WAL.replay(p, tolerate_corruption=False))
        # tolerant mode stops at the bad record
        assert list(WAL.replay(p, tolerate_corruption=True)) == []


class TestPersistentEngine:
    def _open(self, tmp_path, **kw):
        kw.setdefault("sync_on_write", True)
        kw.setdefault("snapshot_interval", 3600)
        return PersistentEngine(str(tmp_path / "db"), **kw)

    def test_restart_recovers(self, tmp_path):
        e = self._open(tmp_path)
        e.create_node(mk("a", "Person", name="Ada"))
        e.create_node(mk("b", "Person"))
        e.create_edge(Edge("e1", "KNOWS", "a", "b"))
        e._wal.sync()
        e._wal.close()  # simulate crash: no snapshot
        e._stop.set()

        e2 = self._open(tmp_path)
        assert e2.node_count() == 2
        assert e2.get_node("a").properties["name"] == "Ada"
        assert e2.edge_count() == 1
        e2.close()

    def test_snapshot_and_wal_truncate(self, tmp_path):
        e = self._open(tmp_path)
        for i in range(20):
            e.create_node(mk(f"n{i}", "X", i=i))
        e.snapshot()
        assert e._wal.size() == 0
        e.create_node(mk("after", "X"))
        e._wal.sync()
        e._wal.close()
        e._stop.set()

        e2 = self._open(tmp_path)
        assert e2.node_count() == 21
        assert e2.has_node("after")
        e2.close()

    def test_embedding_update_skippable(self, tmp_path):
        e = self._open(tmp_path)
        e.create_node(mk("a", "Doc"))
        e.update_embedding("a", [0.1, 0.2])
        e._wal.sync()
        e._wal.close()
        e._stop.set()

        e2 = PersistentEngine(str(tmp_path / "db"), sync_on_write=True,
                              snapshot_interval=3600, replay_embeddings=False)
        # embedding dropped on replay but node re-queued for embed
        assert e2.get_node("a").embedding is None
        assert "a" in e2.pending_embeddings()
        e2.close()

    def test_transaction_commit_and_rollback(self, tmp_path):
        e = self._open(tmp_path)
        tx = e.begin()
        tx.create_node(mk("a"))
        tx.create_node(mk("b"))
        tx.create_edge(Edge("e1", "R", "a", "b"))
        tx.commit()
        assert e.node_count() == 2 and e.edge_count() == 1

        tx2 = e.begin()
        tx2.create_node(mk("c"))
        tx2.rollback()
        assert not e.has_node("c")

        # failing tx: duplicate node id -> rolled back atomically
        tx3 = e.begin()
        tx3.create_node(mk("d"))
        tx3.create_node(mk("a"))  # conflict
        with pytest.raises(Exception):
            tx3.commit()
        assert not e.has_node("d")
        e.close()

    def test_uncommitted_tx_not_replayed(self, tmp_path):
        e = self._open(tmp_path)
        from nornicdb_amd.storage import wal as W
        e._wal.append(W.OP_TX_BEGIN, {"tx": "t1"})
        e._wal.append(W.OP_CREATE_NODE, {"id": "ghost", "labels": [], "props": {},
                                         "emb": None, "ca": 0, "ua": 0})
        # no commit marker
        e._wal.sync()
        e._wal.close()
        e._stop.set()

        e2 = self._open(tmp_path)
        assert not e2.has_node("ghost")
        e2.close()


class TestAsyncEngine:
    def test_read_your_writes_and_flush(self):
        inner = MemoryEngine()
        e = AsyncEngine(inner, flush_interval=10)  # manual flush only
        e.create_node(mk("a", "X"))
        assert e.get_node("a").id == "a"  # served from buffer
        e.flush()
        assert inner.get_node("a").id == "a"
        e.close()

    def test_count_after_flush_consistency(self):
        inner = MemoryEngine()
        e = AsyncEngine(inner, flush_interval=10)
        for i in range(100):
            e.create_node(mk(f"n{i}"))
        assert e.node_count() == 100  # count forces flush
        e.close()

    def test_concurrent_writers(self):
        inner = MemoryEngine()
        e = AsyncEngine(inner, flush_interval=0.001)

        def writer(base):
            for i in range(50):
                e.create_node(mk(f"{base}-{i}"))

        ts = [threading.Thread(target=writer, args=(b,)) for b in "abcd"]
        [t.start() for t in ts]
        [t.join() for t in ts]
        assert e.node_count() == 200
        e.close()


class TestNamespacedEngine:
    def test_isolation(self):
        base = MemoryEngine()
        a = NamespacedEngine(base, "dba")
        b = NamespacedEngine(base, "dbb")
        a.create_node(mk("x", "P", v=1))
        b.create_node(mk("x", "P", v=2))
        assert a.get_node("x").properties["v"] == 1
        assert b.get_node("x").properties["v"] == 2
        assert a.node_count() == 1 and b.node_count() == 1
        assert base.node_count() == 2
        a.create_node(mk("y"))
        a.create_edge(Edge("e1", "R", "x", "y"))
        assert b.edge_count() == 0
        assert a.neighbors("x") == ["y"]

    def test_event_filtering(self):
        base = MemoryEngine()
        a = NamespacedEngine(base, "dba")
        b = NamespacedEngine(base, "dbb")
        seen = []
        a.register_callback(lambda ev, o: seen.append(o.id))
        a.create_node(mk("mine"))
        b.create_node(mk("theirs"))
        assert seen == ["mine"]


class TestDegradedWAL:
    def test_write_failure_degrades_but_reads_work(self, tmp_path):
        from nornicdb_amd.storage import WALDegraded
        e = PersistentEngine(str(tmp_path / "db"), sync_on_write=True,
                             snapshot_interval=3600)
        e.create_node(mk("a", "X", v=1))
        # simulate device failure: close the underlying file
        e._wal._f.close()
        with pytest.raises(Exception):
            e.create_node(mk("b", "X"))
        # reads still served from RAM (reference wal_degraded.go)
        assert e.get_node("a").properties["v"] == 1
        assert e.node_count() == 1
        e._stop.set()


class TestConcurrency:
    def test_concurrent_cypher_writers_and_readers(self):
        """Race-style test (reference -race suites): concurrent writers
        against one engine must not corrupt counts or indexes."""
        import threading
        from nornicdb_amd.cypher import Executor
        eng = MemoryEngine()
        ex = Executor(eng)
        errors = []

        def writer(base):
            try:
                for i in range(50):
                    ex.execute("CREATE (:RC {k: $k})", {"k": f"{base}-{i}"})
            except Exception as exn:  # pragma: no cover
                errors.append(exn)

        def reader():
            try:
                for _ in range(50):
                    ex.execute("MATCH (n:RC) RETURN count(n)")
            except Exception as exn:  # pragma: no cover
                errors.append(exn)

        ts = [threading.Thread(target=writer, args=(b,)) for b in range(4)]
        ts += [threading.Thread(target=reader) for _ in range(2)]
        [t.start() for t in ts]
        [t.join() for t in ts]
        assert not errors
        assert eng.node_count() == 200
        assert eng.node_count_by_label("RC") == 200

    def test_concurrent_embed_queue_and_search(self):
        import threading
        from nornicdb_amd.db import open_db
        from nornicdb_amd.embed import MockEmbedder
        mgr = open_db(embedder=MockEmbedder(16), dims=16)
        db = mgr.get()
        errs = []

        def storer():
            try:
                for i in range(30):
                    db.store(f"doc number {i}")
            except Exception as e:  # pragma: no cover
                errs.append(e)

        def drainer():
            try:
                for _ in range(20):
                    db.embed_queue._process_batch()
            except Exception as e:  # pragma: no cover
                errs.append(e)

        ts = [threading.Thread(target=storer) for _ in range(2)]
        ts += [threading.Thread(target=drainer) for _ in range(2)]
        [t.start() for t in ts]
        [t.join() for t in ts]
        db.embed_queue.drain()
        assert not errs
        assert db.engine.node_count() == 60
        assert len(db.search.emb) == 60
        mgr.close()


class TestWALSegments:
    """WAL segment rotation (reference pkg/storage/wal.go segment files)."""

    def test_rotation_and_replay(self, tmp_path):
        from nornicdb_amd.storage.wal import OP_CREATE_NODE, WAL
        p = str(tmp_path / "seg.wal")
        w = WAL(p, sync_on_write=True, segment_bytes=256)
        for i in range(50):
            w.append(OP_CREATE_NODE, {"id": f"n{i}", "i": i})
        w.close()
        import os
        segs = [f for f in os.listdir(tmp_path) if f.startswith("seg.wal.")]
        assert len(segs) >= 2  # rotated at least twice
        got = [payload["i"] for op, payload in WAL.replay(p)]
        assert got == list(range(50))  # ordered across segments

    def test_truncate_removes_segments(self, tmp_path):
        from nornicdb_amd.storage.wal import OP_CREATE_NODE, WAL
        import os
        p = str(tmp_path / "seg2.wal")
        w = WAL(p, sync_on_write=True, segment_bytes=128)
        for i in range(30):
            w.append(OP_CREATE_NODE, {"id": f"n{i}"})
        assert w.size() > 128
        w.truncate()
        assert w.size() == 0
        assert not [f for f in os.listdir(tmp_path) if f.startswith("seg2.wal.")]
        w.close()
        assert list(WAL.replay(p)) == []


class TestTemporalPersistence:
    """Temporal property values survive WAL replay and snapshots
    (storage/codec.py msgpack hooks)."""

    def test_wal_replay_roundtrip(self, tmp_path):
        from nornicdb_amd.db import open_db
        d = str(tmp_path / "tdb")
        mgr = open_db(d)
        mgr.get().cypher(
            "CREATE (:E {d: date('2026-01-01'), "
            "ts: datetime('2026-06-01T12:00:00Z'), dur: duration('P1DT2H')})")
        mgr.close()
        mgr2 = open_db(d)
        r = mgr2.get().cypher(
            "MATCH (e:E) RETURN e.d.year, e.ts.month, e.dur.hours")
        assert r.rows == [[2026, 6, 2]]
        mgr2.close()

    def test_snapshot_roundtrip(self, tmp_path):
        from nornicdb_amd.db import open_db
        d = str(tmp_path / "tdb2")
        mgr = open_db(d)
        db = mgr.get()
        db.cypher("CREATE (:E {d: date('2025-12-31')})")
        if hasattr(db.engine, "snapshot"):
            db.engine.snapshot()
        mgr.close()
        mgr2 = open_db(d)
        assert mgr2.get().cypher(
            "MATCH (e:E) RETURN toString(e.d)").rows == [["2025-12-31"]]
        mgr2.close()


def test_wal_engine_durability_suite(tmp_path):
    """The round-1 WAL engine gets the same restart guarantees as the
    disk engine: schema + enforcement, vector recall, db registry."""
    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder

    d = str(tmp_path / "wal")
    mgr = open_db(d, embedder=MockEmbedder(8), dims=8, engine="wal")
    db = mgr.get()
    db.cypher("CREATE CONSTRAINT u FOR (n:P) REQUIRE n.x IS UNIQUE")
    m = db.store("wal engine memory")
    db.embed_queue.drain()
    mgr.create("side")
    mgr.close()

    mgr2 = open_db(d, embedder=MockEmbedder(8), dims=8, engine="wal")
    db2 = mgr2.get()
    assert len(db2.cypher("SHOW CONSTRAINTS").rows) == 1
    db2.cypher("CREATE (:P {x: 1})")
    import pytest as _pt
    with _pt.raises(Exception):
        db2.cypher("CREATE (:P {x: 1})")
    hits = db2.recall("wal engine", limit=1)
    assert hits and hits[0].id == m.id
    assert "side" in mgr2.list()
    mgr2.close()
