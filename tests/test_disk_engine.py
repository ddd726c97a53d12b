"""DiskEngine / LSMStore tests: the on-disk storage engine.

Mirrors the reference's durability test matrix
(pkg/storage/wal_corruption_test.go, wal_durability_test.go,
badger_backup.go) plus LSM-specific behavior: bounded-memory scans,
compaction, crash-mid-commit (kill -9 equivalent), torn tails,
encryption at rest, and backup/restore.
"""

import os
import random
import shutil
import signal
import struct
import subprocess
import sys

import pytest

from nornicdb_amd.storage.disk import DiskEngine
from nornicdb_amd.storage.lsm import LSMStore, SSTable
from nornicdb_amd.storage.types import (ConstraintViolation, Edge, Node,
                                        NotFoundError)


@pytest.fixture
def tdir(tmp_path):
    return str(tmp_path / "db")


# ---------------------------------------------------------------------------
# LSM store
# ---------------------------------------------------------------------------

class TestLSM:
    def test_basic_roundtrip(self, tdir):
        s = LSMStore(tdir, memtable_bytes=2048)
        kv = {}
        random.seed(1)
        for i in range(1500):
            k = f"k{random.randrange(400):04d}".encode()
            v = os.urandom(random.randrange(1, 50))
            s.put(k, v)
            kv[k] = v
            if random.random() < 0.15:
                dk = f"k{random.randrange(400):04d}".encode()
                s.delete(dk)
                kv.pop(dk, None)
        assert dict(s.scan(b"k")) == kv
        for k, v in list(kv.items())[:100]:
            assert s.get(k) == v
        assert s.get(b"nope") is None
        s.close()
        s2 = LSMStore(tdir)
        assert dict(s2.scan(b"k")) == kv
        s2.close()

    def test_batch_atomic_and_ordering(self, tdir):
        s = LSMStore(tdir)
        s.write_batch([(b"a", b"1"), (b"c", b"3"), (b"b", b"2")])
        assert [k for k, _ in s.scan(b"")] == [b"a", b"b", b"c"]
        s.close()

    def test_compaction_preserves_newest(self, tdir):
        s = LSMStore(tdir, memtable_bytes=512)
        for gen in range(5):
            for i in range(60):
                s.put(f"x{i:03d}".encode(), f"g{gen}".encode())
        s.compact_all()
        assert s.stats()["tables"] == 1
        for i in range(60):
            assert s.get(f"x{i:03d}".encode()) == b"g4"
        # tombstones dropped by full merge
        s.delete(b"x000")
        s.compact_all()
        assert s.get(b"x000") is None
        assert all(k != b"x000" for k, _ in s.scan(b"x"))
        s.close()

    def test_torn_log_tail_recovers(self, tdir):
        s = LSMStore(tdir, sync_on_write=True)
        s.put(b"good", b"1")
        s.put(b"good2", b"2")
        log_path = s._log_path
        s._log.flush()
        s.close = lambda: None  # simulate crash: no clean close
        # append a torn record
        with open(log_path, "ab") as f:
            f.write(b"\x44\x4c\xff\xff\x00\x00partial")
        s2 = LSMStore(tdir)
        assert s2.get(b"good") == b"1"
        assert s2.get(b"good2") == b"2"
        s2.close()

    def test_corrupt_log_record_stops_replay(self, tdir):
        s = LSMStore(tdir, sync_on_write=True)
        s.put(b"a", b"1")
        s.put(b"b", b"2")
        log_path = s._log_path
        s._log.close()
        raw = open(log_path, "rb").read()
        # flip a payload byte of the first record
        raw = bytearray(raw)
        raw[12] ^= 0xFF
        open(log_path, "wb").write(bytes(raw))
        s2 = LSMStore(tdir)
        # first record corrupt -> CRC stops replay; store opens empty but sane
        assert s2.get(b"a") is None
        s2.put(b"c", b"3")
        assert s2.get(b"c") == b"3"
        s2.close()

    def test_block_crc_detects_sst_corruption(self, tdir):
        s = LSMStore(tdir)
        for i in range(500):
            s.put(f"k{i:04d}".encode(), os.urandom(40))
        s.flush()
        path = s._tables[0].path
        s.close()
        raw = bytearray(open(path, "rb").read())
        raw[100] ^= 0xFF  # corrupt inside first block
        open(path, "wb").write(bytes(raw))
        s2 = LSMStore(tdir)
        from nornicdb_amd.storage.lsm import CorruptTable
        with pytest.raises(CorruptTable):
            for _ in s2.scan(b"k"):
                pass
        s2.close()

    def test_kill9_mid_write_durable(self, tmp_path):
        """Hard-kill a child that wrote synced records; reopen and verify."""
        d = str(tmp_path / "kdb")
        code = f"""
import os, sys
sys.path.insert(0, {os.path.dirname(os.path.dirname(os.path.abspath(__file__)))!r})
from nornicdb_amd.storage.lsm import LSMStore
s = LSMStore({d!r}, sync_on_write=True)
for i in range(50):
    s.put(f"k{{i:03d}}".encode(), b"v%d" % i)
print("READY", flush=True)
import time
time.sleep(30)
"""
        p = subprocess.Popen([sys.executable, "-c", code],
                             stdout=subprocess.PIPE)
        assert p.stdout.readline().strip() == b"READY"
        os.kill(p.pid, signal.SIGKILL)
        p.wait()
        s = LSMStore(d)
        for i in range(50):
            assert s.get(f"k{i:03d}".encode()) == b"v%d" % i
        s.close()

    def test_backup_restore(self, tdir, tmp_path):
        s = LSMStore(tdir, memtable_bytes=1024)
        kv = {f"k{i:04d}".encode(): os.urandom(30) for i in range(300)}
        for k, v in kv.items():
            s.put(k, v)
        bak = str(tmp_path / "backup.sst")
        s.backup(bak)
        # backup is consistent even as writes continue
        s.put(b"after", b"x")
        rdir = str(tmp_path / "restored")
        r = LSMStore.restore(bak, rdir)
        assert dict(r.scan(b"k")) == kv
        assert r.get(b"after") is None
        r.close()
        s.close()

    def test_encryption_at_rest(self, tdir):
        from nornicdb_amd.utils.encryption import EncryptionManager
        em = EncryptionManager("secret", salt=b"s" * 16)
        s = LSMStore(tdir, memtable_bytes=512, crypt=em)
        kv = {f"name{i}".encode(): f"PLAINTEXT-{i}".encode() * 3
              for i in range(100)}
        for k, v in kv.items():
            s.put(k, v)
        s.flush()
        s.close()
        blob = b"".join(open(os.path.join(tdir, fn), "rb").read()
                        for fn in os.listdir(tdir))
        for k, v in kv.items():
            assert v not in blob and k not in blob
        em2 = EncryptionManager("secret", salt=b"s" * 16)
        s2 = LSMStore(tdir, crypt=em2)
        assert dict(s2.scan(b"name")) == kv
        s2.close()
        with pytest.raises(Exception):
            LSMStore(tdir, crypt=EncryptionManager("wrong", salt=b"s" * 16))


# ---------------------------------------------------------------------------
# DiskEngine
# ---------------------------------------------------------------------------

class TestDiskEngine:
    def test_graph_crud_and_restart(self, tdir):
        e = DiskEngine(tdir, memtable_bytes=8192)
        for i in range(50):
            e.create_node(Node(f"n{i}", ["P"] + (["Q"] if i % 3 == 0 else []),
                               {"i": i}))
        for i in range(49):
            e.create_edge(Edge(f"e{i}", "R", f"n{i}", f"n{i+1}", {}))
        assert e.node_count() == 50
        assert e.edge_count() == 49
        assert e.node_count_by_label("Q") == 17
        assert sorted(e.neighbors("n10")) == ["n11", "n9"]
        with pytest.raises(ConstraintViolation):
            e.create_node(Node("n0", [], {}))
        with pytest.raises(ConstraintViolation):
            e.delete_node("n10")  # has edges
        e.detach_delete_node("n10")
        assert e.node_count() == 49 and e.edge_count() == 47
        e.close()
        e2 = DiskEngine(tdir)
        assert e2.node_count() == 49 and e2.edge_count() == 47
        with pytest.raises(NotFoundError):
            e2.get_node("n10")
        assert e2.get_node("n11").properties["i"] == 11
        e2.close()

    def test_embedding_chunk_overflow(self, tdir):
        e = DiskEngine(tdir)
        e.create_node(Node("big", ["V"], {}))
        emb = [float(i) / 3 for i in range(30000)]  # 120 KB > 50 KB cap
        e.update_embedding("big", emb)
        # chunks on disk under prefix 0x08
        nch = sum(1 for _ in e._kv.scan(b"\x08"))
        assert nch == 4  # ceil(120KB / 32KB)
        e.close()
        e2 = DiskEngine(tdir)
        got = e2.get_node("big").embedding
        assert len(got) == 30000
        assert abs(got[12345] - emb[12345]) < 1e-6
        # shrinking the embedding removes stale chunks
        e2.update_embedding("big", [1.0] * 100)
        assert sum(1 for _ in e2._kv.scan(b"\x08")) == 0
        assert len(e2.get_node("big").embedding) == 100
        e2.close()

    def test_dataset_larger_than_caches(self, tdir):
        """Graph bigger than memtable + node cache: reads hit disk."""
        e = DiskEngine(tdir, memtable_bytes=16 << 10,
                       node_cache_bytes=32 << 10, block_cache_bytes=32 << 10)
        n = 2000
        for i in range(n):
            e.create_node(Node(f"n{i:05d}", ["D"], {"pad": "x" * 100, "i": i}))
        assert e.node_count() == n
        # far more data than the caches can hold; spot-check random reads
        random.seed(3)
        for _ in range(100):
            i = random.randrange(n)
            assert e.get_node(f"n{i:05d}").properties["i"] == i
        # streaming full scan
        assert sum(1 for _ in e.all_nodes()) == n
        st = e.stats()
        assert st["memtable_bytes"] <= 16 << 10
        e.close()
        # restart: no full-history replay (active log is small)
        e2 = DiskEngine(tdir, memtable_bytes=16 << 10)
        assert e2.node_count() == n
        assert e2.get_node("n00042").properties["i"] == 42
        e2.close()

    def test_transaction_commit_rollback(self, tdir):
        e = DiskEngine(tdir)
        tx = e.begin()
        tx.create_node(Node("a", [], {}))
        tx.create_node(Node("b", [], {}))
        tx.create_edge(Edge("ab", "T", "a", "b", {}))
        tx.commit()
        assert e.node_count() == 2 and e.edge_count() == 1
        tx2 = e.begin()
        tx2.create_node(Node("c", [], {}))
        tx2.create_node(Node("a", [], {}))  # duplicate -> rollback
        with pytest.raises(ConstraintViolation):
            tx2.commit()
        assert e.node_count() == 2  # "c" rolled back
        e.close()

    def test_property_index(self, tdir):
        e = DiskEngine(tdir)
        for i in range(20):
            e.create_node(Node(f"u{i}", ["U"], {"email": f"u{i}@x.io"}))
        e.create_property_index("U", "email")
        hits = e.lookup_property_index("U", "email", "u7@x.io")
        assert [n.id for n in hits] == ["u7"]
        n = e.get_node("u7")
        n.properties["email"] = "new@x.io"
        e.update_node(n)
        assert e.lookup_property_index("U", "email", "u7@x.io") == []
        assert [n.id for n in e.lookup_property_index("U", "email", "new@x.io")] == ["u7"]
        e.close()
        e2 = DiskEngine(tdir)  # defs persist
        assert [n.id for n in e2.lookup_property_index("U", "email", "new@x.io")] == ["u7"]
        e2.close()

    def test_backup_restore_engine(self, tdir, tmp_path):
        e = DiskEngine(tdir)
        e.create_node(Node("n1", ["B"], {"k": "v"}))
        e.create_node(Node("n2", ["B"], {}))
        e.create_edge(Edge("e1", "T", "n1", "n2", {}))
        bak = str(tmp_path / "db.backup")
        e.backup(bak)
        e.close()
        r = DiskEngine.restore(bak, str(tmp_path / "restored"))
        assert r.node_count() == 2 and r.edge_count() == 1
        assert r.get_node("n1").properties["k"] == "v"
        assert r.neighbors("n1") == ["n2"]
        r.close()

    def test_events_and_validators(self, tdir):
        e = DiskEngine(tdir)
        seen = []
        e.register_callback(lambda ev, obj: seen.append(ev))
        e.add_validator(lambda n, upd: (_ for _ in ()).throw(
            ConstraintViolation("no X")) if "X" in n.labels else None)
        e.create_node(Node("ok", ["Y"], {}))
        with pytest.raises(ConstraintViolation):
            e.create_node(Node("bad", ["X"], {}))
        assert seen == ["node_created"]
        assert e.node_count() == 1
        e.close()

    def test_open_db_disk_integration(self, tmp_path):
        from nornicdb_amd.db import open_db
        mgr = open_db(str(tmp_path / "d"), dims=8)
        db = mgr.get("neo4j")
        db.execute_cypher("CREATE (n:T {name:'x'})")
        r = db.execute_cypher("MATCH (n:T) RETURN n.name")
        assert r.rows == [["x"]]
        mgr.close()
        mgr2 = open_db(str(tmp_path / "d"), dims=8)
        r = mgr2.get("neo4j").execute_cypher("MATCH (n:T) RETURN n.name")
        assert r.rows == [["x"]]
        mgr2.close()

    def test_open_db_encrypted(self, tmp_path):
        from nornicdb_amd.db import open_db
        d = str(tmp_path / "enc")
        mgr = open_db(d, dims=8, encryption_passphrase="pw123")
        mgr.get("neo4j").execute_cypher("CREATE (n:Secret {code:'TOPSECRET'})")
        mgr.close()
        blob = b""
        for root, _, files in os.walk(d):
            for fn in files:
                blob += open(os.path.join(root, fn), "rb").read()
        assert b"TOPSECRET" not in blob
        mgr2 = open_db(d, dims=8, encryption_passphrase="pw123")
        r = mgr2.get("neo4j").execute_cypher("MATCH (n:Secret) RETURN n.code")
        assert r.rows == [["TOPSECRET"]]
        mgr2.close()
        with pytest.raises(Exception):
            open_db(d, dims=8, encryption_passphrase="wrong")


def test_vector_search_survives_reopen(tmp_path):
    """Reopening a persisted store rebuilds the embedding/fulltext
    indexes from stored embeddings — recall works immediately after a
    restart (regression: the index used to start empty)."""
    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder

    d = str(tmp_path / "store")
    mgr = open_db(d, embedder=MockEmbedder(16), dims=16)
    db = mgr.get()
    m = db.store("persistent memory about graph databases")
    db.store("unrelated note about cooking")
    db.embed_queue.drain()
    assert db.recall("graph databases", limit=1)[0].id == m.id
    mgr.close()

    mgr2 = open_db(d, embedder=MockEmbedder(16), dims=16)
    after = mgr2.get().recall("graph databases", limit=1)
    assert after and after[0].id == m.id
    # fulltext survives too
    ft = mgr2.get().cypher(
        "CALL db.index.fulltext.queryNodes('x', 'cooking') "
        "YIELD node RETURN node.content")
    assert any("cooking" in r[0] for r in ft.rows)
    mgr2.close()


def test_schema_survives_reopen(tmp_path):
    """Constraints/indexes persist (sidecar schema.json) AND enforce
    after a restart; enforcement also works through the DatabaseManager
    facade (regression: the namespaced wrapper dropped validators)."""
    import pytest as _pt

    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder
    from nornicdb_amd.storage.types import ConstraintViolation

    d = str(tmp_path / "store")
    mgr = open_db(d, embedder=MockEmbedder(8), dims=8)
    db = mgr.get()
    db.cypher("CREATE CONSTRAINT uq FOR (n:P) REQUIRE n.x IS UNIQUE")
    db.cypher("CREATE INDEX pidx FOR (n:P) ON (n.x)")
    db.cypher("CREATE (:P {x: 1})")
    with _pt.raises(Exception):
        db.cypher("CREATE (:P {x: 1})")
    mgr.close()

    mgr2 = open_db(d, embedder=MockEmbedder(8), dims=8)
    db2 = mgr2.get()
    assert len(db2.cypher("SHOW CONSTRAINTS").rows) == 1
    assert len(db2.cypher("SHOW INDEXES").rows) >= 1
    with _pt.raises(Exception):
        db2.cypher("CREATE (:P {x: 1})")
    db2.cypher("CREATE (:P {x: 2})")   # non-duplicates still fine
    mgr2.close()


def test_database_registry_survives_reopen(tmp_path):
    """User-created databases + aliases persist (system-namespace
    _Database/_DbAlias nodes) — without this, created databases vanished
    from the manager on restart while their data stayed orphaned."""
    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder

    d = str(tmp_path / "store")
    mgr = open_db(d, embedder=MockEmbedder(8), dims=8)
    mgr.create("analytics")
    mgr.alias("stats", "analytics")
    mgr.get("analytics").cypher("CREATE (:T {x: 1})")
    mgr.close()

    mgr2 = open_db(d, embedder=MockEmbedder(8), dims=8)
    assert "analytics" in mgr2.list()
    assert mgr2.get("stats").cypher(
        "MATCH (n:T) RETURN count(n)").rows == [[1]]
    mgr2.drop("analytics")
    mgr2.close()

    mgr3 = open_db(d, embedder=MockEmbedder(8), dims=8)
    assert "analytics" not in mgr3.list()
    mgr3.close()


def test_backup_restore_carries_schema(tmp_path):
    """Online backup + restore preserves constraints (sidecar
    schema.json travels with the backup file)."""
    import pytest as _pt

    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder
    from nornicdb_amd.storage.disk import DiskEngine

    d = str(tmp_path / "src")
    b = str(tmp_path / "backup.snap")
    r = str(tmp_path / "restored")
    mgr = open_db(d, embedder=MockEmbedder(8), dims=8)
    db = mgr.get()
    db.cypher("CREATE CONSTRAINT u FOR (n:P) REQUIRE n.x IS UNIQUE")
    db.cypher("CREATE (:P {x: 1})")
    mgr._base.backup(b)
    mgr.close()

    DiskEngine.restore(b, r).close()
    mgr2 = open_db(r, embedder=MockEmbedder(8), dims=8)
    db2 = mgr2.get()
    assert db2.cypher("MATCH (n:P) RETURN count(n)").rows == [[1]]
    assert len(db2.cypher("SHOW CONSTRAINTS").rows) == 1
    with _pt.raises(Exception):
        db2.cypher("CREATE (:P {x: 1})")
    mgr2.close()


def test_schema_sidecar_encrypted_at_rest(tmp_path):
    """With encryption on, the schema sidecar is ciphertext (no label
    names leak next to the encrypted store) and reloads correctly."""
    import os

    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder

    d = str(tmp_path / "store")
    mgr = open_db(d, embedder=MockEmbedder(8), dims=8,
                  encryption_passphrase="pw")
    mgr.get().cypher(
        "CREATE CONSTRAINT u FOR (n:Secret) REQUIRE n.k IS UNIQUE")
    mgr.close()
    raw = open(os.path.join(d, "schema.json"), "rb").read()
    assert b"Secret" not in raw
    mgr2 = open_db(d, embedder=MockEmbedder(8), dims=8,
                   encryption_passphrase="pw")
    assert len(mgr2.get().cypher("SHOW CONSTRAINTS").rows) == 1
    mgr2.close()


def test_typed_properties_roundtrip_reopen(tmp_path):
    """Temporal, spatial (cartesian + wgs-84), nested and unicode
    property values survive the disk codec and a reopen (regression:
    point() properties could not be stored at all)."""
    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder

    d = str(tmp_path / "store")
    mgr = open_db(d, embedder=MockEmbedder(8), dims=8)
    mgr.get().cypher(
        "CREATE (:X {dt: datetime('2026-01-02T03:04:05Z'), "
        "pt: point({x: 1.5, y: 2.5}), "
        "gp: point({longitude: 9.9, latitude: 48.4}), "
        "m: {a: [1, 2, {b: 'c'}]}, s: 'ünïcode ✓', f: 1.25})")
    mgr.close()
    mgr2 = open_db(d, embedder=MockEmbedder(8), dims=8)
    r = mgr2.get().cypher(
        "MATCH (n:X) RETURN n.dt.year, n.pt.x, n.gp.latitude, "
        "n.m.a[2].b, n.s, n.f")
    assert r.rows == [[2026, 1.5, 48.4, "c", "ünïcode ✓", 1.25]]
    r2 = mgr2.get().cypher(
        "MATCH (n:X) RETURN point.distance(n.pt, point({x: 1.5, y: 0.5}))")
    assert abs(r2.rows[0][0] - 2.0) < 1e-9
    mgr2.close()
