"""DB facade tests: memory API, embed queue, procedures, multidb.

Models reference pkg/nornicdb tests (db_test.go, embed_queue behavior,
multi-database e2e).
"""

import time

import pytest

from nornicdb_amd.db import DatabaseManager, NornicDB, open_db
from nornicdb_amd.embed import CachedEmbedder, MockEmbedder, chunk_text
from nornicdb_amd.storage import MemoryEngine


@pytest.fixture
def mgr():
    m = open_db(embedder=MockEmbedder(32), dims=32)
    yield m
    m.close()


class TestMemoryAPI:
    def test_store_recall(self, mgr):
        db = mgr.get()
        m1 = db.store("the mitochondria is the powerhouse of the cell",
                      title="biology")
        db.store("graph databases store nodes and edges", title="databases")
        assert db.embed_queue.drain()
        res = db.recall("mitochondria cell powerhouse", limit=2)
        assert res and res[0].id == m1.id
        assert res[0].access_count >= 0

    def test_remember_touch(self, mgr):
        db = mgr.get()
        m = db.store("remember me")
        got = db.remember(m.id)
        assert got.content == "remember me"
        got2 = db.remember(m.id)
        assert got2.access_count >= 1

    def test_link_neighbors_forget(self, mgr):
        db = mgr.get()
        a = db.store("memory a")
        b = db.store("memory b")
        c = db.store("memory c")
        db.link(a.id, b.id, "RELATES_TO")
        db.link(b.id, c.id, "CAUSED_BY")
        nb = db.neighbors(a.id, depth=1)
        assert [m.id for m in nb] == [b.id]
        nb2 = db.neighbors(a.id, depth=2)
        assert {m.id for m in nb2} == {b.id, c.id}
        assert db.forget(b.id)
        assert db.neighbors(a.id) == []

    def test_memory_tiers(self, mgr):
        db = mgr.get()
        db.store("fact", memory_type="semantic")
        db.store("event", memory_type="episodic")
        r = db.cypher("MATCH (n:Semantic) RETURN n.content")
        assert r.rows == [["fact"]]


class TestEmbedQueue:
    def test_auto_embed_pipeline(self, mgr):
        db = mgr.get()
        m = db.store("auto embedded content")
        assert m.id in db.engine.pending_embeddings()
        assert db.embed_queue.drain()
        node = db.engine.get_node(m.id)
        assert node.embedding is not None and len(node.embedding) == 32
        # indexed in the vector index
        qv = db.embedder.embed_query("auto embedded content")
        res = db.search.vector_search(qv, 1)
        assert res[0].id == m.id

    def test_chunk_text(self):
        words = " ".join(f"w{i}" for i in range(1200))
        chunks = chunk_text(words, chunk_tokens=512, overlap=50)
        assert len(chunks) >= 3
        assert chunks[0].split()[0] == "w0"
        # overlap: chunk 2 starts 462 words in
        assert chunks[1].split()[0] == "w462"


class TestProcedures:
    def test_vector_query_nodes_string_autoembed(self, mgr):
        db = mgr.get()
        db.store("vector searchable text")
        db.embed_queue.drain()
        r = db.cypher(
            "CALL db.index.vector.queryNodes('idx', 5, 'vector searchable text') "
            "YIELD node, score RETURN node.content, score")
        assert r.rows[0][0] == "vector searchable text"
        assert r.rows[0][1] > 0.9

    def test_introspection(self, mgr):
        db = mgr.get()
        db.cypher("CREATE (:A {x: 1})-[:REL]->(:B {y: 2})")
        assert ["A"] in db.cypher("CALL db.labels()").rows
        assert db.cypher("CALL db.relationshipTypes()").rows == [["REL"]]
        keys = [r[0] for r in db.cypher("CALL db.propertyKeys()").rows]
        assert "x" in keys and "y" in keys

    def test_dbms_components(self, mgr):
        r = mgr.get().cypher("CALL dbms.components() YIELD name, edition "
                             "RETURN name, edition")
        assert r.rows[0][0] == "NornicDB-AMD"


class TestMultiDB:
    def test_isolation(self, mgr):
        mgr.create("tenant1")
        db1 = mgr.get("tenant1")
        db0 = mgr.get()
        db1.cypher("CREATE (:T {v: 1})")
        assert db0.cypher("MATCH (n:T) RETURN count(n)").rows == [[0]]
        assert db1.cypher("MATCH (n:T) RETURN count(n)").rows == [[1]]

    def test_create_drop_list(self, mgr):
        mgr.create("x")
        assert "x" in mgr.list()
        mgr.drop("x")
        assert "x" not in mgr.list()
        with pytest.raises(ValueError):
            mgr.drop("system")

    def test_alias(self, mgr):
        mgr.create("real")
        mgr.alias("nick", "real")
        assert mgr.get("nick") is mgr.get("real")


class TestPersistentFacade:
    def test_reopen_with_data(self, tmp_path):
        mgr = open_db(str(tmp_path / "d"), embedder=MockEmbedder(16), dims=16,
                      durable_sync=True)
        db = mgr.get()
        db.store("durable memory")
        db.embed_queue.drain()
        mgr.close()

        mgr2 = open_db(str(tmp_path / "d"), embedder=MockEmbedder(16), dims=16)
        db2 = mgr2.get()
        r = db2.cypher("MATCH (n:Memory) RETURN n.content")
        assert r.rows == [["durable memory"]]
        # embeddings survived and are searchable after index rebuild
        db2.search.build_indexes()
        res = db2.recall("durable memory")
        assert res and res[0].content == "durable memory"
        mgr2.close()


class TestCachedEmbedder:
    def test_cache_hit(self):
        inner = MockEmbedder(8)
        ce = CachedEmbedder(inner, capacity=2)
        v1 = ce.embed("hello")
        v2 = ce.embed("hello")
        assert (v1 == v2).all()
        assert ce.hits == 1 and ce.misses == 1
        ce.embed("a")
        ce.embed("b")  # evicts "hello"
        ce.embed("hello")
        assert ce.misses == 4


class TestQueryCache:
    def test_read_cached_and_invalidated(self, mgr):
        db = mgr.get()
        db.cypher("CREATE (:QC {v: 1})")
        r1 = db.cypher("MATCH (n:QC) RETURN n.v")
        assert db.query_cache.hits == 0
        r2 = db.cypher("MATCH (n:QC) RETURN n.v")
        assert db.query_cache.hits == 1
        assert r2.rows == r1.rows
        # a write invalidates
        db.cypher("CREATE (:QC {v: 2})")
        r3 = db.cypher("MATCH (n:QC) RETURN count(n)")
        assert r3.rows == [[2]]

    def test_write_queries_not_cached(self, mgr):
        db = mgr.get()
        db.cypher("CREATE (:W1)")
        db.cypher("CREATE (:W1)")
        assert db.cypher("MATCH (n:W1) RETURN count(n)").rows == [[2]]


def test_default_tokenizer_is_trained_bpe():
    """HashTokenizer must no longer be the default —
    the shipped BPE artifact drives the text->tokens->vector path."""
    from nornicdb_amd.embed.tokenizer import (HFTokenizer, default_tokenizer,
                                              BOS, EOS, PAD)
    tok = default_tokenizer()
    assert isinstance(tok, HFTokenizer), type(tok)
    ids = tok.encode("NornicDB stores graph memories with vector search.")
    assert ids[0] == BOS and ids[-1] == EOS
    assert all(0 <= i < 250002 for i in ids)
    # subword merges learned: common English words are single tokens
    short = tok.encode("the")
    assert len(short) <= 4
    batch_ids, mask = tok.encode_batch(["a tiny text", "a much longer text "
                                        "with several additional words"])
    assert len(batch_ids[0]) == len(batch_ids[1])
    assert mask[0][-1] == 0 and batch_ids[0][-1] == PAD
    # deterministic
    assert tok.encode("same input") == tok.encode("same input")
