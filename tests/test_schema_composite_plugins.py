"""Schema constraints, composite engine, plugin loader, query cache."""

import pytest

from nornicdb_amd.storage import (CompositeEngine, ConstraintViolation,
                                  MemoryEngine, Node, Edge, SchemaManager)


class TestSchema:
    def test_unique_constraint_enforced(self):
        eng = MemoryEngine()
        sm = SchemaManager(eng)
        sm.create_unique_constraint("uq", "User", "email")
        eng.create_node(Node("a", ["User"], {"email": "x@y.z"}))
        with pytest.raises(ConstraintViolation):
            eng.create_node(Node("b", ["User"], {"email": "x@y.z"}))
        eng.create_node(Node("c", ["User"], {"email": "other"}))
        # update into violation also blocked
        n = eng.get_node("c")
        n.properties["email"] = "x@y.z"
        with pytest.raises(ConstraintViolation):
            eng.update_node(n)

    def test_unique_constraint_rejects_existing_dupes(self):
        eng = MemoryEngine()
        eng.create_node(Node("a", ["U"], {"k": 1}))
        eng.create_node(Node("b", ["U"], {"k": 1}))
        sm = SchemaManager(eng)
        with pytest.raises(ConstraintViolation):
            sm.create_unique_constraint("uq", "U", "k")

    def test_exists_constraint(self):
        eng = MemoryEngine()
        sm = SchemaManager(eng)
        sm.create_exists_constraint("ex", "Doc", "title")
        with pytest.raises(ConstraintViolation):
            eng.create_node(Node("a", ["Doc"], {}))
        eng.create_node(Node("b", ["Doc"], {"title": "ok"}))
        assert len(sm.list_constraints()) == 1


class TestComposite:
    def _mk(self):
        a, b = MemoryEngine(), MemoryEngine()
        comp = CompositeEngine({"dba": a, "dbb": b}, default="dba",
                               label_routes={"B": "dbb"})
        return comp, a, b

    def test_label_routing(self):
        comp, a, b = self._mk()
        comp.create_node(Node("x", ["A"], {}))
        comp.create_node(Node("y", ["B"], {}))
        assert a.node_count() == 1 and b.node_count() == 1
        assert comp.node_count() == 2

    def test_fanout_reads_and_gid(self):
        comp, a, b = self._mk()
        n1 = comp.create_node(Node("x", ["A"], {"v": 1}))
        n2 = comp.create_node(Node("x", ["B"], {"v": 2}))
        assert n1.id == "dba:x" and n2.id == "dbb:x"
        assert comp.get_node("dba:x").properties["v"] == 1
        assert comp.get_node("dbb:x").properties["v"] == 2
        assert len(comp.get_nodes_by_label("A")) == 1

    def test_edges_stay_within_constituent(self):
        comp, a, b = self._mk()
        comp.create_node(Node("x", ["A"], {}))
        comp.create_node(Node("y", ["A"], {}))
        comp.create_node(Node("z", ["B"], {}))
        comp.create_edge(Edge("e1", "R", "dba:x", "dba:y"))
        assert comp.edge_count() == 1
        with pytest.raises(Exception):
            comp.create_edge(Edge("e2", "R", "dba:x", "dbb:z"))
        assert comp.neighbors("dba:x") == ["dba:y"]


class TestPlugins:
    def test_load_plugins(self, tmp_path):
        plug = tmp_path / "myplug.py"
        plug.write_text('''
def register_functions(register):
    register("my.double", lambda x: x * 2)

def register_procedures(db):
    def hello(ex):
        return ["msg"], [["hi from plugin"]]
    return {"my.hello": hello}
''')
        from nornicdb_amd.db import open_db
        from nornicdb_amd.embed import MockEmbedder
        from nornicdb_amd.plugins import load_plugins_from_dir
        mgr = open_db(embedder=MockEmbedder(8), dims=8)
        db = mgr.get()
        loaded = load_plugins_from_dir(db, str(tmp_path))
        assert loaded["modules"] == ["myplug.py"]
        assert db.cypher("RETURN my.double(21)").rows == [[42]]
        assert db.cypher("CALL my.hello() YIELD msg RETURN msg").rows == [["hi from plugin"]]
        mgr.close()
