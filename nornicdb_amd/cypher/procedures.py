"""Built-in CALL procedures.

Parity: reference pkg/cypher/call.go dispatch + call_vector.go:35
(db.index.vector.queryNodes with string auto-embed), call_fulltext.go,
call_index_mgmt.go, and the db.*/dbms.* introspection procedures.

A procedure is fn(executor, *args) -> (columns, rows).
"""

from __future__ import annotations

from typing import TYPE_CHECKING, Any, Dict, List

import numpy as np

if TYPE_CHECKING:
    from ..db import NornicDB


def build_procedures(db: "NornicDB") -> Dict[str, Any]:
    procs: Dict[str, Any] = {}

    def register(name):
        def deco(fn):
            procs[name.lower()] = fn
            return fn
        return deco

    @register("db.index.vector.queryNodes")
    def vector_query(ex, index_name, k, query, *rest):
        # string query -> auto-embed (reference call_vector.go string input)
        if isinstance(query, str):
            qv = db.embedder.embed_query(query)
        else:
            qv = np.asarray(query, dtype=np.float32)
        res = db.search.vector_search(qv, int(k))
        return ["node", "score"], [[r.node, r.score] for r in res]

    @register("db.index.vector.createNodeIndex")
    def vector_create(ex, name, label, prop, dims, similarity="cosine"):
        from ..search.vectorspace import GLOBAL, VectorSpace
        GLOBAL.register(VectorSpace(db.name, label, name, int(dims), similarity))
        return ["name"], [[name]]

    @register("db.index.fulltext.queryNodes")
    def fulltext_query(ex, index_name, query, *rest):
        res = db.search.text_search(query, 25)
        return ["node", "score"], [[r.node, r.score] for r in res]

    @register("db.index.fulltext.createNodeIndex")
    def fulltext_create(ex, name, labels=None, props=None):
        return ["name"], [[name]]

    @register("db.labels")
    def labels(ex):
        seen = set()
        for n in db.engine.all_nodes():
            seen.update(n.labels)
        return ["label"], [[lb] for lb in sorted(seen)]

    @register("db.relationshipTypes")
    def rel_types(ex):
        seen = set()
        for e in db.engine.all_edges():
            seen.add(e.type)
        return ["relationshipType"], [[t] for t in sorted(seen)]

    @register("db.propertyKeys")
    def prop_keys(ex):
        seen = set()
        for n in db.engine.all_nodes():
            seen.update(n.properties.keys())
        for e in db.engine.all_edges():
            seen.update(e.properties.keys())
        return ["propertyKey"], [[k] for k in sorted(seen)]

    @register("db.indexes")
    def indexes(ex):
        from ..search.vectorspace import GLOBAL
        rows = []
        for s in GLOBAL.list(db.name):
            rows.append([s.name, "VECTOR", s.entity_type, s.dims, s.distance])
        return ["name", "type", "label", "dims", "similarity"], rows

    @register("dbms.components")
    def components(ex):
        from .. import __version__
        return (["name", "versions", "edition"],
                [["NornicDB-AMD", [__version__], "mi355x"]])

    @register("db.info")
    def info(ex):
        return (["name", "nodes", "relationships"],
                [[db.name, db.engine.node_count(), db.engine.edge_count()]])

    @register("db.ping")
    def ping(ex):
        return ["success"], [[True]]

    @register("nornic.search")
    def nornic_search(ex, query, k=10):
        res = db.search.search(query=query, k=int(k))
        return ["node", "score"], [[r.node, r.score] for r in res]

    @register("nornic.recluster")
    def recluster(ex, k=None):
        db.search.recluster(int(k) if k else None)
        return ["clusters"], [[db.search.clusters.k]]

    return procs
