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

    @register("dbms.routing.getRoutingTable")
    def _routing_table(ex, context=None, database=None):
        # single-server topology (drivers use this for session routing)
        addr = "localhost:7687"
        servers = [{"addresses": [addr], "role": r}
                   for r in ("WRITE", "READ", "ROUTE")]
        return ["ttl", "servers"], [[300, servers]]

    @register("db.schema.visualization")
    def _schema_viz(ex):
        from ..storage.types import Node as _N
        labels = sorted({lb for n in db.engine.all_nodes()
                         for lb in n.labels})
        vnodes = {lb: _N(id=f"schema:{lb}", labels=[lb],
                         properties={"name": lb}) for lb in labels}
        vrels = []
        seen = set()
        for e in db.engine.all_edges():
            try:
                a = db.engine.get_node(e.start_node)
                b = db.engine.get_node(e.end_node)
            except Exception:
                continue
            for la in a.labels:
                for lb2 in b.labels:
                    key = (la, e.type, lb2)
                    if key not in seen:
                        seen.add(key)
                        vrels.append({"type": e.type, "from": la, "to": lb2})
        return ["nodes", "relationships"], [[list(vnodes.values()), vrels]]

    @register("db.schema.nodeTypeProperties")
    def _schema_ntp(ex):
        from collections import defaultdict
        props = defaultdict(set)
        for n in db.engine.all_nodes():
            for lb in n.labels:
                for k, v in n.properties.items():
                    props[(lb, k)].add(type(v).__name__)
        return (["nodeType", "nodeLabels", "propertyName", "propertyTypes",
                 "mandatory"],
                [[f":`{lb}`", [lb], k, sorted(ts), False]
                 for (lb, k), ts in sorted(props.items())])

    @register("db.schema.relTypeProperties")
    def _schema_rtp(ex):
        from collections import defaultdict
        props = defaultdict(set)
        for e in db.engine.all_edges():
            for k, v in e.properties.items():
                props[(e.type, k)].add(type(v).__name__)
        return (["relType", "propertyName", "propertyTypes", "mandatory"],
                [[f":`{t}`", k, sorted(ts), False]
                 for (t, k), ts in sorted(props.items())])

    @register("db.awaitIndexes")
    def _await_indexes(ex, timeout=300):
        return [], []

    @register("db.awaitIndex")
    def _await_index(ex, name=None, timeout=300):
        return [], []

    @register("db.resampleIndex")
    def _resample_index(ex, name=None):
        return [], []

    @register("db.resampleOutdatedIndexes")
    def _resample_outdated(ex):
        return [], []

    @register("dbms.listConfig")
    def _list_config(ex, search=None):
        from ..utils.config import Config
        cfg = Config()
        rows = [[k, str(getattr(cfg, k))] for k in sorted(vars(cfg))
                if not k.startswith("_")]
        if search:
            rows = [r for r in rows if search.lower() in r[0].lower()]
        return ["name", "value"], rows

    @register("dbms.info")
    def _dbms_info(ex):
        return ["id", "name", "creationDate"], [["nornicdb-amd", db.name, ""]]

    @register("db.stats.retrieve")
    def _stats_retrieve(ex, section="GRAPH COUNTS"):
        from collections import Counter
        lc = Counter()
        for n in db.engine.all_nodes():
            lc.update(n.labels)
        tc = Counter(e.type for e in db.engine.all_edges())
        data = {"nodes": db.engine.node_count(),
                "relationships": db.engine.edge_count(),
                "labels": dict(lc), "relTypes": dict(tc)}
        return ["section", "data"], [[section, data]]

    return procs
