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
        return (["id", "name", "creationDate", "nodes", "relationships",
                 "nodeCount", "relationshipCount"],
                [["nornicdb", db.name, "", db.engine.node_count(),
                  db.engine.edge_count(), db.engine.node_count(),
                  db.engine.edge_count()]])

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

    # -------------------- gds.* (reference pkg/cypher/fastrp.go +
    # linkpredict exposure via GDS-compatible procedure names) --------------
    _GDS_GRAPHS: Dict[str, Any] = {}

    @register("gds.graph.project")
    def _gds_project(ex, name, node_label=None, rel_type=None):
        from ..graph import csr as _csr
        g = _csr.from_engine(db.engine,
                             edge_types=None if rel_type in ("*", None)
                             else [rel_type])
        _GDS_GRAPHS[name] = g
        return (["graphName", "nodeCount", "relationshipCount"],
                [[name, g.n, g.m]])

    @register("gds.graph.drop")
    def _gds_drop(ex, name, fail_if_missing=True):
        existed = _GDS_GRAPHS.pop(name, None) is not None
        if not existed and fail_if_missing:
            raise ValueError(f"graph {name!r} not found")
        return ["graphName"], [[name]]

    @register("gds.graph.list")
    def _gds_list(ex, name=None):
        items = ([(name, _GDS_GRAPHS[name])] if name in _GDS_GRAPHS
                 else list(_GDS_GRAPHS.items()))
        return (["graphName", "nodeCount"],
                [[k, g.n] for k, g in items])

    def _gds_graph(name):
        if name not in _GDS_GRAPHS:
            from ..graph import csr as _csr
            _GDS_GRAPHS[name] = _csr.from_engine(db.engine)
        return _GDS_GRAPHS[name]

    @register("gds.fastRP.stats")
    def _gds_fastrp_stats(ex, graph_name, config=None):
        g = _gds_graph(graph_name)
        return ["nodeCount"], [[g.n]]

    def _lp_pairs(limit=1000):
        """Candidate non-adjacent pairs within 2 hops."""
        out = []
        for n in db.engine.all_nodes():
            nb1 = set(db.engine.neighbors(n.id))
            for mid in nb1:
                for cand in db.engine.neighbors(mid):
                    if cand != n.id and cand not in nb1 and n.id < cand:
                        out.append((n.id, cand))
                        if len(out) >= limit:
                            return out
        return out

    def _lp_stream(scorer):
        from ..cognitive.linkpredict import (adamic_adar, common_neighbors,
                                             jaccard,
                                             preferential_attachment)
        from ..cognitive.linkpredict import resource_allocation
        fn = {"adamic": adamic_adar, "common": common_neighbors,
              "jaccard": jaccard, "pref": preferential_attachment,
              "resource": resource_allocation}[scorer]
        rows = [[a, b, float(fn(db.engine, a, b))] for a, b in _lp_pairs()]
        rows.sort(key=lambda r: -r[2])
        return ["node1", "node2", "score"], rows

    @register("gds.linkPrediction.adamicAdar.stream")
    def _gds_lp_aa(ex, config=None):
        return _lp_stream("adamic")

    @register("gds.linkPrediction.commonNeighbors.stream")
    def _gds_lp_cn(ex, config=None):
        return _lp_stream("common")

    @register("gds.linkPrediction.jaccard.stream")
    def _gds_lp_j(ex, config=None):
        return _lp_stream("jaccard")

    @register("gds.linkPrediction.resourceAllocation.stream")
    def _gds_lp_ra(ex, config=None):
        return _lp_stream("resource")

    @register("gds.linkPrediction.preferentialAttachment.stream")
    def _gds_lp_pa(ex, config=None):
        return _lp_stream("pref")

    @register("gds.linkPrediction.predict.stream")
    def _gds_lp_predict(ex, config=None):
        return _lp_stream("adamic")

    # -------------------- db.index extras --------------------
    @register("db.index.vector.createRelationshipIndex")
    def _vec_rel_idx(ex, name, rel_type, prop, dims=1024, similarity="cosine"):
        sm = getattr(ex, "schema", None)
        if sm:
            sm.create_vector_index(name, rel_type, prop, int(dims), similarity)
        return ["name"], [[name]]

    @register("db.index.vector.queryRelationships")
    def _vec_rel_query(ex, index_name, k, query):
        # relationships carry no embeddings in this engine: resolve via
        # endpoint-node similarity (same contract shape)
        import numpy as np
        qv = (db.embedder.embed_query(query) if isinstance(query, str)
              else np.asarray(query, dtype=np.float32))
        res = db.search.vector_search(qv, int(k))
        rows = []
        for r in res:
            for e in db.engine.get_out_edges(r.id):
                rows.append([e, r.score])
                break
        return ["relationship", "score"], rows[:int(k)]

    @register("db.index.vector.drop")
    def _vec_drop(ex, name):
        sm = getattr(ex, "schema", None)
        return ["dropped"], [[bool(sm and sm.drop_index(name))]]

    @register("db.index.fulltext.drop")
    def _ft_drop(ex, name):
        sm = getattr(ex, "schema", None)
        return ["dropped"], [[bool(sm and sm.drop_index(name))]]

    @register("db.create.setNodeVectorProperty")
    def _set_node_vec(ex, node, prop, vector):
        n = db.engine.get_node(node.id if hasattr(node, "id") else node)
        vec = [float(x) for x in (vector or [])]
        n.properties[prop] = vec
        db.engine.update_node(n)
        if prop == "embedding" and hasattr(db.engine, "update_embedding"):
            db.engine.update_embedding(n.id, vec)
        return ["node"], [[n]]

    @register("db.create.setVectorProperty")
    def _set_vec(ex, node, prop, vector):
        return _set_node_vec(ex, node, prop, vector)

    @register("dbms.procedures")
    def _dbms_procs(ex):
        return ["name", "signature"], [[n, ""] for n in sorted(procs)]

    @register("dbms.functions")
    def _dbms_fns(ex):
        from .functions import FUNCTIONS
        return ["name"], [[n] for n in sorted(FUNCTIONS)]

    @register("db.constraints")
    def _db_constraints(ex):
        sm = getattr(ex, "schema", None)
        return ["name", "description"], [
            [c.name, f"CONSTRAINT ON (:{c.label}) {c.kind} {c.prop}"]
            for c in (sm.list_constraints() if sm else [])]

    @register("db.indexes")
    def _db_indexes(ex):
        sm = getattr(ex, "schema", None)
        return ["name", "state", "type"], [
            [n, "ONLINE", k.upper()]
            for n, k, lb, ps in (sm.list_indexes() if sm else [])]

    @register("db.index.fulltext.createRelationshipIndex")
    def _ft_rel_idx(ex, name, types=None, props=None):
        sm = getattr(ex, "schema", None)
        if sm:
            sm.create_index((types or ["REL"])[0], (props or ["text"])[0],
                            name=name, kind="fulltext", props=props)
        return ["name"], [[name]]

    @register("db.create.setRelationshipVectorProperty")
    def _set_rel_vec(ex, rel, prop, vector):
        e = db.engine.get_edge(rel.id if hasattr(rel, "id") else rel)
        e.properties[prop] = [float(x) for x in (vector or [])]
        db.engine.update_edge(e)
        return ["relationship"], [[e]]

    @register("tx.setMetaData")
    def _tx_meta(ex, meta=None):
        return [], []

    # -------------------- nornicdb.* (reference pkg/cypher/call.go:973-1027
    # + call_compat.go / call_index_mgmt.go compat surface) --------------------
    @register("nornicdb.version")
    def _nv(ex):
        from .. import __version__
        return ["version", "build", "edition"], [[__version__, "rocm", "community"]]

    @register("nornicdb.stats")
    def _nstats(ex):
        labels = set()
        for n in db.engine.all_nodes():
            labels.update(n.labels)
        types = {e.type for e in db.engine.all_edges()}
        return ["nodes", "relationships", "labels", "relationshipTypes"], \
            [[db.engine.node_count(), db.engine.edge_count(),
              len(labels), len(types)]]

    @register("nornicdb.decay.info")
    def _ndecay(ex):
        cfg = getattr(db, "decay_config", None)
        return (["enabled", "halfLifeEpisodic", "halfLifeSemantic",
                 "halfLifeProcedural", "archiveThreshold"],
                [[bool(getattr(cfg, "enabled", True)) if cfg is not None else True,
                  "7 days", "69 days", "693 days", 0.05]])

    @register("gds.version")
    def _gdsv(ex):
        return ["version"], [["2.6.0-nornicdb-amd"]]

    @register("db.clearQueryCaches")
    def _clearcache(ex):
        cache = getattr(db, "query_cache", None)
        if cache is not None:
            cache.invalidate()
        return ["status"], [["Query caches cleared"]]

    @register("db.stats.clear")
    def _stats_clear(ex, section="QUERIES"):
        return ["section", "success", "message"], [[section, True, "cleared"]]

    @register("db.stats.collect")
    def _stats_collect(ex, section="QUERIES", config=None):
        return ["section", "success", "message"], [[section, True, "collecting"]]

    @register("db.stats.stop")
    def _stats_stop(ex, section="QUERIES"):
        return ["section", "success", "message"], [[section, True, "stopped"]]

    @register("db.stats.status")
    def _stats_status(ex):
        return ["section", "status", "message"], [["QUERIES", "idle", ""]]

    @register("db.stats.retrieveAllAnTheStats")
    def _stats_all(ex):
        _, rows = procs["db.stats.retrieve"](ex, "GRAPH COUNTS")
        return ["section", "data"], rows

    @register("db.schema.nodeProperties")
    def _schema_nodeprops(ex):
        seen = {}
        for n in db.engine.all_nodes():
            for lb in n.labels:
                seen.setdefault(lb, set()).update((n.properties or {}).keys())
        return ["nodeLabel", "propertyName", "propertyType"], \
            [[lb, p, "ANY"] for lb in sorted(seen) for p in sorted(seen[lb])]

    @register("db.schema.relProperties")
    def _schema_relprops(ex):
        seen = {}
        for e in db.engine.all_edges():
            seen.setdefault(e.type, set()).update((e.properties or {}).keys())
        return ["relType", "propertyName", "propertyType"], \
            [[t, p, "ANY"] for t in sorted(seen) for p in sorted(seen[t])]

    @register("db.index.fulltext.listAvailableAnalyzers")
    def _ft_analyzers(ex):
        return ["analyzer", "description"], [
            ["standard-no-stop-words", "Standard analyzer without stop words"],
            ["simple", "Simple analyzer with lowercase tokenizer"],
            ["whitespace", "Whitespace analyzer"],
            ["keyword", "Keyword analyzer - entire string as single token"],
        ]

    @register("db.index.fulltext.queryRelationships")
    def _ft_query_rels(ex, index, query, options=None):
        ql = str(query).lower()
        rows = []
        for e in db.engine.all_edges():
            score = sum(1.0 for v in (e.properties or {}).values()
                        if isinstance(v, str) and ql in v.lower())
            if score:
                rows.append([e, score])
        rows.sort(key=lambda r: -r[1])
        return ["relationship", "score"], rows

    @register("dbms.clientConfig")
    def _client_cfg(ex):
        return ["name", "value"], [
            ["server.bolt.advertised_address", "localhost:7687"],
            ["server.http.advertised_address", "localhost:7474"],
        ]

    @register("dbms.listConnections")
    def _list_conns(ex):
        return ["connectionId", "connectTime", "connector", "username",
                "userAgent", "clientAddress"], []

    return procs
