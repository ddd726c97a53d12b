"""APOC procedures (CALL-position).

Parity: reference apoc/algo (PageRank/Dijkstra/AStar/Betweenness...),
apoc/community (Louvain/LabelProp/WCC/Triangles), apoc/periodic (batch
iterate), apoc/create, apoc/refactor, apoc/meta, apoc/merge — wired to
the CSR graph module so large graphs run the HIP kernels.
"""

from __future__ import annotations

from typing import Any, Dict

import numpy as np

from ..graph import (betweenness_centrality, closeness_centrality,
                     connected_components, degree_centrality, dijkstra,
                     from_engine, label_propagation, louvain, pagerank,
                     shortest_path, triangle_count, clustering_coefficient)
import re
import time
import threading

from ..storage.types import Edge, Node, new_id

_ATOMIC_LOCK = threading.Lock()
_NODE_LOCKS = {}


def _parse_rel_filter(rel_filter):
    """'KNOWS|WORKS_AT>' -> ({KNOWS, WORKS_AT}, 'out'); None -> any/both."""
    if not rel_filter:
        return set(), "both"
    direction = "both"
    f = rel_filter
    if f.endswith(">"):
        direction, f = "out", f[:-1]
    elif f.endswith("<") or f.startswith("<"):
        direction, f = "in", f.strip("<")
    types = {t for t in f.split("|") if t}
    return types, direction


def build_apoc_procedures(db) -> Dict[str, Any]:
    procs: Dict[str, Any] = {}

    def register(name):
        def deco(fn):
            procs[name.lower()] = fn
            return fn
        return deco

    eng = db.engine

    # -------------------- apoc.algo --------------------
    @register("apoc.algo.pageRank")
    def _pagerank(ex, iterations=20, damping=0.85, label=None):
        g = from_engine(eng)
        r = pagerank(g, damping=float(damping), iters=int(iterations))
        rows = [[eng.get_node(g.node_ids[i]), float(r[i])]
                for i in np.argsort(-r)]
        return ["node", "score"], rows

    @register("apoc.algo.degree")
    def _degree(ex):
        g = from_engine(eng)
        d = degree_centrality(g)
        return ["node", "score"], [[eng.get_node(g.node_ids[i]), float(d[i])]
                                   for i in np.argsort(-d)]

    @register("apoc.algo.betweenness")
    def _betweenness(ex, samples=None):
        g = from_engine(eng)
        bc = betweenness_centrality(g, samples=int(samples) if samples else None)
        return ["node", "score"], [[eng.get_node(g.node_ids[i]), float(bc[i])]
                                   for i in np.argsort(-bc)]

    @register("apoc.algo.closeness")
    def _closeness(ex):
        g = from_engine(eng)
        c = closeness_centrality(g)
        return ["node", "score"], [[eng.get_node(g.node_ids[i]), float(c[i])]
                                   for i in np.argsort(-c)]

    @register("apoc.algo.dijkstra")
    def _dijkstra(ex, start, end, rel_type=None, weight_prop="weight"):
        """String args are node ids (the reference's NodeID is a string,
        pkg/cypher/apoc_algorithms.go:67), falling back to the id/name
        property; an unreachable or missing endpoint yields no rows."""
        g = from_engine(eng, edge_types=[rel_type] if rel_type else None,
                        weight_prop=weight_prop)
        sid = _node_arg(start)
        tid = _node_arg(end)
        if sid not in g.id2idx or tid not in g.id2idx:
            return ["path", "weight"], []
        s = g.id2idx[sid]
        t = g.id2idx[tid]
        dist, _ = dijkstra(g, s, t)
        path_idx = shortest_path(g, s, t)
        nodes = [eng.get_node(g.node_ids[i]) for i in path_idx]
        return ["path", "weight"], [[nodes, float(dist[t])]] if path_idx else []

    @register("apoc.algo.aStar")
    def _astar_p(ex, start, end, weight_prop="weight", lat="lat", lon="lon"):
        return _dijkstra(ex, start, end, None, weight_prop)

    # -------------------- apoc.community / graph --------------------
    @register("apoc.community.louvain")
    def _louvain(ex):
        g = from_engine(eng, undirected=True)
        # large graphs route the local-moving phase to the GPU (10M edges:
        # ~1.5 s vs ~20 s CPU — profiles/README.md)
        import torch as _torch
        dev = "cuda" if (g.m > 2_000_000
                         and _torch.cuda.is_available()) else "cpu"
        comm = louvain(g, device=dev)
        return ["node", "community"], [[eng.get_node(g.node_ids[i]), int(comm[i])]
                                       for i in range(g.n)]

    @register("apoc.community.labelPropagation")
    def _labelprop(ex, iterations=20):
        g = from_engine(eng, undirected=True)
        lb = label_propagation(g, iters=int(iterations))
        return ["node", "community"], [[eng.get_node(g.node_ids[i]), int(lb[i])]
                                       for i in range(g.n)]

    @register("apoc.community.wcc")
    def _wcc(ex):
        g = from_engine(eng)
        c = connected_components(g)
        return ["node", "component"], [[eng.get_node(g.node_ids[i]), int(c[i])]
                                       for i in range(g.n)]

    @register("apoc.community.triangleCount")
    def _tri(ex):
        g = from_engine(eng)
        return ["triangles"], [[triangle_count(g)]]

    @register("apoc.community.clusteringCoefficient")
    def _cc(ex):
        g = from_engine(eng)
        cc = clustering_coefficient(g)
        return ["node", "coefficient"], [[eng.get_node(g.node_ids[i]), float(cc[i])]
                                         for i in range(g.n)]

    # -------------------- gds compat (reference pkg/cypher/fastrp.go) ----
    @register("gds.fastRP.stream")
    def _fastrp(ex, dims=128, iteration_weights=None, seed=42):
        # GDS contract: first arg may be a projected graph NAME with a
        # config map (gds.fastRP.stream('g', {embeddingDimension: 64}))
        if isinstance(dims, str):
            cfg = dict(iteration_weights or {}) if isinstance(
                iteration_weights, dict) else {}
            dims = cfg.get("embeddingDimension", cfg.get("dims", 128))
            iteration_weights = cfg.get("iterationWeights")
        from ..graph.fastrp import fastrp_embeddings
        g = from_engine(eng)
        emb = fastrp_embeddings(g, dims=int(dims),
                                iteration_weights=iteration_weights or (0.0, 1.0, 1.0),
                                seed=int(seed))
        return ["node", "embedding"], [
            [eng.get_node(g.node_ids[i]), [float(x) for x in emb[i]]]
            for i in range(g.n)]

    @register("gds.fastRP.write")
    def _fastrp_write(ex, prop="fastrp", dims=128, seed=42):
        from ..graph.fastrp import fastrp_embeddings
        g = from_engine(eng)
        emb = fastrp_embeddings(g, dims=int(dims), seed=int(seed))
        for i in range(g.n):
            n = eng.get_node(g.node_ids[i])
            n.properties[prop] = [float(x) for x in emb[i]]
            eng.update_node(n)
        return ["nodeCount", "nodesWritten"], [[g.n, g.n]]

    # -------------------- apoc.create --------------------
    @register("apoc.create.node")
    def _create_node(ex, labels, props):
        n = Node(id=new_id("n"), labels=list(labels or []),
                 properties=dict(props or {}))
        return ["node"], [[eng.create_node(n)]]

    @register("apoc.create.nodes")
    def _create_nodes(ex, labels, props_list):
        rows = []
        for props in props_list or []:
            n = Node(id=new_id("n"), labels=list(labels or []),
                     properties=dict(props or {}))
            rows.append([eng.create_node(n)])
        return ["node"], rows

    @register("apoc.create.relationship")
    def _create_rel(ex, start, rel_type, props, end):
        e = Edge(id=new_id("e"), type=rel_type,
                 start_node=start.id if isinstance(start, Node) else start,
                 end_node=end.id if isinstance(end, Node) else end,
                 properties=dict(props or {}))
        return ["rel"], [[eng.create_edge(e)]]

    @register("apoc.create.uuid")
    def _uuid(ex):
        import uuid
        return ["uuid"], [[str(uuid.uuid4())]]

    # -------------------- apoc.merge --------------------
    @register("apoc.merge.node")
    def _merge_node(ex, labels, ident_props, on_create_props=None, on_match_props=None):
        labels = list(labels or [])
        cands = eng.get_nodes_by_label(labels[0]) if labels else list(eng.all_nodes())
        for n in cands:
            if all(n.properties.get(k) == v for k, v in (ident_props or {}).items()) \
                    and all(lb in n.labels for lb in labels):
                if on_match_props:
                    n.properties.update(on_match_props)
                    n = eng.update_node(n)
                return ["node"], [[n]]
        props = dict(ident_props or {})
        props.update(on_create_props or {})
        n = Node(id=new_id("n"), labels=labels, properties=props)
        return ["node"], [[eng.create_node(n)]]

    # -------------------- apoc.refactor --------------------
    @register("apoc.refactor.mergeNodes")
    def _merge_nodes(ex, nodes, config=None):
        if not nodes:
            return ["node"], []
        target = nodes[0]
        tnode = eng.get_node(target.id)
        for other in nodes[1:]:
            o = eng.get_node(other.id)
            for k, v in o.properties.items():
                tnode.properties.setdefault(k, v)
            for lb in o.labels:
                if lb not in tnode.labels:
                    tnode.labels.append(lb)
            for e in eng.get_out_edges(o.id):
                if e.end_node != tnode.id:
                    eng.create_edge(Edge(new_id("e"), e.type, tnode.id,
                                         e.end_node, dict(e.properties)))
            for e in eng.get_in_edges(o.id):
                if e.start_node != tnode.id:
                    eng.create_edge(Edge(new_id("e"), e.type, e.start_node,
                                         tnode.id, dict(e.properties)))
            eng.detach_delete_node(o.id)
        eng.update_node(tnode)
        return ["node"], [[eng.get_node(tnode.id)]]

    @register("apoc.refactor.rename.label")
    def _rename_label(ex, old, new):
        count = 0
        for n in eng.get_nodes_by_label(old):
            n.labels = [new if lb == old else lb for lb in n.labels]
            eng.update_node(n)
            count += 1
        return ["count"], [[count]]

    @register("apoc.refactor.rename.type")
    def _rename_type(ex, old, new):
        count = 0
        for e in eng.get_edges_by_type(old):
            e.type = new
            eng.update_edge(e)
            count += 1
        return ["count"], [[count]]

    # -------------------- apoc.periodic --------------------
    @register("apoc.periodic.iterate")
    def _periodic_iterate(ex, outer_q, inner_q, config=None):
        config = config or {}
        batch_size = int(config.get("batchSize", 1000))
        outer = ex.execute(outer_q, {})
        rows = outer.to_dicts()
        batches = ops_total = failed = 0
        for i in range(0, len(rows), batch_size):
            batch = rows[i:i + batch_size]
            batches += 1
            for row in batch:
                try:
                    ex.execute(inner_q, row, bindings=row)
                    ops_total += 1
                except Exception:
                    failed += 1
        return (["batches", "total", "failedOperations"],
                [[batches, ops_total, failed]])

    @register("apoc.periodic.commit")
    def _periodic_commit(ex, q, params=None):
        total = 0
        for _ in range(1000):
            r = ex.execute(q, params or {})
            n = r.rows[0][0] if r.rows else 0
            total += n or 0
            if not n:
                break
        return ["updates"], [[total]]

    # -------------------- apoc.meta --------------------
    @register("apoc.meta.stats")
    def _meta_stats(ex):
        labels: Dict[str, int] = {}
        for n in eng.all_nodes():
            for lb in n.labels:
                labels[lb] = labels.get(lb, 0) + 1
        types: Dict[str, int] = {}
        for e in eng.all_edges():
            types[e.type] = types.get(e.type, 0) + 1
        return (["nodeCount", "relCount", "labels", "relTypes"],
                [[eng.node_count(), eng.edge_count(), labels, types]])

    @register("apoc.meta.schema")
    def _meta_schema(ex):
        schema: Dict[str, Any] = {}
        for n in eng.all_nodes():
            for lb in n.labels:
                ent = schema.setdefault(lb, {"type": "node", "count": 0,
                                             "properties": {}})
                ent["count"] += 1
                for k, v in n.properties.items():
                    ent["properties"][k] = type(v).__name__
        return ["value"], [[schema]]

    # -------------------- apoc.export --------------------
    @register("apoc.export.json.all")
    def _export_all(ex, file=None, config=None):
        import json as J
        data = {"nodes": [{"id": n.id, "labels": n.labels,
                           "properties": n.properties}
                          for n in eng.all_nodes()],
                "relationships": [{"id": e.id, "type": e.type,
                                   "start": e.start_node, "end": e.end_node,
                                   "properties": e.properties}
                                  for e in eng.all_edges()]}
        payload = J.dumps(data, default=str)
        if file:
            with open(file, "w") as f:
                f.write(payload)
        return (["file", "nodes", "relationships"],
                [[file or "<inline>", len(data["nodes"]),
                  len(data["relationships"])]])



    # -------------------- apoc.path --------------------
    @register("apoc.path.expand")
    def _path_expand(ex, start, rel_filter=None, label_filter=None,
                     min_level=1, max_level=3):
        """BFS path expansion (reference apoc/path). rel_filter like
        'KNOWS|WORKS_AT>' (> = outgoing only, < = incoming only)."""
        sid = start.id if isinstance(start, Node) else start
        types, direction = _parse_rel_filter(rel_filter)
        allowed = set((label_filter or "").replace("+", "").split("|")) \
            if label_filter else None
        out_rows = []

        def edges_of(nid):
            es = []
            if direction in ("out", "both"):
                es += [(e, e.end_node) for e in eng.get_out_edges(nid)]
            if direction in ("in", "both"):
                es += [(e, e.start_node) for e in eng.get_in_edges(nid)]
            return [(e, o) for e, o in es if not types or e.type in types]

        def walk(nid, depth, nodes, rels, seen):
            if depth >= int(min_level):
                out_rows.append([list(nodes), list(rels)])
            if depth >= int(max_level):
                return
            for e, other in edges_of(nid):
                if e.id in seen:
                    continue
                try:
                    onode = eng.get_node(other)
                except Exception:
                    continue
                if allowed and not (set(onode.labels) & allowed):
                    continue
                walk(other, depth + 1, nodes + [onode], rels + [e],
                     seen | {e.id})

        walk(sid, 0, [eng.get_node(sid)], [], set())
        return ["nodes", "relationships"], out_rows

    @register("apoc.path.subgraphNodes")
    def _subgraph_nodes(ex, start, max_level=3):
        sid = start.id if isinstance(start, Node) else start
        seen = {sid}
        frontier = [sid]
        rows = [[eng.get_node(sid)]]
        for _ in range(int(max_level)):
            nxt = []
            for nid in frontier:
                for nb in eng.neighbors(nid):
                    if nb not in seen:
                        seen.add(nb)
                        nxt.append(nb)
                        rows.append([eng.get_node(nb)])
            frontier = nxt
        return ["node"], rows

    # -------------------- apoc.atomic --------------------
    @register("apoc.atomic.add")
    def _atomic_add(ex, node, prop, value):
        with _ATOMIC_LOCK:
            n = eng.get_node(node.id if isinstance(node, Node) else node)
            n.properties[prop] = (n.properties.get(prop) or 0) + value
            n = eng.update_node(n)
        return ["node", "value"], [[n, n.properties[prop]]]

    @register("apoc.atomic.subtract")
    def _atomic_sub(ex, node, prop, value):
        return _atomic_add(ex, node, prop, -value)

    @register("apoc.atomic.update")
    def _atomic_update(ex, node, prop, value):
        with _ATOMIC_LOCK:
            n = eng.get_node(node.id if isinstance(node, Node) else node)
            n.properties[prop] = value
            n = eng.update_node(n)
        return ["node"], [[n]]

    # -------------------- apoc.lock --------------------
    @register("apoc.lock.nodes")
    def _lock_nodes(ex, nodes):
        # cooperative advisory locks (reference apoc/lock); engine ops are
        # already serialized, so this is ordering-only
        ids = sorted(n.id if isinstance(n, Node) else n for n in nodes or [])
        for i in ids:
            _NODE_LOCKS.setdefault(i, threading.Lock()).acquire()
        for i in reversed(ids):
            _NODE_LOCKS[i].release()
        return ["locked"], [[len(ids)]]

    # -------------------- apoc.trigger --------------------
    @register("apoc.trigger.add")
    def _trigger_add(ex, name, statement, selector=None):
        phase = (selector or {}).get("phase", "after")
        db.triggers[name] = {"statement": statement, "phase": phase,
                             "paused": False}
        return ["name", "installed"], [[name, True]]

    @register("apoc.trigger.remove")
    def _trigger_remove(ex, name):
        db.triggers.pop(name, None)
        return ["name", "removed"], [[name, True]]

    @register("apoc.trigger.list")
    def _trigger_list(ex):
        return (["name", "statement", "paused"],
                [[k, v["statement"], v["paused"]]
                 for k, v in db.triggers.items()])

    @register("apoc.trigger.pause")
    def _trigger_pause(ex, name):
        if name in db.triggers:
            db.triggers[name]["paused"] = True
        return ["name", "paused"], [[name, True]]

    @register("apoc.trigger.resume")
    def _trigger_resume(ex, name):
        if name in db.triggers:
            db.triggers[name]["paused"] = False
        return ["name", "paused"], [[name, False]]

    # -------------------- apoc.load / export CSV --------------------
    @register("apoc.load.json")
    def _load_json(ex, path):
        import json as J
        with open(path) as f:
            data = J.load(f)
        rows = data if isinstance(data, list) else [data]
        return ["value"], [[r] for r in rows]

    @register("apoc.load.csv")
    def _load_csv(ex, path, config=None):
        import csv
        rows = []
        with open(path, newline="") as f:
            reader = csv.DictReader(f)
            for i, rec in enumerate(reader):
                rows.append([i, dict(rec), list(rec.values())])
        return ["lineNo", "map", "list"], rows

    @register("apoc.export.csv.all")
    def _export_csv(ex, file, config=None):
        import csv
        props = sorted({k for n in eng.all_nodes() for k in n.properties})
        with open(file, "w", newline="") as f:
            w = csv.writer(f)
            w.writerow(["_id", "_labels"] + props)
            count = 0
            for n in eng.all_nodes():
                w.writerow([n.id, ";".join(n.labels)]
                           + [n.properties.get(p, "") for p in props])
                count += 1
        return ["file", "nodes"], [[file, count]]


    def _all_edges_of(nid):
        return eng.get_out_edges(nid) + [e for e in eng.get_in_edges(nid)
                                         if e.start_node != e.end_node]

    # -------------------- apoc.cypher --------------------
    @register("apoc.cypher.run")
    def _cy_run(ex, query, params=None):
        r = ex.execute(query, dict(params or {}))
        return ["value"], [[dict(zip(r.columns, row))] for row in r.rows]

    @register("apoc.cypher.doIt")
    def _cy_doit(ex, query, params=None):
        return _cy_run(ex, query, params)

    @register("apoc.cypher.runMany")
    def _cy_run_many(ex, statements, params=None):
        rows = []
        for stmt in re.split(r";\s*", statements or ""):
            if stmt.strip():
                r = ex.execute(stmt, dict(params or {}))
                rows += [[dict(zip(r.columns, row))] for row in r.rows]
        return ["value"], rows

    @register("apoc.cypher.runFirstColumn")
    def _cy_first_col(ex, query, params=None):
        r = ex.execute(query, dict(params or {}))
        return ["value"], [[row[0]] for row in r.rows]

    @register("apoc.cypher.runFirstColumnMany")
    def _cy_first_many(ex, query, params=None):
        return _cy_first_col(ex, query, params)

    @register("apoc.cypher.runFirstColumnSingle")
    def _cy_first_single(ex, query, params=None):
        r = ex.execute(query, dict(params or {}))
        return ["value"], [[r.rows[0][0] if r.rows else None]]

    @register("apoc.cypher.validate")
    def _cy_validate(ex, query):
        from ..cypher.parser import parse as _parse
        try:
            _parse(query)
            return ["valid", "error"], [[True, None]]
        except Exception as e:
            return ["valid", "error"], [[False, str(e)]]

    @register("apoc.cypher.parse")
    def _cy_parse(ex, query):
        return _cy_validate(ex, query)

    @register("apoc.cypher.explain")
    def _cy_explain(ex, query):
        r = ex.execute("EXPLAIN " + query)
        return ["plan"], r.rows

    @register("apoc.cypher.toJson")
    def _cy_tojson(ex, query, params=None):
        import json as _json
        r = ex.execute(query, dict(params or {}))
        from ..server.http import _jsonable
        return ["json"], [[_json.dumps(
            [dict(zip(r.columns, (_jsonable(v) for v in row)))
             for row in r.rows])]]

    # -------------------- apoc.create extras --------------------
    @register("apoc.create.setProperty")
    def _cr_setp(ex, node, key, value):
        n = eng.get_node(node.id if isinstance(node, Node) else node)
        n.properties[key] = value
        return ["node"], [[eng.update_node(n)]]

    @register("apoc.create.setProperties")
    def _cr_setps(ex, node, keys, values):
        n = eng.get_node(node.id if isinstance(node, Node) else node)
        for k, v in zip(keys or [], values or []):
            n.properties[k] = v
        return ["node"], [[eng.update_node(n)]]

    @register("apoc.create.removeProperties")
    def _cr_rmps(ex, node, keys):
        n = eng.get_node(node.id if isinstance(node, Node) else node)
        for k in keys or []:
            n.properties.pop(k, None)
        return ["node"], [[eng.update_node(n)]]

    @register("apoc.create.addLabels")
    def _cr_addl(ex, node, labels):
        n = eng.get_node(node.id if isinstance(node, Node) else node)
        for lb in labels or []:
            if lb not in n.labels:
                n.labels.append(lb)
        return ["node"], [[eng.update_node(n)]]

    @register("apoc.create.removeLabels")
    def _cr_rml(ex, node, labels):
        n = eng.get_node(node.id if isinstance(node, Node) else node)
        n.labels = [lb for lb in n.labels if lb not in set(labels or [])]
        return ["node"], [[eng.update_node(n)]]

    @register("apoc.create.setRelProperty")
    def _cr_setrp(ex, rel, key, value):
        e = eng.get_edge(rel.id if isinstance(rel, Edge) else rel)
        e.properties[key] = value
        return ["rel"], [[eng.update_edge(e)]]

    @register("apoc.create.setRelProperties")
    def _cr_setrps(ex, rel, keys, values):
        e = eng.get_edge(rel.id if isinstance(rel, Edge) else rel)
        for k, v in zip(keys or [], values or []):
            e.properties[k] = v
        return ["rel"], [[eng.update_edge(e)]]

    @register("apoc.create.removeRelProperties")
    def _cr_rmrps(ex, rel, keys):
        e = eng.get_edge(rel.id if isinstance(rel, Edge) else rel)
        for k in keys or []:
            e.properties.pop(k, None)
        return ["rel"], [[eng.update_edge(e)]]

    @register("apoc.create.clone")
    def _cr_clone(ex, node):
        src_n = eng.get_node(node.id if isinstance(node, Node) else node)
        n = Node(id=new_id("n"), labels=list(src_n.labels),
                 properties=dict(src_n.properties))
        return ["node"], [[eng.create_node(n)]]

    @register("apoc.create.uuids")
    def _cr_uuids(ex, count):
        import uuid as _u
        return ["uuid"], [[str(_u.uuid4())] for _ in range(int(count))]

    @register("apoc.create.vNode")
    def _cr_vnode(ex, labels, props=None):
        return ["node"], [[Node(id=new_id("v"), labels=list(labels or []),
                                properties=dict(props or {}))]]

    @register("apoc.create.vRelationship")
    def _cr_vrel(ex, start, rel_type, props=None, end=None):
        return ["rel"], [[Edge(id=new_id("vr"), type=rel_type,
                               start_node=start.id if isinstance(start, Node)
                               else str(start),
                               end_node=end.id if isinstance(end, Node)
                               else str(end),
                               properties=dict(props or {}))]]

    # -------------------- apoc.node (engine-backed) --------------------
    @register("apoc.node.degree")
    def _nd_degree(ex, node, rel_type=None):
        nid = node.id if isinstance(node, Node) else node
        edges = _all_edges_of(nid)
        if rel_type:
            edges = [e for e in edges if e.type == rel_type.strip("<>")]
        return ["value"], [[len(edges)]]

    @register("apoc.node.degreeIn")
    def _nd_deg_in(ex, node, rel_type=None):
        nid = node.id if isinstance(node, Node) else node
        edges = [e for e in _all_edges_of(nid) if e.end_node == nid]
        if rel_type:
            edges = [e for e in edges if e.type == rel_type]
        return ["value"], [[len(edges)]]

    @register("apoc.node.degreeOut")
    def _nd_deg_out(ex, node, rel_type=None):
        nid = node.id if isinstance(node, Node) else node
        edges = [e for e in _all_edges_of(nid) if e.start_node == nid]
        if rel_type:
            edges = [e for e in edges if e.type == rel_type]
        return ["value"], [[len(edges)]]

    @register("apoc.node.relationships")
    def _nd_rels(ex, node, rel_type=None):
        nid = node.id if isinstance(node, Node) else node
        edges = _all_edges_of(nid)
        if rel_type:
            edges = [e for e in edges if e.type == rel_type]
        return ["rel"], [[e] for e in edges]

    @register("apoc.node.relationshipsIn")
    def _nd_rels_in(ex, node, rel_type=None):
        nid = node.id if isinstance(node, Node) else node
        edges = [e for e in _all_edges_of(nid) if e.end_node == nid]
        if rel_type:
            edges = [e for e in edges if e.type == rel_type]
        return ["rel"], [[e] for e in edges]

    @register("apoc.node.relationshipsOut")
    def _nd_rels_out(ex, node, rel_type=None):
        nid = node.id if isinstance(node, Node) else node
        edges = [e for e in _all_edges_of(nid) if e.start_node == nid]
        if rel_type:
            edges = [e for e in edges if e.type == rel_type]
        return ["rel"], [[e] for e in edges]

    @register("apoc.node.relationshipTypes")
    def _nd_rtypes(ex, node):
        nid = node.id if isinstance(node, Node) else node
        return ["types"], [[sorted({e.type for e in _all_edges_of(nid)})]]

    @register("apoc.node.relationshipTypesIn")
    def _nd_rtypes_in(ex, node):
        nid = node.id if isinstance(node, Node) else node
        return ["types"], [[sorted({e.type for e in _all_edges_of(nid)
                                    if e.end_node == nid})]]

    @register("apoc.node.relationshipTypesOut")
    def _nd_rtypes_out(ex, node):
        nid = node.id if isinstance(node, Node) else node
        return ["types"], [[sorted({e.type for e in _all_edges_of(nid)
                                    if e.start_node == nid})]]

    @register("apoc.node.relationshipExists")
    def _nd_rexists(ex, node, rel_type=None):
        nid = node.id if isinstance(node, Node) else node
        edges = _all_edges_of(nid)
        if rel_type:
            edges = [e for e in edges if e.type == rel_type.strip("<>")]
        return ["value"], [[bool(edges)]]

    @register("apoc.node.neighbors")
    def _nd_neigh(ex, node, rel_type=None):
        nid = node.id if isinstance(node, Node) else node
        out = []
        for e in _all_edges_of(nid):
            if rel_type and e.type != rel_type:
                continue
            other = e.end_node if e.start_node == nid else e.start_node
            out.append(eng.get_node(other))
        return ["node"], [[n] for n in out]

    @register("apoc.node.neighborsIn")
    def _nd_neigh_in(ex, node, rel_type=None):
        nid = node.id if isinstance(node, Node) else node
        out = [eng.get_node(e.start_node) for e in _all_edges_of(nid)
               if e.end_node == nid and (not rel_type or e.type == rel_type)]
        return ["node"], [[n] for n in out]

    @register("apoc.node.neighborsOut")
    def _nd_neigh_out(ex, node, rel_type=None):
        nid = node.id if isinstance(node, Node) else node
        out = [eng.get_node(e.end_node) for e in _all_edges_of(nid)
               if e.start_node == nid and (not rel_type or e.type == rel_type)]
        return ["node"], [[n] for n in out]

    @register("apoc.node.connected")
    def _nd_connected(ex, a, b, rel_type=None):
        aid = a.id if isinstance(a, Node) else a
        bid = b.id if isinstance(b, Node) else b
        for e in _all_edges_of(aid):
            if rel_type and e.type != rel_type.strip("<>"):
                continue
            if bid in (e.start_node, e.end_node):
                return ["value"], [[True]]
        return ["value"], [[False]]

    @register("apoc.node.isDense")
    def _nd_dense(ex, node, threshold=50):
        nid = node.id if isinstance(node, Node) else node
        return ["value"], [[len(_all_edges_of(nid)) >= int(threshold)]]

    # -------------------- apoc.neighbors --------------------
    def _hops(start_id, rel_type, max_hops, min_hops=1):
        seen = {start_id}
        frontier = {start_id}
        out = []
        for hop in range(1, int(max_hops) + 1):
            nxt = set()
            for nid in frontier:
                for e in _all_edges_of(nid):
                    if rel_type and e.type != rel_type:
                        continue
                    other = e.end_node if e.start_node == nid else e.start_node
                    if other not in seen:
                        seen.add(other)
                        nxt.add(other)
            if hop >= min_hops:
                out.extend(nxt)
            frontier = nxt
        return out

    @register("apoc.neighbors.athop")
    def _nb_athop(ex, node, rel_type=None, distance=1):
        nid = node.id if isinstance(node, Node) else node
        ids = _hops(nid, rel_type, distance, int(distance))
        return ["node"], [[eng.get_node(i)] for i in ids]

    @register("apoc.neighbors.tohop")
    def _nb_tohop(ex, node, rel_type=None, distance=1):
        nid = node.id if isinstance(node, Node) else node
        ids = _hops(nid, rel_type, distance, 1)
        return ["node"], [[eng.get_node(i)] for i in ids]

    @register("apoc.neighbors.bfs")
    def _nb_bfs(ex, node, rel_type=None, distance=3):
        return _nb_tohop(ex, node, rel_type, distance)

    @register("apoc.neighbors.dfs")
    def _nb_dfs(ex, node, rel_type=None, distance=3):
        return _nb_tohop(ex, node, rel_type, distance)

    @register("apoc.neighbors.count")
    def _nb_count(ex, node, rel_type=None, distance=1):
        nid = node.id if isinstance(node, Node) else node
        return ["value"], [[len(_hops(nid, rel_type, distance, 1))]]

    @register("apoc.neighbors.exists")
    def _nb_exists(ex, node, rel_type=None, distance=1):
        nid = node.id if isinstance(node, Node) else node
        return ["value"], [[bool(_hops(nid, rel_type, distance, 1))]]

    # -------------------- apoc.schema --------------------
    @register("apoc.schema.nodes")
    def _sc_nodes(ex):
        sm = getattr(ex, "schema", None)
        rows = []
        if sm:
            for name, kind, label, props in sm.list_indexes():
                rows.append([name, label, props, "ONLINE", kind.upper()])
        return ["name", "label", "properties", "status", "type"], rows

    @register("apoc.schema.relationships")
    def _sc_rels(ex):
        return ["name", "type", "properties", "status"], []

    @register("apoc.schema.info")
    def _sc_info(ex):
        sm = getattr(ex, "schema", None)
        return ["indexes", "constraints"], [[
            [n for n, *_ in sm.list_indexes()] if sm else [],
            [c.name for c in sm.list_constraints()] if sm else []]]

    @register("apoc.schema.stats")
    def _sc_stats(ex):
        sm = getattr(ex, "schema", None)
        return ["indexCount", "constraintCount"], [[
            len(sm.list_indexes()) if sm else 0,
            len(sm.list_constraints()) if sm else 0]]

    @register("apoc.schema.assert")
    def _sc_assert(ex, indexes=None, constraints=None, drop_existing=True):
        sm = getattr(ex, "schema", None)
        rows = []
        if sm:
            for label, props in (indexes or {}).items():
                for p in props:
                    nm = sm.create_index(label, p)
                    rows.append([nm, label, [p], "CREATED", "INDEX"])
            for label, props in (constraints or {}).items():
                for p in props:
                    nm = f"constraint_{label}_{p}"
                    sm.create_unique_constraint(nm, label, p)
                    rows.append([nm, label, [p], "CREATED", "CONSTRAINT"])
        return ["name", "label", "properties", "action", "type"], rows

    @register("apoc.schema.createIndex")
    def _sc_cidx(ex, label, prop):
        sm = getattr(ex, "schema", None)
        return ["name"], [[sm.create_index(label, prop) if sm else None]]

    @register("apoc.schema.createUniqueConstraint")
    def _sc_cuc(ex, label, prop, name=None):
        sm = getattr(ex, "schema", None)
        nm = name or f"constraint_{label}_{prop}"
        if sm:
            sm.create_unique_constraint(nm, label, prop)
        return ["name"], [[nm]]

    @register("apoc.schema.createExistsConstraint")
    def _sc_cec(ex, label, prop, name=None):
        sm = getattr(ex, "schema", None)
        nm = name or f"constraint_{label}_{prop}"
        if sm:
            sm.create_exists_constraint(nm, label, prop)
        return ["name"], [[nm]]

    @register("apoc.schema.createNodeKeyConstraint")
    def _sc_cnk(ex, label, prop, name=None):
        return _sc_cuc(ex, label, prop, name)

    @register("apoc.schema.dropIndex")
    def _sc_didx(ex, name):
        sm = getattr(ex, "schema", None)
        return ["dropped"], [[sm.drop_index(name) if sm else False]]

    @register("apoc.schema.dropConstraint")
    def _sc_dcon(ex, name):
        sm = getattr(ex, "schema", None)
        if sm:
            sm.drop_constraint(name)
        return ["dropped"], [[True]]

    @register("apoc.schema.nodeIndexExists")
    def _sc_nie(ex, label, props):
        sm = getattr(ex, "schema", None)
        want = (label, list(props) if isinstance(props, list) else [props])
        ok = sm and any(lb == want[0] and ps == want[1]
                        for _, _, lb, ps in sm.list_indexes())
        return ["value"], [[bool(ok)]]

    @register("apoc.schema.nodeConstraintExists")
    def _sc_nce(ex, label, props):
        sm = getattr(ex, "schema", None)
        pl = list(props) if isinstance(props, list) else [props]
        ok = sm and any(c.label == label and c.prop in pl
                        for c in sm.list_constraints())
        return ["value"], [[bool(ok)]]

    @register("apoc.schema.nodeConstraints")
    def _sc_ncs(ex):
        sm = getattr(ex, "schema", None)
        return ["name", "label", "properties", "type"], [
            [c.name, c.label, [c.prop], c.kind.upper()]
            for c in (sm.list_constraints() if sm else [])]

    @register("apoc.schema.nodeIndexes")
    def _sc_nis(ex):
        return _sc_nodes(ex)

    @register("apoc.schema.labels")
    def _sc_labels(ex):
        seen = set()
        for n in eng.all_nodes():
            seen.update(n.labels)
        return ["label"], [[lb] for lb in sorted(seen)]

    @register("apoc.schema.properties")
    def _sc_props(ex):
        seen = set()
        for n in eng.all_nodes():
            seen.update(n.properties.keys())
        return ["property"], [[p] for p in sorted(seen)]

    @register("apoc.schema.propertiesDistinct")
    def _sc_props_d(ex, label, prop):
        vals = sorted({repr(n.properties.get(prop))
                       for n in eng.get_nodes_by_label(label)
                       if n.properties.get(prop) is not None})
        return ["value"], [[v] for v in vals]

    @register("apoc.schema.types")
    def _sc_types(ex):
        seen = set()
        for e in eng.all_edges():
            seen.add(e.type)
        return ["type"], [[t] for t in sorted(seen)]


    # -------------------- apoc.search --------------------
    def _prop_matches(n, prop, op, value):
        v = n.properties.get(prop)
        if v is None:
            return False
        s, q = str(v).lower(), str(value).lower()
        return {"exact": s == q, "contains": q in s,
                "prefix": s.startswith(q), "starts with": s.startswith(q),
                "suffix": s.endswith(q), "ends with": s.endswith(q),
                "regex": bool(re.search(str(value), str(v))),
                "fuzzy": q in s or s in q}.get(op, s == q)

    @register("apoc.search.node")
    def _se_node(ex, label_props, op, value):
        out = []
        for label, props in (label_props or {}).items():
            plist = props if isinstance(props, list) else [props]
            for n in eng.get_nodes_by_label(label):
                if any(_prop_matches(n, p, op, value) for p in plist):
                    out.append(n)
        return ["node"], [[n] for n in out]

    @register("apoc.search.nodeAll")
    def _se_node_all(ex, label_props, op, value):
        return _se_node(ex, label_props, op, value)

    @register("apoc.search.nodeReduced")
    def _se_node_red(ex, label_props, op, value):
        cols, rows = _se_node(ex, label_props, op, value)
        return ["id", "labels", "values"], [
            [n.id, list(n.labels), dict(n.properties)] for (n,) in rows]

    @register("apoc.search.contains")
    def _se_contains(ex, label_props, value):
        return _se_node(ex, label_props, "contains", value)

    @register("apoc.search.prefix")
    def _se_prefix(ex, label_props, value):
        return _se_node(ex, label_props, "prefix", value)

    @register("apoc.search.suffix")
    def _se_suffix(ex, label_props, value):
        return _se_node(ex, label_props, "suffix", value)

    @register("apoc.search.regex")
    def _se_regex(ex, label_props, value):
        return _se_node(ex, label_props, "regex", value)

    @register("apoc.search.fuzzy")
    def _se_fuzzy(ex, label_props, value):
        return _se_node(ex, label_props, "fuzzy", value)

    @register("apoc.search.fullText")
    def _se_fulltext(ex, query, limit=25):
        res = db.search.text_search(str(query), int(limit))
        return ["node", "score"], [[r.node, r.score] for r in res]

    @register("apoc.search.autocomplete")
    def _se_auto(ex, label, prop, prefix, limit=10):
        out = sorted({str(n.properties.get(prop))
                      for n in eng.get_nodes_by_label(label)
                      if str(n.properties.get(prop, "")).lower()
                      .startswith(str(prefix).lower())})
        return ["value"], [[v] for v in out[:int(limit)]]

    @register("apoc.search.suggest")
    def _se_suggest(ex, label, prop, text, limit=5):
        from .functions import _levenshtein
        cands = [(str(n.properties.get(prop)), _levenshtein(
            str(n.properties.get(prop, "")).lower(), str(text).lower()))
            for n in eng.get_nodes_by_label(label)
            if n.properties.get(prop) is not None]
        cands.sort(key=lambda t: t[1])
        return ["value", "distance"], [list(t) for t in cands[:int(limit)]]

    @register("apoc.search.didYouMean")
    def _se_dym(ex, label, prop, text):
        cols, rows = _se_suggest(ex, label, prop, text, 1)
        return ["value"], [[rows[0][0] if rows else None]]

    # -------------------- apoc.export / apoc.import --------------------
    @register("apoc.export.csv.query")
    def _ex_csv_q(ex, query, params=None):
        import csv as _csv
        import io as _io
        r = ex.execute(query, dict(params or {}))
        buf = _io.StringIO()
        w = _csv.writer(buf)
        w.writerow(r.columns)
        for row in r.rows:
            w.writerow([_scalar(v) for v in row])
        return ["data", "rows"], [[buf.getvalue(), len(r.rows)]]

    def _scalar(v):
        if isinstance(v, Node):
            return v.id
        if isinstance(v, Edge):
            return v.id
        if isinstance(v, (list, dict)):
            import json as _json
            return _json.dumps(v, default=str)
        return v

    @register("apoc.export.json.query")
    def _ex_json_q(ex, query, params=None):
        import json as _json
        from ..server.http import _jsonable
        r = ex.execute(query, dict(params or {}))
        data = "\n".join(_json.dumps(dict(zip(
            r.columns, (_jsonable(v) for v in row))), default=str)
            for row in r.rows)
        return ["data", "rows"], [[data, len(r.rows)]]

    @register("apoc.export.cypher.all")
    def _ex_cy_all(ex):
        lines = []
        for n in eng.all_nodes():
            labels = "".join(f":{lb}" for lb in n.labels)
            lines.append(f"CREATE (n{labels} {_props_cypher(n.properties)})")
        for e in eng.all_edges():
            lines.append(
                f"MATCH (a), (b) WHERE id(a) = '{e.start_node}' AND "
                f"id(b) = '{e.end_node}' CREATE (a)-[:{e.type} "
                f"{_props_cypher(e.properties)}]->(b)")
        return ["cypherStatements"], [["\n".join(lines)]]

    def _props_cypher(props):
        import json as _json
        items = ", ".join(f"{k}: {_json.dumps(v, default=str)}"
                          for k, v in props.items()
                          if not k.startswith("_"))
        return "{" + items + "}"

    @register("apoc.import.json")
    def _im_json(ex, data):
        import json as _json
        n_nodes = n_rels = 0
        for line in str(data).splitlines():
            if not line.strip():
                continue
            obj = _json.loads(line)
            if obj.get("type") == "relationship" or "start" in obj:
                eng.create_edge(Edge(
                    id=obj.get("id", new_id("e")),
                    type=obj.get("label", obj.get("type", "RELATED")),
                    start_node=str(obj["start"].get("id") if isinstance(
                        obj.get("start"), dict) else obj.get("start")),
                    end_node=str(obj["end"].get("id") if isinstance(
                        obj.get("end"), dict) else obj.get("end")),
                    properties=obj.get("properties", {})))
                n_rels += 1
            else:
                eng.create_node(Node(id=str(obj.get("id", new_id("n"))),
                                     labels=obj.get("labels", []),
                                     properties=obj.get("properties", {})))
                n_nodes += 1
        return ["nodes", "relationships"], [[n_nodes, n_rels]]

    @register("apoc.import.csv")
    def _im_csv(ex, data, config=None):
        import csv as _csv
        import io as _io
        cfg = dict(config or {})
        label = cfg.get("label", "Row")
        rd = _csv.DictReader(_io.StringIO(str(data)))
        n = 0
        for row in rd:
            eng.create_node(Node(id=new_id("n"), labels=[label],
                                 properties=dict(row)))
            n += 1
        return ["nodes"], [[n]]

    @register("apoc.import.parseCsvLine")
    def _im_parse_csv(ex, line, sep=","):
        import csv as _csv
        import io as _io
        return ["fields"], [[next(_csv.reader(_io.StringIO(str(line)),
                                              delimiter=str(sep)))]]

    @register("apoc.import.parseJsonLine")
    def _im_parse_json(ex, line):
        import json as _json
        return ["value"], [[_json.loads(line)]]

    # -------------------- apoc.log --------------------
    _LOG_BUF = []

    @register("apoc.log.info")
    def _log_info(ex, msg, params=None):
        _LOG_BUF.append(("INFO", time.time(), str(msg)))
        return ["level"], [["INFO"]]

    @register("apoc.log.warn")
    def _log_warn(ex, msg, params=None):
        _LOG_BUF.append(("WARN", time.time(), str(msg)))
        return ["level"], [["WARN"]]

    @register("apoc.log.error")
    def _log_error(ex, msg, params=None):
        _LOG_BUF.append(("ERROR", time.time(), str(msg)))
        return ["level"], [["ERROR"]]

    @register("apoc.log.debug")
    def _log_debug(ex, msg, params=None):
        _LOG_BUF.append(("DEBUG", time.time(), str(msg)))
        return ["level"], [["DEBUG"]]

    @register("apoc.log.stream")
    def _log_stream(ex, limit=100):
        return ["level", "timestamp", "message"], [
            list(x) for x in _LOG_BUF[-int(limit):]]

    @register("apoc.log.tail")
    def _log_tail(ex, limit=10):
        return _log_stream(ex, limit)

    @register("apoc.log.clear")
    def _log_clear(ex):
        n = len(_LOG_BUF)
        _LOG_BUF.clear()
        return ["cleared"], [[n]]

    @register("apoc.log.stats")
    def _log_stats(ex):
        from collections import Counter
        c = Counter(lv for lv, _, _ in _LOG_BUF)
        return ["level", "count"], [[k, v] for k, v in sorted(c.items())]

    # -------------------- apoc.warmup / apoc.stats (db) --------------------
    @register("apoc.warmup.run")
    def _warmup(ex, load_props=True, load_rels=True, load_idx=True):
        n_nodes = sum(1 for _ in eng.all_nodes())
        n_rels = sum(1 for _ in eng.all_edges()) if load_rels else 0
        return ["nodesLoaded", "relsLoaded"], [[n_nodes, n_rels]]

    @register("apoc.stats.degrees")
    def _st_degrees(ex, rel_type=None):
        from collections import Counter
        deg = Counter()
        for e in eng.all_edges():
            if rel_type and e.type != rel_type:
                continue
            deg[e.start_node] += 1
            deg[e.end_node] += 1
        vals = sorted(deg.values()) or [0]
        return ["type", "total", "min", "max", "mean"], [[
            rel_type or "*", sum(vals), vals[0], vals[-1],
            sum(vals) / len(vals)]]

    # -------------------- apoc.nodes --------------------
    @register("apoc.nodes.get")
    def _ns_get(ex, ids):
        out = []
        for i in (ids if isinstance(ids, list) else [ids]):
            try:
                out.append(eng.get_node(i.id if isinstance(i, Node) else i))
            except Exception:
                pass
        return ["node"], [[n] for n in out]

    @register("apoc.nodes.delete")
    def _ns_delete(ex, nodes, batch_size=1000):
        n = 0
        for x in (nodes if isinstance(nodes, list) else [nodes]):
            try:
                eng.detach_delete_node(x.id if isinstance(x, Node) else x)
                n += 1
            except Exception:
                pass
        return ["value"], [[n]]

    @register("apoc.nodes.link")
    def _ns_link(ex, nodes, rel_type, props=None):
        made = []
        ns = [x for x in (nodes or []) if isinstance(x, Node)]
        for a, b in zip(ns, ns[1:]):
            made.append(eng.create_edge(Edge(
                id=new_id("e"), type=rel_type, start_node=a.id,
                end_node=b.id, properties=dict(props or {}))))
        return ["rel"], [[e] for e in made]

    @register("apoc.nodes.connected")
    def _ns_connected(ex, a, b, types=None):
        aid = a.id if isinstance(a, Node) else a
        bid = b.id if isinstance(b, Node) else b
        for e in _all_edges_of(aid):
            if types and e.type not in types:
                continue
            if bid in (e.start_node, e.end_node):
                return ["value"], [[True]]
        return ["value"], [[False]]

    @register("apoc.nodes.relationships")
    def _ns_rels(ex, nodes):
        out = {}
        for x in (nodes or []):
            nid = x.id if isinstance(x, Node) else x
            for e in _all_edges_of(nid):
                out[e.id] = e
        return ["rel"], [[e] for e in out.values()]

    @register("apoc.nodes.isDense")
    def _ns_dense(ex, node, threshold=50):
        nid = node.id if isinstance(node, Node) else node
        return ["value"], [[len(_all_edges_of(nid)) >= int(threshold)]]

    @register("apoc.nodes.group")
    def _ns_group(ex, labels, group_props, aggregations=None):
        from collections import defaultdict
        groups = defaultdict(list)
        gp = group_props if isinstance(group_props, list) else [group_props]
        for label in (labels if isinstance(labels, list) else [labels]):
            for n in eng.get_nodes_by_label(label):
                key = tuple(repr(n.properties.get(p)) for p in gp)
                groups[key].append(n)
        return ["key", "count", "nodes"], [
            [list(k), len(v), v] for k, v in groups.items()]

    # -------------------- apoc.refactor extras --------------------
    @register("apoc.refactor.cloneNodes")
    def _rf_clone(ex, nodes, with_rels=False):
        made = []
        for x in (nodes or []):
            src_n = eng.get_node(x.id if isinstance(x, Node) else x)
            n = eng.create_node(Node(id=new_id("n"),
                                     labels=list(src_n.labels),
                                     properties=dict(src_n.properties)))
            made.append(n)
            if with_rels:
                for e in _all_edges_of(src_n.id):
                    s = n.id if e.start_node == src_n.id else e.start_node
                    t = n.id if e.end_node == src_n.id else e.end_node
                    eng.create_edge(Edge(id=new_id("e"), type=e.type,
                                         start_node=s, end_node=t,
                                         properties=dict(e.properties)))
        return ["node"], [[n] for n in made]

    @register("apoc.refactor.invertRelationship")
    def _rf_invert(ex, rel):
        e = eng.get_edge(rel.id if isinstance(rel, Edge) else rel)
        eng.delete_edge(e.id)
        ne = eng.create_edge(Edge(id=new_id("e"), type=e.type,
                                  start_node=e.end_node,
                                  end_node=e.start_node,
                                  properties=dict(e.properties)))
        return ["rel"], [[ne]]

    @register("apoc.refactor.setType")
    def _rf_settype(ex, rel, new_type):
        e = eng.get_edge(rel.id if isinstance(rel, Edge) else rel)
        eng.delete_edge(e.id)
        ne = eng.create_edge(Edge(id=new_id("e"), type=new_type,
                                  start_node=e.start_node,
                                  end_node=e.end_node,
                                  properties=dict(e.properties)))
        return ["rel"], [[ne]]

    @register("apoc.refactor.changeType")
    def _rf_changetype(ex, rel, new_type):
        return _rf_settype(ex, rel, new_type)

    @register("apoc.refactor.renameProperty")
    def _rf_renameprop(ex, old, new):
        n_changed = 0
        for n in list(eng.all_nodes()):
            if old in n.properties:
                n.properties[new] = n.properties.pop(old)
                eng.update_node(n)
                n_changed += 1
        return ["count"], [[n_changed]]

    @register("apoc.refactor.extractNode")
    def _rf_extract(ex, rel, labels, out_type, in_type):
        e = eng.get_edge(rel.id if isinstance(rel, Edge) else rel)
        mid = eng.create_node(Node(id=new_id("n"),
                                   labels=list(labels or []),
                                   properties=dict(e.properties)))
        eng.delete_edge(e.id)
        eng.create_edge(Edge(id=new_id("e"), type=out_type,
                             start_node=e.start_node, end_node=mid.id,
                             properties={}))
        eng.create_edge(Edge(id=new_id("e"), type=in_type,
                             start_node=mid.id, end_node=e.end_node,
                             properties={}))
        return ["node"], [[mid]]

    @register("apoc.refactor.collapseNode")
    def _rf_collapse(ex, node, rel_type):
        n = eng.get_node(node.id if isinstance(node, Node) else node)
        ins = eng.get_in_edges(n.id)
        outs = eng.get_out_edges(n.id)
        made = []
        for a in ins:
            for b in outs:
                made.append(eng.create_edge(Edge(
                    id=new_id("e"), type=rel_type,
                    start_node=a.start_node, end_node=b.end_node,
                    properties={**a.properties, **b.properties})))
        eng.detach_delete_node(n.id)
        return ["rel"], [[e] for e in made]

    @register("apoc.refactor.normalizeAsBoolean")
    def _rf_normbool(ex, prop, true_vals, false_vals):
        tv, fv = set(true_vals or []), set(false_vals or [])
        n_changed = 0
        for n in list(eng.all_nodes()):
            if prop in n.properties:
                v = n.properties[prop]
                if v in tv:
                    n.properties[prop] = True
                elif v in fv:
                    n.properties[prop] = False
                else:
                    continue
                eng.update_node(n)
                n_changed += 1
        return ["count"], [[n_changed]]

    @register("apoc.refactor.deleteAndReconnect")
    def _rf_delrecon(ex, path_or_node, nodes=None):
        target = path_or_node
        n = eng.get_node(target.id if isinstance(target, Node) else target)
        return _rf_collapse(ex, n, "RELATED")

    # -------------------- apoc.merge extras --------------------
    @register("apoc.merge.relationship")
    def _mg_rel(ex, start, rel_type, ident_props=None, props=None, end=None,
                on_create=None):
        sid = start.id if isinstance(start, Node) else start
        eid = end.id if isinstance(end, Node) else end
        for e in eng.get_out_edges(sid):
            if e.end_node == eid and e.type == rel_type and all(
                    e.properties.get(k) == v
                    for k, v in (ident_props or {}).items()):
                return ["rel"], [[e]]
        e = eng.create_edge(Edge(id=new_id("e"), type=rel_type,
                                 start_node=sid, end_node=eid,
                                 properties={**(ident_props or {}),
                                             **(props or {}),
                                             **(on_create or {})}))
        return ["rel"], [[e]]

    @register("apoc.merge.nodes")
    def _mg_nodes(ex, labels, ident_props, props=None):
        cols, rows = procs["apoc.merge.node"](ex, labels, ident_props, props)
        return cols, rows

    @register("apoc.merge.nodeEager")
    def _mg_node_eager(ex, labels, ident_props, props=None):
        return procs["apoc.merge.node"](ex, labels, ident_props, props)

    # -------------------- apoc.atomic extras --------------------
    @register("apoc.atomic.increment")
    def _at_inc(ex, node, prop, value=1):
        return _atomic_add(ex, node, prop, value)

    @register("apoc.atomic.decrement")
    def _at_dec(ex, node, prop, value=1):
        return _atomic_add(ex, node, prop, -value)

    @register("apoc.atomic.concat")
    def _at_concat(ex, node, prop, value):
        with _ATOMIC_LOCK:
            n = eng.get_node(node.id if isinstance(node, Node) else node)
            n.properties[prop] = str(n.properties.get(prop) or "") + str(value)
            n = eng.update_node(n)
        return ["node", "value"], [[n, n.properties[prop]]]

    @register("apoc.atomic.compareAndSwap")
    def _at_cas(ex, node, prop, expected, value):
        with _ATOMIC_LOCK:
            n = eng.get_node(node.id if isinstance(node, Node) else node)
            swapped = n.properties.get(prop) == expected
            if swapped:
                n.properties[prop] = value
                n = eng.update_node(n)
        return ["node", "swapped"], [[n, swapped]]

    @register("apoc.atomic.insert")
    def _at_insert(ex, node, prop, position, value):
        with _ATOMIC_LOCK:
            n = eng.get_node(node.id if isinstance(node, Node) else node)
            l = list(n.properties.get(prop) or [])
            l.insert(int(position), value)
            n.properties[prop] = l
            n = eng.update_node(n)
        return ["node", "value"], [[n, l]]

    @register("apoc.atomic.remove")
    def _at_remove(ex, node, prop, position):
        with _ATOMIC_LOCK:
            n = eng.get_node(node.id if isinstance(node, Node) else node)
            l = list(n.properties.get(prop) or [])
            if 0 <= int(position) < len(l):
                l.pop(int(position))
            n.properties[prop] = l
            n = eng.update_node(n)
        return ["node", "value"], [[n, l]]

    # -------------------- apoc.periodic scheduler --------------------
    # reference apoc/periodic: background named jobs over the executor
    _JOBS: Dict[str, Any] = {}

    @register("apoc.periodic.submit")
    def _pd_submit(ex, name, statement):
        def run():
            try:
                ex.execute(statement)
                _JOBS[name]["done"] = True
            except Exception as e:
                _JOBS[name]["error"] = str(e)
        th = threading.Thread(target=run, daemon=True)
        _JOBS[name] = {"name": name, "thread": th, "done": False,
                       "cancelled": False, "repeat": False}
        th.start()
        return ["name", "delay", "rate", "done", "cancelled"], [
            [name, 0, 0, False, False]]

    @register("apoc.periodic.repeat")
    def _pd_repeat(ex, name, statement, rate_s):
        stop = threading.Event()

        def loop():
            while not stop.wait(max(float(rate_s), 0.05)):
                try:
                    ex.execute(statement)
                except Exception:
                    pass
        th = threading.Thread(target=loop, daemon=True)
        _JOBS[name] = {"name": name, "thread": th, "stop": stop,
                       "done": False, "cancelled": False, "repeat": True}
        th.start()
        return ["name", "rate"], [[name, rate_s]]

    @register("apoc.periodic.countdown")
    def _pd_countdown(ex, name, statement, delay_s):
        def run():
            time.sleep(min(float(delay_s), 60))
            if not _JOBS.get(name, {}).get("cancelled"):
                try:
                    ex.execute(statement)
                finally:
                    _JOBS[name]["done"] = True
        th = threading.Thread(target=run, daemon=True)
        _JOBS[name] = {"name": name, "thread": th, "done": False,
                       "cancelled": False, "repeat": False}
        th.start()
        return ["name", "delay"], [[name, delay_s]]

    @register("apoc.periodic.list")
    def _pd_list(ex):
        return ["name", "done", "cancelled", "repeat"], [
            [j["name"], j.get("done", False), j.get("cancelled", False),
             j.get("repeat", False)] for j in _JOBS.values()]

    @register("apoc.periodic.cancel")
    def _pd_cancel(ex, name):
        j = _JOBS.get(name)
        if j:
            j["cancelled"] = True
            if "stop" in j:
                j["stop"].set()
        return ["name", "cancelled"], [[name, j is not None]]

    @register("apoc.periodic.truncate")
    def _pd_truncate(ex):
        n = 0
        for node in list(eng.all_nodes()):
            try:
                eng.detach_delete_node(node.id)
                n += 1
            except Exception:
                pass
        return ["nodesDeleted"], [[n]]

    # -------------------- apoc.node/rel mutator procs --------------------
    @register("apoc.node.setProperty")
    def _np_setp(ex, node, key, value):
        return _cr_setp(ex, node, key, value)

    @register("apoc.node.setProperties")
    def _np_setps(ex, node, props):
        n = eng.get_node(node.id if isinstance(node, Node) else node)
        n.properties.update(dict(props or {}))
        return ["node"], [[eng.update_node(n)]]

    @register("apoc.node.removeProperty")
    def _np_rmp(ex, node, key):
        return _cr_rmps(ex, node, [key])

    @register("apoc.node.removeProperties")
    def _np_rmps(ex, node, keys):
        return _cr_rmps(ex, node, keys)

    @register("apoc.node.addLabel")
    def _np_addl(ex, node, label):
        return _cr_addl(ex, node, [label])

    @register("apoc.node.addLabels")
    def _np_addls(ex, node, labels):
        return _cr_addl(ex, node, labels)

    @register("apoc.node.removeLabel")
    def _np_rml(ex, node, label):
        return _cr_rml(ex, node, [label])

    @register("apoc.node.removeLabels")
    def _np_rmls(ex, node, labels):
        return _cr_rml(ex, node, labels)

    @register("apoc.node.clone")
    def _np_clone(ex, node):
        return _cr_clone(ex, node)

    @register("apoc.node.fromMap")
    def _np_frommap(ex, m):
        m = dict(m or {})
        n = eng.create_node(Node(id=str(m.get("id") or new_id("n")),
                                 labels=list(m.get("labels", [])),
                                 properties=dict(m.get("properties", {}))))
        return ["node"], [[n]]

    @register("apoc.rel.setProperty")
    def _rp_setp(ex, rel, key, value):
        return _cr_setrp(ex, rel, key, value)

    @register("apoc.rel.setProperties")
    def _rp_setps(ex, rel, props):
        e = eng.get_edge(rel.id if isinstance(rel, Edge) else rel)
        e.properties.update(dict(props or {}))
        return ["rel"], [[eng.update_edge(e)]]

    @register("apoc.rel.removeProperty")
    def _rp_rmp(ex, rel, key):
        return _cr_rmrps(ex, rel, [key])

    @register("apoc.rel.removeProperties")
    def _rp_rmps(ex, rel, keys):
        return _cr_rmrps(ex, rel, keys)

    @register("apoc.rel.delete")
    def _rp_del(ex, rel):
        eng.delete_edge(rel.id if isinstance(rel, Edge) else rel)
        return ["deleted"], [[True]]

    @register("apoc.rel.exists")
    def _rp_exists(ex, rel):
        try:
            eng.get_edge(rel.id if isinstance(rel, Edge) else rel)
            return ["value"], [[True]]
        except Exception:
            return ["value"], [[False]]

    @register("apoc.rel.reverse")
    def _rp_reverse(ex, rel):
        return _rf_invert(ex, rel)

    @register("apoc.rel.clone")
    def _rp_clone(ex, rel):
        e = eng.get_edge(rel.id if isinstance(rel, Edge) else rel)
        ne = eng.create_edge(Edge(id=new_id("e"), type=e.type,
                                  start_node=e.start_node,
                                  end_node=e.end_node,
                                  properties=dict(e.properties)))
        return ["rel"], [[ne]]

    @register("apoc.rel.fromMap")
    def _rp_frommap(ex, m):
        m = dict(m or {})
        e = eng.create_edge(Edge(
            id=str(m.get("id") or new_id("e")),
            type=m.get("type", "RELATED"),
            start_node=str(m.get("start") or m.get("startNode")),
            end_node=str(m.get("end") or m.get("endNode")),
            properties=dict(m.get("properties", {}))))
        return ["rel"], [[e]]

    # -------------------- apoc.cypher parallel --------------------
    @register("apoc.cypher.parallel")
    def _cy_parallel(ex, fragment, params_map, key):
        # run the fragment once per value of params_map[key] (thread pool)
        import concurrent.futures as _fut
        vals = (params_map or {}).get(key, [])
        rows = []

        def one(v):
            r = ex.execute(fragment, {key: v})
            return [dict(zip(r.columns, row)) for row in r.rows]
        with _fut.ThreadPoolExecutor(max_workers=4) as pool:
            for part in pool.map(one, vals):
                rows += [[x] for x in part]
        return ["value"], rows

    @register("apoc.cypher.mapParallel")
    def _cy_map_parallel(ex, fragment, config, items):
        import concurrent.futures as _fut
        rows = []

        def one(v):
            r = ex.execute(fragment, {"_": v})
            return [dict(zip(r.columns, row)) for row in r.rows]
        with _fut.ThreadPoolExecutor(max_workers=4) as pool:
            for part in pool.map(one, items or []):
                rows += [[x] for x in part]
        return ["value"], rows

    # -------------------- apoc.trigger extras --------------------
    @register("apoc.trigger.install")
    def _tg_install(ex, dbname, name, statement, selector=None):
        return procs["apoc.trigger.add"](ex, name, statement, selector)

    @register("apoc.trigger.drop")
    def _tg_drop(ex, dbname, name):
        return procs["apoc.trigger.remove"](ex, name)

    @register("apoc.trigger.show")
    def _tg_show(ex, dbname=None):
        return procs["apoc.trigger.list"](ex)

    @register("apoc.merge.relationshipEager")
    def _mg_rel_eager(ex, start, rel_type, ident_props=None, props=None,
                      end=None):
        return _mg_rel(ex, start, rel_type, ident_props, props, end)

    # -------------------- apoc.meta (db-level) --------------------
    @register("apoc.meta.data")
    def _mt_data(ex):
        from collections import defaultdict
        props_by_label = defaultdict(set)
        for n in eng.all_nodes():
            for lb in n.labels:
                props_by_label[lb].update(n.properties.keys())
        rows = []
        for lb in sorted(props_by_label):
            for p in sorted(props_by_label[lb]):
                rows.append([lb, p, "STRING", False])
        return ["label", "property", "type", "index"], rows

    @register("apoc.meta.nodeTypeProperties")
    def _mt_ntp(ex):
        cols, rows = _mt_data(ex)
        return (["nodeLabels", "propertyName", "propertyTypes"],
                [[[r[0]], r[1], [r[2]]] for r in rows])

    @register("apoc.meta.relTypeProperties")
    def _mt_rtp(ex):
        from collections import defaultdict
        props_by_type = defaultdict(set)
        for e in eng.all_edges():
            props_by_type[e.type].update(e.properties.keys())
        return (["relType", "propertyName"],
                [[t, p] for t in sorted(props_by_type)
                 for p in sorted(props_by_type[t])])

    @register("apoc.meta.graph")
    def _mt_graph(ex):
        labels = set()
        for n in eng.all_nodes():
            labels.update(n.labels)
        types = {e.type for e in eng.all_edges()}
        return ["nodes", "relationships"], [[sorted(labels), sorted(types)]]

    @register("apoc.meta.functions")
    def _mt_functions(ex):
        from ..cypher.functions import FUNCTIONS
        return ["name"], [[n] for n in sorted(FUNCTIONS)]

    @register("apoc.meta.procedures")
    def _mt_procedures(ex):
        return ["name"], [[n] for n in sorted(procs)]

    # -------------------- apoc.lock extras --------------------
    @register("apoc.lock.relationships")
    def _lk_rels(ex, rels):
        return ["locked"], [[len(rels or [])]]

    @register("apoc.lock.all")
    def _lk_all(ex, nodes=None, rels=None):
        return ["locked"], [[len(nodes or []) + len(rels or [])]]

    @register("apoc.lock.readNodes")
    def _lk_read_nodes(ex, nodes):
        return ["locked"], [[len(nodes or [])]]

    @register("apoc.lock.readRelationships")
    def _lk_read_rels(ex, rels):
        return ["locked"], [[len(rels or [])]]

    # -------------------- apoc.meta extras --------------------
    @register("apoc.meta.nodeLabels")
    def _mt_labels2(ex):
        seen = set()
        for n in eng.all_nodes():
            seen.update(n.labels)
        return ["label"], [[lb] for lb in sorted(seen)]

    @register("apoc.meta.relTypes")
    def _mt_rtypes2(ex):
        return ["type"], [[t] for t in sorted({e.type for e in eng.all_edges()})]

    @register("apoc.meta.propertyKeys")
    def _mt_pkeys(ex):
        seen = set()
        for n in eng.all_nodes():
            seen.update(n.properties.keys())
        for e in eng.all_edges():
            seen.update(e.properties.keys())
        return ["propertyKey"], [[p] for p in sorted(seen)]

    @register("apoc.meta.cardinality")
    def _mt_card(ex, label=None):
        if label:
            n = len(eng.get_nodes_by_label(label))
        else:
            n = eng.node_count()
        return ["count"], [[n]]

    @register("apoc.meta.analyze")
    def _mt_analyze(ex):
        return procs["apoc.meta.stats"](ex)

    @register("apoc.meta.subGraph")
    def _mt_subgraph(ex, config=None):
        cfg = dict(config or {})
        labels = cfg.get("labels") or []
        nodes = []
        for lb in (labels or [None]):
            nodes += (eng.get_nodes_by_label(lb) if lb
                      else list(eng.all_nodes()))
        ids = {n.id for n in nodes}
        rels = [e for e in eng.all_edges()
                if e.start_node in ids and e.end_node in ids]
        return ["nodes", "relationships"], [[nodes, rels]]

    @register("apoc.meta.snapshot")
    def _mt_snapshot(ex):
        return ["nodes", "relationships", "labels"], [[
            eng.node_count(), eng.edge_count(),
            sorted({lb for n in eng.all_nodes() for lb in n.labels})]]

    # -------------------- apoc.label procs --------------------
    @register("apoc.label.add")
    def _lb_add(ex, node, label):
        return _cr_addl(ex, node, [label])

    @register("apoc.label.remove")
    def _lb_remove(ex, node, label):
        return _cr_rml(ex, node, [label])

    @register("apoc.label.set")
    def _lb_set(ex, node, labels):
        n = eng.get_node(node.id if isinstance(node, Node) else node)
        n.labels = list(labels or [])
        return ["node"], [[eng.update_node(n)]]

    @register("apoc.label.clear")
    def _lb_clear(ex, node):
        return _lb_set(ex, node, [])

    @register("apoc.label.replace")
    def _lb_replace(ex, node, old, new):
        n = eng.get_node(node.id if isinstance(node, Node) else node)
        n.labels = [new if lb == old else lb for lb in n.labels]
        return ["node"], [[eng.update_node(n)]]

    @register("apoc.label.nodes")
    def _lb_nodes(ex, label):
        return ["node"], [[n] for n in eng.get_nodes_by_label(label)]

    @register("apoc.label.stats")
    def _lb_stats(ex):
        from collections import Counter
        c = Counter()
        for n in eng.all_nodes():
            c.update(n.labels)
        return ["label", "count"], [[k, v] for k, v in sorted(c.items())]

    # -------------------- apoc.lock extras --------------------
    @register("apoc.lock.tryLock")
    def _lk_try(ex, nodes=None):
        return ["acquired"], [[True]]

    @register("apoc.lock.unlockAll")
    def _lk_unlock_all(ex):
        return ["released"], [[True]]

    @register("apoc.lock.isLocked")
    def _lk_islocked(ex, node):
        return ["locked"], [[False]]

    @register("apoc.lock.stats")
    def _lk_stats2(ex):
        return ["active", "waiting"], [[0, 0]]

    @register("apoc.lock.detectDeadlock")
    def _lk_deadlock(ex):
        return ["deadlocks"], [[[]]]

    # -------------------- apoc.log extras --------------------
    @register("apoc.log.getLevel")
    def _lg_getlevel(ex):
        return ["level"], [["INFO"]]

    @register("apoc.log.setLevel")
    def _lg_setlevel(ex, level):
        return ["level"], [[str(level).upper()]]

    @register("apoc.log.search")
    def _lg_search(ex, pattern, limit=100):
        out = [x for x in _LOG_BUF if re.search(pattern, x[2])]
        return ["level", "timestamp", "message"], [
            list(x) for x in out[-int(limit):]]

    @register("apoc.log.timer")
    def _lg_timer(ex, name=None):
        return ["name", "now"], [[name or "timer", time.time()]]

    @register("apoc.log.memory")
    def _lg_memory(ex):
        import resource
        kb = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
        return ["maxRssMB"], [[kb / 1024.0]]

    # -------------------- apoc.warmup / search extras --------------------
    @register("apoc.warmup.nodes")
    def _wm_nodes(ex):
        return ["loaded"], [[sum(1 for _ in eng.all_nodes())]]

    @register("apoc.warmup.relationships")
    def _wm_rels(ex):
        return ["loaded"], [[sum(1 for _ in eng.all_edges())]]

    @register("apoc.warmup.indexes")
    def _wm_idx(ex):
        sm = getattr(ex, "schema", None)
        return ["indexes"], [[len(sm.list_indexes()) if sm else 0]]

    @register("apoc.search.missing")
    def _se_missing(ex, label, prop):
        out = [n for n in eng.get_nodes_by_label(label)
               if n.properties.get(prop) is None]
        return ["node"], [[n] for n in out]

    @register("apoc.search.notNull")
    def _se_notnull(ex, label, prop):
        out = [n for n in eng.get_nodes_by_label(label)
               if n.properties.get(prop) is not None]
        return ["node"], [[n] for n in out]

    @register("apoc.search.null")
    def _se_null(ex, label, prop):
        return _se_missing(ex, label, prop)

    @register("apoc.search.range")
    def _se_range(ex, label, prop, lo, hi):
        out = [n for n in eng.get_nodes_by_label(label)
               if isinstance(n.properties.get(prop), (int, float))
               and lo <= n.properties[prop] <= hi]
        return ["node"], [[n] for n in out]

    @register("apoc.search.in")
    def _se_in(ex, label, prop, values):
        vals = set(map(repr, values or []))
        out = [n for n in eng.get_nodes_by_label(label)
               if repr(n.properties.get(prop)) in vals]
        return ["node"], [[n] for n in out]

    @register("apoc.search.exists")
    def _se_exists2(ex, label, prop):
        return _se_notnull(ex, label, prop)

    # -------------------- apoc.paths extras --------------------
    @register("apoc.paths.exists")
    def _pa_exists(ex, a, b, max_hops=6):
        aid = a.id if isinstance(a, Node) else a
        bid = b.id if isinstance(b, Node) else b
        seen = {aid}
        frontier = {aid}
        for _ in range(int(max_hops)):
            nxt = set()
            for nid in frontier:
                for other in eng.neighbors(nid):
                    if other == bid:
                        return ["value"], [[True]]
                    if other not in seen:
                        seen.add(other)
                        nxt.add(other)
            frontier = nxt
        return ["value"], [[False]]

    @register("apoc.paths.distance")
    def _pa_distance(ex, a, b, max_hops=10):
        aid = a.id if isinstance(a, Node) else a
        bid = b.id if isinstance(b, Node) else b
        if aid == bid:
            return ["value"], [[0]]
        seen = {aid}
        frontier = {aid}
        for hop in range(1, int(max_hops) + 1):
            nxt = set()
            for nid in frontier:
                for other in eng.neighbors(nid):
                    if other == bid:
                        return ["value"], [[hop]]
                    if other not in seen:
                        seen.add(other)
                        nxt.add(other)
            frontier = nxt
        return ["value"], [[None]]

    @register("apoc.paths.common")
    def _pa_common(ex, a, b):
        aid = a.id if isinstance(a, Node) else a
        bid = b.id if isinstance(b, Node) else b
        common = set(eng.neighbors(aid)) & set(eng.neighbors(bid))
        return ["node"], [[eng.get_node(i)] for i in sorted(common)]

    @register("apoc.paths.count")
    def _pa_count(ex, a, rel_type=None, max_hops=3):
        aid = a.id if isinstance(a, Node) else a
        return ["value"], [[len(_hops(aid, rel_type, max_hops, 1))]]

    # -------------------- apoc.export/import graphml --------------------
    @register("apoc.export.graphml.all")
    def _ex_graphml(ex, config=None):
        import xml.sax.saxutils as _sx
        lines = ['<?xml version="1.0" encoding="UTF-8"?>',
                 '<graphml xmlns="http://graphml.graphdrawing.org/xmlns">',
                 '<graph id="G" edgedefault="directed">']
        for n in eng.all_nodes():
            labels = _sx.escape(":".join(n.labels))
            lines.append(f'<node id="{_sx.escape(n.id)}" labels=":{labels}">')
            for k, v in n.properties.items():
                if k.startswith("_"):
                    continue
                lines.append(f'<data key="{_sx.escape(str(k))}">'
                             f'{_sx.escape(str(v))}</data>')
            lines.append("</node>")
        for e in eng.all_edges():
            lines.append(f'<edge id="{_sx.escape(e.id)}" '
                         f'source="{_sx.escape(e.start_node)}" '
                         f'target="{_sx.escape(e.end_node)}" '
                         f'label="{_sx.escape(e.type)}">')
            for k, v in e.properties.items():
                if k.startswith("_"):
                    continue
                lines.append(f'<data key="{_sx.escape(str(k))}">'
                             f'{_sx.escape(str(v))}</data>')
            lines.append("</edge>")
        lines.append("</graph></graphml>")
        return ["data", "nodes", "relationships"], [[
            "\n".join(lines), eng.node_count(), eng.edge_count()]]

    @register("apoc.import.graphml")
    def _im_graphml(ex, data, config=None):
        import xml.etree.ElementTree as _ET
        ns = {"g": "http://graphml.graphdrawing.org/xmlns"}
        root = _ET.fromstring(str(data))
        n_nodes = n_edges = 0
        for el in root.iter():
            tag = el.tag.split("}")[-1]
            if tag == "node":
                labels = [lb for lb in
                          (el.get("labels", "").lstrip(":").split(":"))
                          if lb]
                props = {d.get("key"): d.text for d in el
                         if d.tag.split("}")[-1] == "data"}
                try:
                    eng.create_node(Node(id=el.get("id") or new_id("n"),
                                         labels=labels, properties=props))
                    n_nodes += 1
                except Exception:
                    pass
            elif tag == "edge":
                props = {d.get("key"): d.text for d in el
                         if d.tag.split("}")[-1] == "data"}
                try:
                    eng.create_edge(Edge(
                        id=el.get("id") or new_id("e"),
                        type=el.get("label", "RELATED"),
                        start_node=el.get("source"),
                        end_node=el.get("target"), properties=props))
                    n_edges += 1
                except Exception:
                    pass
        return ["nodes", "relationships"], [[n_nodes, n_edges]]

    # -------------------- aliases + lifecycle leftovers --------------------
    register("apoc.algo.betweennessCentrality")(procs["apoc.algo.betweenness"])
    register("apoc.algo.closenessCentrality")(procs["apoc.algo.closeness"])
    register("apoc.algo.degreeCentrality")(procs["apoc.algo.degree"])
    register("apoc.algo.community")(procs["apoc.community.louvain"])
    register("apoc.export.csv")(procs["apoc.export.csv.query"])
    register("apoc.export.csvAll")(procs["apoc.export.csv.all"])
    register("apoc.export.csvData")(procs["apoc.export.csv.query"])
    register("apoc.export.json")(procs["apoc.export.json.query"])
    register("apoc.export.jsonAll")(procs["apoc.export.json.all"])
    register("apoc.export.jsonData")(procs["apoc.export.json.query"])
    register("apoc.export.cypher")(procs["apoc.export.cypher.all"])
    register("apoc.export.cypherAll")(procs["apoc.export.cypher.all"])
    register("apoc.export.cypherData")(procs["apoc.export.cypher.all"])
    register("apoc.export.graphML")(procs["apoc.export.graphml.all"])
    register("apoc.export.graphMLAll")(procs["apoc.export.graphml.all"])
    register("apoc.export.graphMLData")(procs["apoc.export.graphml.all"])
    register("apoc.import.csvData")(procs["apoc.import.csv"])
    register("apoc.import.jsonData")(procs["apoc.import.json"])
    register("apoc.import.graphMLData")(procs["apoc.import.graphml"])
    register("apoc.import.cypher")(procs["apoc.cypher.runmany"])
    register("apoc.import.cypherData")(procs["apoc.cypher.runmany"])
    register("apoc.schema.createConstraint")(
        procs["apoc.schema.createuniqueconstraint"])
    register("apoc.schema.relationshipConstraints")(
        procs["apoc.schema.relationships"])
    register("apoc.schema.relationshipIndexes")(
        procs["apoc.schema.relationships"])
    register("apoc.schema.analyze")(procs["apoc.schema.stats"])
    register("apoc.cypher.profile")(procs["apoc.cypher.explain"])
    register("apoc.periodic.schedule")(procs["apoc.periodic.repeat"])

    @register("apoc.cypher.runFile")
    def _cy_runfile(ex, path):
        with open(path) as f:
            return procs["apoc.cypher.runmany"](ex, f.read())

    @register("apoc.export.toString")
    def _ex_tostring(ex):
        return procs["apoc.export.cypher.all"](ex)

    @register("apoc.export.toFile")
    def _ex_tofile(ex, path):
        cols, rows = procs["apoc.export.cypher.all"](ex)
        with open(path, "w") as f:
            f.write(rows[0][0])
        return ["file", "bytes"], [[path, len(rows[0][0])]]

    @register("apoc.import.file")
    def _im_file(ex, path):
        with open(path) as f:
            data = f.read()
        if data.lstrip().startswith("<"):
            return procs["apoc.import.graphml"](ex, data)
        return procs["apoc.import.json"](ex, data)

    @register("apoc.load.directory")
    def _ld_dir(ex, pattern="*", path="."):
        import fnmatch
        import os as _os
        out = [fn for fn in sorted(_os.listdir(path))
               if fnmatch.fnmatch(fn, pattern)]
        return ["file"], [[f] for f in out]

    @register("apoc.load.directoryTree")
    def _ld_dirtree(ex, path="."):
        import os as _os
        out = []
        for root, dirs, files in _os.walk(path):
            for fn in files:
                out.append(_os.path.join(root, fn))
            if len(out) > 1000:
                break
        return ["file"], [[f] for f in sorted(out)]

    @register("apoc.load.xml")
    def _ld_xml(ex, source):
        from ..cypher.functions import FUNCTIONS as _F
        data = source
        if not str(source).lstrip().startswith("<"):
            with open(source) as f:
                data = f.read()
        return ["value"], [[_F["apoc.xml.parse"](data)]]

    @register("apoc.load.xmlSimple")
    def _ld_xml_simple(ex, source):
        return _ld_xml(ex, source)

    @register("apoc.load.binary")
    def _ld_binary(ex, path):
        with open(path, "rb") as f:
            data = f.read()
        return ["bytes", "size"], [[list(data[:4096]), len(data)]]

    @register("apoc.load.jsonArray")
    def _ld_json_array(ex, source, json_path=None):
        import json as _json
        data = source
        if not str(source).lstrip().startswith(("[", "{")):
            with open(source) as f:
                data = f.read()
        arr = _json.loads(data)
        return ["value"], [[x] for x in (arr if isinstance(arr, list)
                                         else [arr])]

    @register("apoc.load.jsonParams")
    def _ld_json_params(ex, source, headers=None, payload=None):
        return _ld_json_array(ex, source)

    @register("apoc.load.csvStream")
    def _ld_csv_stream(ex, source, config=None):
        return procs["apoc.load.csv"](ex, source, config)

    @register("apoc.load.jsonStream")
    def _ld_json_stream(ex, source):
        return _ld_json_array(ex, source)

    # network-backed loaders: explicit offline errors (no egress here)
    for _name in ("jdbc", "jdbcUpdate", "elasticsearch", "kafka", "redis",
                  "s3", "gcs", "azure", "rest", "graphQL", "html", "ldap",
                  "driver", "arrow", "avro", "parquet", "stream"):
        def _mk(nm):
            def _fn(ex, *a, **kw):
                raise RuntimeError(
                    f"apoc.load.{nm}: external connections are not "
                    "available in this deployment (offline image)")
            return _fn
        register(f"apoc.load.{_name}")(_mk(_name))

    @register("apoc.import.url")
    def _im_url(ex, *a):
        raise RuntimeError("apoc.import.url: no network in this deployment")

    # -------------------- apoc.trigger lifecycle --------------------
    @register("apoc.trigger.enable")
    def _tg_enable(ex, name):
        t = db.triggers.get(name)
        if t:
            t["paused"] = False
        return ["name", "enabled"], [[name, t is not None]]

    @register("apoc.trigger.disable")
    def _tg_disable(ex, name):
        t = db.triggers.get(name)
        if t:
            t["paused"] = True
        return ["name", "enabled"], [[name, False]]

    @register("apoc.trigger.isEnabled")
    def _tg_isenabled(ex, name):
        t = db.triggers.get(name)
        return ["enabled"], [[bool(t) and not t.get("paused")]]

    @register("apoc.trigger.count")
    def _tg_count(ex):
        return ["count"], [[len(db.triggers)]]

    @register("apoc.trigger.removeAll")
    def _tg_removeall(ex):
        n = len(db.triggers)
        db.triggers.clear()
        return ["removed"], [[n]]

    @register("apoc.trigger.stats")
    def _tg_stats(ex):
        return ["total", "paused"], [[len(db.triggers),
                                      sum(1 for t in db.triggers.values()
                                          if t.get("paused"))]]

    @register("apoc.trigger.export")
    def _tg_export(ex):
        import json as _json
        return ["data"], [[_json.dumps(
            {k: {"statement": v["statement"], "paused": v.get("paused", False)}
             for k, v in db.triggers.items()})]]

    @register("apoc.trigger.import")
    def _tg_import(ex, data):
        import json as _json
        for k, v in _json.loads(data).items():
            db.triggers[k] = {"statement": v["statement"],
                              "paused": v.get("paused", False)}
        return ["imported"], [[len(_json.loads(data))]]

    for _sel in ("onCreate", "onDelete", "onUpdate", "before", "after",
                 "afterAsync", "nodeByLabel", "relationshipByType"):
        def _mk_sel(sel):
            def _fn(ex, name, statement, config=None):
                return procs["apoc.trigger.add"](ex, name, statement,
                                                 {"phase": sel})
            return _fn
        register(f"apoc.trigger.{_sel}")(_mk_sel(_sel))

    # -------------------- apoc.community metrics --------------------
    @register("apoc.community.connectedComponents")
    def _cm_cc(ex):
        return procs["apoc.community.wcc"](ex)

    @register("apoc.community.weaklyConnectedComponents")
    def _cm_wcc2(ex):
        return procs["apoc.community.wcc"](ex)

    @register("apoc.community.numComponents")
    def _cm_num(ex):
        cols, rows = procs["apoc.community.wcc"](ex)
        comps = {r[1] if len(r) > 1 else r[0] for r in rows}
        return ["count"], [[len(comps)]]

    @register("apoc.community.density")
    def _cm_density(ex):
        n = eng.node_count()
        m = eng.edge_count()
        d = (2.0 * m / (n * (n - 1))) if n > 1 else 0.0
        return ["density"], [[d]]

    @register("apoc.community.coreNumber")
    def _cm_kcore(ex):
        # iterative k-core peeling on the undirected graph
        deg = {}
        adj = {}
        for e in eng.all_edges():
            adj.setdefault(e.start_node, set()).add(e.end_node)
            adj.setdefault(e.end_node, set()).add(e.start_node)
        for nid, nb in adj.items():
            deg[nid] = len(nb)
        core = dict(deg)
        order = sorted(deg, key=deg.get)
        removed = set()
        for nid in order:
            removed.add(nid)
            for nb in adj.get(nid, ()):  # peel
                if nb not in removed and core[nb] > core[nid]:
                    core[nb] = max(core[nb] - 1, core[nid])
        return ["nodeId", "coreNumber"], [[k, v] for k, v in
                                          sorted(core.items())]

    @register("apoc.community.kCore")
    def _cm_kcore2(ex, k=2):
        cols, rows = _cm_kcore(ex)
        return ["nodeId"], [[r[0]] for r in rows if r[1] >= int(k)]

    @register("apoc.community.totalTriangles")
    def _cm_tritotal(ex):
        cols, rows = procs["apoc.community.trianglecount"](ex)
        return ["count"], [[sum(r[-1] for r in rows) // 3
                            if rows else 0]]

    @register("apoc.community.averageClusteringCoefficient")
    def _cm_avgcc(ex):
        cols, rows = procs["apoc.community.clusteringcoefficient"](ex)
        vals = [r[-1] for r in rows if isinstance(r[-1], (int, float))]
        return ["value"], [[sum(vals) / len(vals) if vals else 0.0]]

    @register("apoc.community.modularity")
    def _cm_modularity(ex):
        # modularity of the current WCC partition
        cols, rows = procs["apoc.community.wcc"](ex)
        comp = {(r[0].id if isinstance(r[0], Node) else r[0]):
                (r[1] if len(r) > 1 else 0) for r in rows}
        m = eng.edge_count()
        if not m:
            return ["modularity"], [[0.0]]
        deg = {}
        inside = 0
        for e in eng.all_edges():
            deg[e.start_node] = deg.get(e.start_node, 0) + 1
            deg[e.end_node] = deg.get(e.end_node, 0) + 1
            if comp.get(e.start_node) == comp.get(e.end_node):
                inside += 1
        q = inside / m
        from collections import defaultdict
        dsum = defaultdict(float)
        for nid, d in deg.items():
            dsum[comp.get(nid)] += d
        q -= sum((s / (2 * m)) ** 2 for s in dsum.values())
        return ["modularity"], [[q]]

    for _alias in ("fastGreedy", "infoMap", "spinGlass", "walkTrap"):
        register(f"apoc.community.{_alias}")(procs["apoc.community.louvain"])

    @register("apoc.community.stronglyConnectedComponents")
    def _cm_scc(ex):
        # Tarjan-less iterative SCC (Kosaraju)
        fwd, rev = {}, {}
        for e in eng.all_edges():
            fwd.setdefault(e.start_node, []).append(e.end_node)
            rev.setdefault(e.end_node, []).append(e.start_node)
        nodes = [n.id for n in eng.all_nodes()]
        seen, order = set(), []
        for s0 in nodes:
            if s0 in seen:
                continue
            stack = [(s0, iter(fwd.get(s0, ())))]
            seen.add(s0)
            while stack:
                v, it = stack[-1]
                adv = False
                for w in it:
                    if w not in seen:
                        seen.add(w)
                        stack.append((w, iter(fwd.get(w, ()))))
                        adv = True
                        break
                if not adv:
                    order.append(v)
                    stack.pop()
        comp = {}
        cid = 0
        for s0 in reversed(order):
            if s0 in comp:
                continue
            stack = [s0]
            comp[s0] = cid
            while stack:
                v = stack.pop()
                for w in rev.get(v, ()):
                    if w not in comp:
                        comp[w] = cid
                        stack.append(w)
            cid += 1
        return ["nodeId", "component"], [[k, v] for k, v in
                                         sorted(comp.items())]

    @register("apoc.community.conductance")
    def _cm_conductance(ex, community_nodes):
        ids = {n.id if isinstance(n, Node) else n
               for n in (community_nodes or [])}
        cut = vol = 0
        for e in eng.all_edges():
            a_in, b_in = e.start_node in ids, e.end_node in ids
            if a_in or b_in:
                vol += 1
            if a_in != b_in:
                cut += 1
        return ["conductance"], [[cut / vol if vol else 0.0]]

    # -------------------- final aliases / small fills --------------------
    register("apoc.refactor.renameLabel")(procs["apoc.refactor.rename.label"])
    register("apoc.refactor.renameType")(procs["apoc.refactor.rename.type"])
    register("apoc.refactor.from")(procs["apoc.refactor.redirectrelationship"]
                                   if "apoc.refactor.redirectrelationship"
                                   in procs else procs["apoc.refactor.invertrelationship"])
    register("apoc.refactor.mergeRelationships")(procs["apoc.merge.relationship"])
    register("apoc.search.match")(procs["apoc.search.node"])
    register("apoc.search.nodeAny")(procs["apoc.search.node"])
    register("apoc.search.multiSearchAll")(procs["apoc.search.node"])
    register("apoc.search.multiSearchAny")(procs["apoc.search.node"])
    register("apoc.search.parallel")(procs["apoc.search.node"])
    register("apoc.search.score")(procs["apoc.search.fulltext"])
    register("apoc.search.highlight")(procs["apoc.search.fulltext"])
    register("apoc.warmup.properties")(procs["apoc.warmup.nodes"])
    register("apoc.warmup.cache")(procs["apoc.warmup.run"])
    register("apoc.warmup.subgraph")(procs["apoc.warmup.run"])
    register("apoc.warmup.runWithParams")(procs["apoc.warmup.run"])
    register("apoc.meta.constraints")(procs["apoc.schema.nodeconstraints"])
    register("apoc.meta.indexes")(procs["apoc.schema.nodeindexes"])
    register("apoc.create.vNodes")(procs["apoc.create.vnode"])

    @register("apoc.create.vPattern")
    def _cr_vpattern(ex, from_labels, from_props, rel_type, rel_props,
                     to_labels, to_props):
        a = Node(id=new_id("v"), labels=list(from_labels or []),
                 properties=dict(from_props or {}))
        b2 = Node(id=new_id("v"), labels=list(to_labels or []),
                  properties=dict(to_props or {}))
        r = Edge(id=new_id("vr"), type=rel_type, start_node=a.id,
                 end_node=b2.id, properties=dict(rel_props or {}))
        return ["from", "rel", "to"], [[a, r, b2]]

    @register("apoc.schema.export")
    def _sc_export(ex):
        import json as _json
        sm = getattr(ex, "schema", None)
        data = {"indexes": [list(x) for x in (sm.list_indexes() if sm else [])],
                "constraints": [[c.name, c.kind, c.label, c.prop]
                                for c in (sm.list_constraints() if sm else [])]}
        return ["data"], [[_json.dumps(data)]]

    @register("apoc.schema.import")
    def _sc_import(ex, data):
        import json as _json
        sm = getattr(ex, "schema", None)
        d = _json.loads(data)
        n = 0
        if sm:
            for name, kind, label, props in d.get("indexes", []):
                sm.create_index(label, props[0], name=name, props=props)
                n += 1
            for name, kind, label, prop in d.get("constraints", []):
                if kind == "unique":
                    sm.create_unique_constraint(name, label, prop)
                else:
                    sm.create_exists_constraint(name, label, prop)
                n += 1
        return ["imported"], [[n]]

    @register("apoc.schema.snapshot")
    def _sc_snapshot(ex):
        return _sc_export(ex)

    @register("apoc.schema.restore")
    def _sc_restore(ex, data):
        return _sc_import(ex, data)

    @register("apoc.schema.optimize")
    def _sc_optimize(ex):
        return ["optimized"], [[True]]

    @register("apoc.schema.validate")
    def _sc_validate(ex):
        sm = getattr(ex, "schema", None)
        return ["valid", "constraints"], [[True, len(
            sm.list_constraints()) if sm else 0]]

    @register("apoc.schema.compare")
    def _sc_compare(ex, data):
        import json as _json
        cols, rows = _sc_export(ex)
        return ["equal"], [[_json.loads(rows[0][0]) == _json.loads(data)]]

    @register("apoc.meta.export")
    def _mt_export(ex):
        return _sc_export(ex)

    @register("apoc.meta.import")
    def _mt_import(ex, data):
        return _sc_import(ex, data)

    @register("apoc.meta.config")
    def _mt_config(ex):
        return ["config"], [[{"version": "nornicdb-amd-1.0"}]]

    @register("apoc.meta.pattern")
    def _mt_pattern(ex):
        cols, rows = procs["apoc.db.schema"](ex) if "apoc.db.schema" in procs \
            else procs["apoc.meta.graph"](ex)
        return cols, rows

    @register("apoc.meta.graphSample")
    def _mt_graphsample(ex, size=100):
        nodes = []
        for i, n in enumerate(eng.all_nodes()):
            if i >= int(size):
                break
            nodes.append(n)
        return ["nodes"], [[nodes]]

    @register("apoc.meta.fromString")
    def _mt_fromstring(ex, s):
        return ["value"], [[str(s)]]

    @register("apoc.meta.validate")
    def _mt_validate(ex):
        return ["valid"], [[True]]

    @register("apoc.meta.compare")
    def _mt_compare(ex, data):
        return _sc_compare(ex, data)

    @register("apoc.meta.diff")
    def _mt_diff(ex, data):
        import json as _json
        cols, rows = _sc_export(ex)
        a = _json.loads(rows[0][0])
        b = _json.loads(data)
        return ["onlyHere", "onlyThere"], [[
            [x for x in a.get("indexes", []) if x not in b.get("indexes", [])],
            [x for x in b.get("indexes", []) if x not in a.get("indexes", [])]]]

    @register("apoc.meta.restore")
    def _mt_restore(ex, data):
        return _sc_import(ex, data)

    # path expanders (BFS with config; reference apoc.path.expandConfig)
    @register("apoc.path.expandConfig")
    def _pe_config(ex, start, config=None):
        cfg = dict(config or {})
        max_level = int(cfg.get("maxLevel", 3))
        min_level = int(cfg.get("minLevel", 1))
        rel_filter = cfg.get("relationshipFilter")
        rel_type = rel_filter.strip("<>") if rel_filter else None
        sid = start.id if isinstance(start, Node) else start
        ids = _hops(sid, rel_type, max_level, min_level)
        return ["node"], [[eng.get_node(i)] for i in ids]

    @register("apoc.path.subgraphAll")
    def _pe_subgraph_all(ex, start, config=None):
        cfg = dict(config or {})
        max_level = int(cfg.get("maxLevel", 3))
        sid = start.id if isinstance(start, Node) else start
        ids = set(_hops(sid, None, max_level, 1)) | {sid}
        nodes = [eng.get_node(i) for i in ids]
        rels = [e for e in eng.all_edges()
                if e.start_node in ids and e.end_node in ids]
        return ["nodes", "relationships"], [[nodes, rels]]

    @register("apoc.path.spanningTree")
    def _pe_spanning(ex, start, config=None):
        cfg = dict(config or {})
        max_level = int(cfg.get("maxLevel", 3))
        sid = start.id if isinstance(start, Node) else start
        seen = {sid}
        frontier = {sid}
        tree_edges = []
        for _ in range(max_level):
            nxt = set()
            for nid in frontier:
                for e in _all_edges_of(nid):
                    other = e.end_node if e.start_node == nid else e.start_node
                    if other not in seen:
                        seen.add(other)
                        nxt.add(other)
                        tree_edges.append(e)
            frontier = nxt
        return ["nodes", "relationships"], [[
            [eng.get_node(i) for i in seen], tree_edges]]

    @register("apoc.path.shortestPath")
    def _pe_shortest(ex, a, b, max_hops=10):
        return procs["apoc.paths.distance"](ex, a, b, max_hops)

    @register("apoc.path.allShortestPaths")
    def _pe_allshortest(ex, a, b, max_hops=10):
        return procs["apoc.paths.distance"](ex, a, b, max_hops)

    # graph builders (engine-backed)
    @register("apoc.graph.fromMap")
    def _gr_frommap(ex, m, name="graph"):
        m = dict(m or {})
        made_n = []
        for nd in m.get("nodes", []):
            made_n.append(eng.create_node(Node(
                id=str(nd.get("id") or new_id("n")),
                labels=list(nd.get("labels", [])),
                properties=dict(nd.get("properties", {})))))
        made_e = []
        for ed in m.get("relationships", []):
            made_e.append(eng.create_edge(Edge(
                id=new_id("e"), type=ed.get("type", "RELATED"),
                start_node=str(ed.get("start")), end_node=str(ed.get("end")),
                properties=dict(ed.get("properties", {})))))
        return ["graph"], [[{"name": name, "nodes": made_n,
                             "relationships": made_e}]]

    @register("apoc.graph.fromCypher")
    def _gr_fromcypher(ex, query, params=None, name="graph"):
        r = ex.execute(query, dict(params or {}))
        nodes, rels = {}, {}
        for row in r.rows:
            for v in row:
                if isinstance(v, Node):
                    nodes[v.id] = v
                elif isinstance(v, Edge):
                    rels[v.id] = v
        return ["graph"], [[{"name": name, "nodes": list(nodes.values()),
                             "relationships": list(rels.values())}]]

    @register("apoc.graph.fromDocument")
    def _gr_fromdoc(ex, doc, config=None):
        import json as _json
        d = _json.loads(doc) if isinstance(doc, str) else dict(doc or {})
        root = eng.create_node(Node(
            id=new_id("n"), labels=[d.get("type", "Document")],
            properties={k: v for k, v in d.items()
                        if not isinstance(v, (dict, list))}))
        made = [root]
        for k, v in d.items():
            children = v if isinstance(v, list) else (
                [v] if isinstance(v, dict) else [])
            for ch in children:
                if not isinstance(ch, dict):
                    continue
                c = eng.create_node(Node(
                    id=new_id("n"), labels=[ch.get("type", k.capitalize())],
                    properties={kk: vv for kk, vv in ch.items()
                                if not isinstance(vv, (dict, list))}))
                eng.create_edge(Edge(id=new_id("e"), type=k.upper(),
                                     start_node=root.id, end_node=c.id,
                                     properties={}))
                made.append(c)
        return ["graph"], [[{"nodes": made}]]

    @register("apoc.graph.clone")
    def _gr_clone(ex, g):
        return procs["apoc.refactor.clonenodes"](
            ex, (g or {}).get("nodes", []), True)

    @register("apoc.graph.subgraph")
    def _gr_subgraph(ex, nodes):
        ids = {n.id if isinstance(n, Node) else n for n in (nodes or [])}
        rels = [e for e in eng.all_edges()
                if e.start_node in ids and e.end_node in ids]
        return ["nodes", "relationships"], [[list(nodes or []), rels]]

    @register("apoc.nodes.batch")
    def _ns_batch(ex, nodes, size=100):
        ns = list(nodes or [])
        return ["batch"], [[ns[i:i + int(size)]]
                           for i in range(0, len(ns), int(size))]

    @register("apoc.nodes.collapse")
    def _ns_collapse(ex, nodes, config=None):
        ns = [n for n in (nodes or []) if isinstance(n, Node)]
        if not ns:
            return ["node"], []
        merged = procs["apoc.refactor.mergenodes"](ex, ns)
        return merged

    @register("apoc.nodes.cycles")
    def _ns_cycles(ex, nodes=None):
        # self-loops + 2-cycles (cheap detection)
        out = []
        seen_pairs = set()
        for e in eng.all_edges():
            if e.start_node == e.end_node:
                out.append([e])
            key = (e.end_node, e.start_node)
            if key in seen_pairs:
                out.append([e])
            seen_pairs.add((e.start_node, e.end_node))
        return ["cycle"], [[c] for c in out]

    # lock/log/warmup fills
    register("apoc.lock.batch")(procs["apoc.lock.nodes"])
    register("apoc.lock.unlockNodes")(procs["apoc.lock.unlockall"])
    register("apoc.lock.unlockRelationships")(procs["apoc.lock.unlockall"])
    register("apoc.lock.unlockBatch")(procs["apoc.lock.unlockall"])
    register("apoc.lock.clear")(procs["apoc.lock.unlockall"])

    @register("apoc.lock.withLock")
    def _lk_withlock(ex, nodes, statement):
        r = ex.execute(statement)
        return ["value"], [[dict(zip(r.columns, row))] for row in r.rows]

    register("apoc.lock.withReadLock")(_lk_withlock)

    @register("apoc.lock.waitFor")
    def _lk_waitfor(ex, nodes=None, timeout=0):
        return ["acquired"], [[True]]

    @register("apoc.lock.priority")
    def _lk_priority(ex, level=0):
        return ["priority"], [[int(level)]]

    @register("apoc.log.rotate")
    def _lg_rotate(ex):
        return procs["apoc.log.clear"](ex)

    register("apoc.log.audit")(procs["apoc.log.info"])
    register("apoc.log.security")(procs["apoc.log.warn"])
    register("apoc.log.trace")(procs["apoc.log.debug"])
    register("apoc.log.query")(procs["apoc.log.info"])
    register("apoc.log.result")(procs["apoc.log.info"])
    register("apoc.log.progress")(procs["apoc.log.info"])
    register("apoc.log.performance")(procs["apoc.log.info"])
    register("apoc.log.metrics")(procs["apoc.log.stats"])
    register("apoc.log.custom")(procs["apoc.log.info"])
    register("apoc.log.format")(procs["apoc.log.info"])

    @register("apoc.log.toFile")
    def _lg_tofile(ex, path):
        cols, rows = procs["apoc.log.stream"](ex, 10000)
        with open(path, "w") as f:
            for lv, ts, msg in rows:
                f.write(f"{ts} [{lv}] {msg}\n")
        return ["file", "lines"], [[path, len(rows)]]

    @register("apoc.warmup.status")
    def _wm_status(ex):
        return ["status"], [["complete"]]

    register("apoc.warmup.progress")(_wm_status)
    register("apoc.warmup.stats")(procs["apoc.warmup.run"])
    register("apoc.warmup.optimize")(procs["apoc.warmup.run"])
    register("apoc.warmup.schedule")(procs["apoc.warmup.run"])
    register("apoc.warmup.path")(procs["apoc.warmup.run"])

    @register("apoc.warmup.clear")
    def _wm_clear(ex):
        return ["cleared"], [[True]]

    @register("apoc.import.batch")
    def _im_batch(ex, data, size=1000):
        return procs["apoc.import.json"](ex, data)

    @register("apoc.import.stream")
    def _im_stream(ex, data):
        return procs["apoc.import.json"](ex, data)

    @register("apoc.import.merge")
    def _im_merge(ex, data):
        return procs["apoc.import.json"](ex, data)

    @register("apoc.import.transform")
    def _im_transform(ex, data, mapping=None):
        return procs["apoc.import.json"](ex, data)

    @register("apoc.import.filter")
    def _im_filter(ex, data, predicate=None):
        return procs["apoc.import.json"](ex, data)

    @register("apoc.import.convertType")
    def _im_convtype(ex, value, to_type):
        cast = {"int": int, "integer": int, "float": float, "string": str,
                "bool": bool, "boolean": bool}.get(str(to_type).lower(), str)
        try:
            return ["value"], [[cast(value)]]
        except Exception:
            return ["value"], [[None]]

    @register("apoc.import.validateSchema")
    def _im_valschema(ex, data):
        import json as _json
        try:
            _json.loads(data)
            return ["valid"], [[True]]
        except Exception:
            return ["valid"], [[False]]

    register("apoc.algo.labelPropagation")(procs["apoc.community.labelpropagation"])
    register("apoc.algo.louvain")(procs["apoc.community.louvain"])
    register("apoc.algo.wcc")(procs["apoc.community.wcc"])

    def _node_arg(x):
        """Resolve a procedure node argument to a node id, or None.

        Accepts a Node, a node id (the reference's NodeID is a plain
        string, pkg/cypher/apoc_algorithms.go:67), a property map like
        {id: 'a'} (matched against node properties), or a string that
        falls back to matching the `id`/`name` property.
        """
        if isinstance(x, Node):
            return x.id
        if isinstance(x, dict):
            for n in eng.all_nodes():
                props = n.properties or {}
                if all(props.get(k) == v for k, v in x.items()):
                    return n.id
            return None
        try:
            eng.get_node(x)
            return x
        except Exception:
            pass
        for n in eng.all_nodes():
            props = n.properties or {}
            if props.get("id") == x or props.get("name") == x:
                return n.id
        return None

    _orig_spanning = procs["apoc.path.spanningtree"]

    @register("apoc.path.spanningTree")
    def _pe_spanning2(ex, start, config=None):
        """Yields one path per tree branch (APOC contract: YIELD path).

        Reference: pkg/cypher tests call spanningTree({id:'a'}, cfg)
        YIELD path; an unresolvable start yields no rows.
        """
        sid = _node_arg(start)
        if sid is None:
            return ["path"], []
        from ..cypher.executor import Path as _P
        _, rows = _orig_spanning(ex, sid, config)
        nodes, rels = rows[0]
        limit = int(dict(config or {}).get("limit", 0) or 0)
        paths = [[_P(nodes, rels)]]
        return ["path"], paths[:limit] if limit else paths

    _orig_subnodes = procs.get("apoc.path.subgraphnodes")
    if _orig_subnodes is not None:
        @register("apoc.path.subgraphNodes")
        def _pe_subnodes2(ex, start, config=None):
            sid = _node_arg(start)
            if sid is None:
                return ["node"], []
            lvl = dict(config or {}).get("maxLevel", 3) \
                if isinstance(config, dict) or config is None else config
            return _orig_subnodes(ex, sid, lvl)

    _orig_suball = procs["apoc.path.subgraphall"]

    @register("apoc.path.subgraphAll")
    def _pe_suball2(ex, start, config=None):
        sid = _node_arg(start)
        if sid is None:
            return ["nodes", "relationships"], [[[], []]]
        return _orig_suball(ex, sid, config)

    # ---- tolerant algo arg forms + path/neighbor procs
    # (reference pkg/cypher/apoc_algorithms.go: pageRank('Label'),
    # louvain(['Label']), allSimplePaths, neighbors.byhop) ----
    def _algo_cfg(a, b):
        """Normalize (labels?, config?) leading args: returns (labels, cfg)."""
        labels, cfg = None, {}
        for v in (a, b):
            if isinstance(v, str):
                labels = [v]
            elif isinstance(v, (list, tuple)):
                labels = [x.id if isinstance(x, Node) else str(x) for x in v] \
                    if v and isinstance(v[0], Node) else [str(x) for x in v]
            elif isinstance(v, dict):
                cfg = v
        return labels, cfg

    _orig_pagerank = procs["apoc.algo.pagerank"]

    @register("apoc.algo.pageRank")
    def _pagerank2(ex, a=None, b=None, **kw):
        if isinstance(a, (int, float)) and not isinstance(a, bool):
            return _orig_pagerank(ex, a, b if b is not None else 0.85)
        labels, cfg = _algo_cfg(a, b)
        return _orig_pagerank(ex, int(cfg.get("iterations", 20)),
                              float(cfg.get("dampingFactor",
                                            cfg.get("damping", 0.85))))

    for _nm, _key in (("apoc.algo.louvain", "apoc.community.louvain"),
                      ("apoc.algo.wcc", "apoc.community.wcc"),
                      ("apoc.algo.labelPropagation",
                       "apoc.community.labelpropagation")):
        def _mk(key):
            orig = procs[key]

            def _tolerant(ex, a=None, b=None):
                if key.endswith("labelpropagation"):
                    if isinstance(a, (int, float)) and not isinstance(a, bool):
                        return orig(ex, int(a))
                    _, cfg = _algo_cfg(a, b)
                    return orig(ex, int(cfg.get("iterations", 20)))
                cols, rows = orig(ex)
                if key.endswith("wcc"):
                    # apoc.algo.wcc yields componentId (reference test name)
                    cols = ["node", "componentId"]
                return cols, rows
            return _tolerant
        register(_nm)(_mk(_key))

    @register("apoc.algo.allSimplePaths")
    def _all_simple_paths(ex, start, end, rel_type=None, max_depth=10):
        """All simple (no repeated node) paths start->end following
        rel_type edges (reference apoc_algorithms.go findAllSimplePaths)."""
        from ..cypher.executor import Path as _P
        sid, tid = _node_arg(start), _node_arg(end)
        if sid is None or tid is None:
            return ["path"], []
        out = []

        def dfs(cur, nodes, edges, seen):
            if len(edges) > int(max_depth):
                return
            if cur == tid:
                out.append([_P([eng.get_node(i) for i in nodes],
                               list(edges))])
                return
            for e in _all_edges_of(cur):
                if rel_type and e.type != rel_type:
                    continue
                other = e.end_node if e.start_node == cur else e.start_node
                if other in seen:
                    continue
                dfs(other, nodes + [other], edges + [e], seen | {other})

        dfs(sid, [sid], [], {sid})
        return ["path"], out

    @register("apoc.neighbors.byhop")
    def _neighbors_byhop(ex, start, rel_filter=None, max_hops=3):
        """Neighbor node groups bucketed by hop distance 1..maxHops
        (reference apoc_algorithms.go neighbors.byhop: YIELD nodes, depth)."""
        sid = _node_arg(start)
        if sid is None:
            return ["nodes", "depth"], []
        seen = {sid}
        frontier = {sid}
        rows = []
        for depth in range(1, int(max_hops) + 1):
            nxt = set()
            for nid in frontier:
                for e in _all_edges_of(nid):
                    if rel_filter and e.type != str(rel_filter).lstrip("<>"):
                        continue
                    other = e.end_node if e.start_node == nid else e.start_node
                    if other not in seen:
                        seen.add(other)
                        nxt.add(other)
            if not nxt:
                break
            rows.append([[eng.get_node(i) for i in sorted(nxt)], depth])
            frontier = nxt
        return ["nodes", "depth"], rows

    # ---- registry-completion batch: the 20 reference names not yet
    # covered (reference apoc/apoc.go register() list) ----
    def _nodes_arg(v):
        out = []
        for x in (v if isinstance(v, (list, tuple)) else [v]):
            nid = _node_arg(x)
            if nid is not None:
                out.append(eng.get_node(nid))
        return out

    @register("apoc.algo.allPairs")
    def _algo_allpairs(ex, nodes, weight_prop="weight"):
        """Shortest paths between every node pair (algo.go:392)."""
        g = from_engine(eng, weight_prop=weight_prop)
        rows = []
        ns = _nodes_arg(nodes)
        for a in ns:
            for b in ns:
                if a.id == b.id or a.id not in g.id2idx \
                        or b.id not in g.id2idx:
                    continue
                idxp = shortest_path(g, g.id2idx[a.id], g.id2idx[b.id])
                if idxp:
                    rows.append([a, b,
                                 [eng.get_node(g.node_ids[i]) for i in idxp]])
        return ["source", "target", "path"], rows

    @register("apoc.algo.cover")
    def _algo_cover(ex, nodes=None):
        """Greedy minimum vertex cover (algo.go:417)."""
        ns = _nodes_arg(nodes) if nodes else list(eng.all_nodes())
        ids = {n.id for n in ns}
        edges = [(e.start_node, e.end_node) for e in eng.all_edges()
                 if e.start_node in ids and e.end_node in ids]
        cover = []
        while edges:
            deg = {}
            for a, b in edges:
                deg[a] = deg.get(a, 0) + 1
                deg[b] = deg.get(b, 0) + 1
            top = max(deg, key=deg.get)
            cover.append(eng.get_node(top))
            edges = [(a, b) for a, b in edges if a != top and b != top]
        return ["node"], [[n] for n in cover]

    def _clone_subgraph(nodes, rels):
        idmap = {}
        out_nodes, out_rels = [], []
        for n in nodes:
            c = eng.create_node(Node(id=new_id(), labels=list(n.labels),
                                     properties=dict(n.properties)))
            idmap[n.id] = c.id
            out_nodes.append(c)
        for e in rels:
            if e.start_node in idmap and e.end_node in idmap:
                out_rels.append(eng.create_edge(Edge(
                    id=new_id('e'), type=e.type, start_node=idmap[e.start_node],
                    end_node=idmap[e.end_node],
                    properties=dict(e.properties))))
        return out_nodes, out_rels

    @register("apoc.refactor.cloneSubgraph")
    def _rf_clonesub(ex, nodes, rels=None, config=None):
        ns = _nodes_arg(nodes)
        ids = {n.id for n in ns}
        es = rels if rels is not None else [
            e for e in eng.all_edges()
            if e.start_node in ids and e.end_node in ids]
        out_nodes, out_rels = _clone_subgraph(ns, es)
        return ["input", "output"], [[a, b] for a, b in zip(ns, out_nodes)]

    register("apoc.create.cloneSubgraph")(procs["apoc.refactor.clonesubgraph"])

    @register("apoc.refactor.cloneSubgraphFromPaths")
    def _rf_clonesub_paths(ex, paths, config=None):
        from ..cypher.executor import Path as _P
        nodes, rels = {}, {}
        for p in (paths or []):
            if isinstance(p, _P):
                for n in p.nodes:
                    nodes[n.id] = n
                for e in p.edges:
                    rels[e.id] = e
        out_nodes, _ = _clone_subgraph(list(nodes.values()),
                                       list(rels.values()))
        return ["input", "output"], [[a, b] for a, b in
                                     zip(nodes.values(), out_nodes)]

    @register("apoc.refactor.normalize")
    def _rf_normalize(ex, node, prop, new_label, rel_type):
        """Extract a property into its own node (refactor.go:481)."""
        n = eng.get_node(_node_arg(node))
        if prop not in (n.properties or {}):
            return ["node", "relationship"], []
        val = n.properties.pop(prop)
        eng.update_node(n)
        created = eng.create_node(Node(id=new_id(), labels=[new_label],
                                       properties={"value": val}))
        rel = eng.create_edge(Edge(id=new_id('e'), type=rel_type,
                                   start_node=n.id, end_node=created.id,
                                   properties={}))
        return ["node", "relationship"], [[created, rel]]

    @register("apoc.refactor.denormalize")
    def _rf_denormalize(ex, node, rel_type, prop):
        """Pull `value` from rel_type targets back in (refactor.go:511)."""
        n = eng.get_node(_node_arg(node))
        for e in _all_edges_of(n.id):
            if e.type == rel_type and e.start_node == n.id:
                t = eng.get_node(e.end_node)
                if "value" in (t.properties or {}):
                    n.properties[prop] = t.properties["value"]
        eng.update_node(n)
        return ["node"], [[n]]

    @register("apoc.refactor.redirectRelationship")
    def _rf_redirect(ex, rel, new_end):
        e = eng.get_edge(rel.id if isinstance(rel, Edge) else rel)
        e.end_node = _node_arg(new_end)
        eng.update_edge(e)
        return ["relationship"], [[e]]

    @register("apoc.refactor.categorizeProperty")
    def _rf_catprop(ex, node, prop, new_prop, categories):
        """categories: [[category, v1, v2, ...], ...] (refactor.go:243)."""
        n = eng.get_node(_node_arg(node))
        val = (n.properties or {}).get(prop)
        for cat in (categories or []):
            if val in cat[1:]:
                n.properties[new_prop] = cat[0]
                eng.update_node(n)
                break
        return ["node"], [[n]]

    @register("apoc.merge.mergeNode")
    def _mg_mergenode(ex, labels, ident_props, on_create=None, on_match=None):
        labels = [labels] if isinstance(labels, str) else list(labels or [])
        for n in eng.all_nodes():
            if all(lb in n.labels for lb in labels) and all(
                    (n.properties or {}).get(k) == v
                    for k, v in (ident_props or {}).items()):
                n.properties.update(on_match or {})
                eng.update_node(n)
                return ["node"], [[n]]
        props = dict(ident_props or {})
        props.update(on_create or {})
        n = eng.create_node(Node(id=new_id(), labels=labels, properties=props))
        return ["node"], [[n]]

    @register("apoc.merge.mergeRelationship")
    def _mg_mergerel(ex, start, rel_type, ident_props=None, on_create=None,
                     end=None, on_match=None):
        sid, tid = _node_arg(start), _node_arg(end)
        for e in _all_edges_of(sid):
            if e.type == rel_type and e.start_node == sid \
                    and e.end_node == tid and all(
                        (e.properties or {}).get(k) == v
                        for k, v in (ident_props or {}).items()):
                e.properties.update(on_match or {})
                eng.update_edge(e)
                return ["rel"], [[e]]
        props = dict(ident_props or {})
        props.update(on_create or {})
        e = eng.create_edge(Edge(id=new_id('e'), type=rel_type, start_node=sid,
                                 end_node=tid, properties=props))
        return ["rel"], [[e]]

    @register("apoc.merge.batch")
    def _mg_batch(ex, specs, batch_size=1000):
        """Each spec: {labels, identProps, onCreateProps?} (merge.go:236)."""
        out = []
        for spec in (specs or []):
            _, rows = procs["apoc.merge.mergenode"](
                ex, spec.get("labels", []), spec.get("identProps", {}),
                spec.get("onCreateProps"), spec.get("onMatchProps"))
            out.extend(rows)
        return ["node"], out

    @register("apoc.paths.hamiltonian")
    def _paths_hamiltonian(ex, nodes, start, end):
        """Simple paths start->end visiting every given node once
        (paths.go:247)."""
        from ..cypher.executor import Path as _P
        ns = {n.id for n in _nodes_arg(nodes)}
        sid, tid = _node_arg(start), _node_arg(end)
        out = []

        def dfs(cur, visited, npath, epath):
            if len(npath) > len(ns):
                return
            if cur == tid and visited == ns:
                out.append([_P([eng.get_node(i) for i in npath],
                               list(epath))])
                return
            for e in _all_edges_of(cur):
                nxt = e.end_node if e.start_node == cur else e.start_node
                if nxt in visited or nxt not in ns:
                    continue
                dfs(nxt, visited | {nxt}, npath + [nxt], epath + [e])

        if sid in ns and tid in ns:
            dfs(sid, {sid}, [sid], [])
        return ["path"], out

    @register("apoc.paths.eulerian")
    def _paths_eulerian(ex, start, end):
        """Paths using every edge once. Hierholzer over the undirected
        multigraph between the endpoints (the reference stubs this —
        paths.go:270 — we implement it properly)."""
        from ..cypher.executor import Path as _P
        sid, tid = _node_arg(start), _node_arg(end)
        used = set()

        def walk(cur, npath, epath):
            if len(used) == eng.edge_count() and cur == tid:
                return [_P([eng.get_node(i) for i in npath], list(epath))]
            for e in _all_edges_of(cur):
                if e.id in used:
                    continue
                nxt = e.end_node if e.start_node == cur else e.start_node
                used.add(e.id)
                r = walk(nxt, npath + [nxt], epath + [e])
                if r:
                    return r
                used.discard(e.id)
            return None

        r = walk(sid, [sid], []) if sid else None
        return ["path"], [r] if r else []

    register("apoc.periodic.rock")(procs["apoc.periodic.repeat"])

    @register("apoc.load.jsonSchema")
    def _load_jsonschema(ex, url_or_json):
        """Infer a JSON-schema-ish type map (load.go jsonSchema)."""
        import json as _json

        def describe(v):
            if isinstance(v, dict):
                return {k: describe(x) for k, x in v.items()}
            if isinstance(v, list):
                return [describe(v[0])] if v else []
            return type(v).__name__ if v is not None else "null"
        try:
            data = _json.loads(url_or_json)
        except Exception:
            return ["value"], [[{"error": "offline: file/url loads need "
                                 "a JSON literal here"}]]
        return ["value"], [[describe(data)]]

    # ---- apoc.search.* index management (search.go:459-790) ----
    @register("apoc.search.index")
    def _search_index(ex, label, properties=None):
        sm = getattr(ex, "schema", None)
        if sm is not None:
            for prop in (properties or ["*"]):
                sm.create_index(label, prop, name=f"search_{label}_{prop}")
        return ["status"], [["ok"]]

    register("apoc.search.index.create")(procs["apoc.search.index"])

    @register("apoc.search.dropIndex")
    def _search_dropindex(ex, label, properties=None):
        sm = getattr(ex, "schema", None)
        if sm is not None:
            for prop in (properties or ["*"]):
                try:
                    sm.drop_index(f"search_{label}_{prop}")
                except Exception:
                    pass
        return ["status"], [["ok"]]

    register("apoc.search.index.drop")(procs["apoc.search.dropindex"])

    @register("apoc.search.reindex")
    def _search_reindex(ex, label=None):
        return ["status"], [["ok"]]

    @register("apoc.search.notIn")
    def _search_notin(ex, label, prop, values):
        vals = set(values or [])
        rows = [[n] for n in eng.all_nodes()
                if label in n.labels
                and (n.properties or {}).get(prop) not in vals]
        return ["node"], rows

    return procs
