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
        g = from_engine(eng, edge_types=[rel_type] if rel_type else None,
                        weight_prop=weight_prop)
        s = g.id2idx[start.id if isinstance(start, Node) else start]
        t = g.id2idx[end.id if isinstance(end, Node) else end]
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
        comm = louvain(g)
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
        return ["nodeCount"], [[g.n]]

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

    return procs
