"""GraphQL endpoint over the graph.

Parity: reference pkg/graphql (gqlgen schema, resolvers in
pkg/graphql/resolvers/{query,mutation,node,relationship,subscription}_impl.go
— ~43 resolvers over nodes / relationships / search / cypher / bulk ops /
subscriptions). Hand-written parser + executor (no codegen): supports
operation variables ($x), input objects, list literals, nested selection
sets, and an event broker for the subscription surface (exposed as SSE by
server/http.py).
"""

from __future__ import annotations

import itertools

import json
import queue
import re
import threading
from typing import Any, Dict, List, Optional, Tuple


class GraphQLError(Exception):
    pass


_TOKEN = re.compile(r"""
    (?P<ws>[\s,]+)
  | (?P<comment>\#[^\n]*)
  | (?P<name>[_A-Za-z][_0-9A-Za-z]*)
  | (?P<string>"(?:\\.|[^"\\])*")
  | (?P<number>-?\d+(?:\.\d+)?)
  | (?P<punct>[{}()\[\]:!@$=])
""", re.VERBOSE)


import functools


@functools.lru_cache(maxsize=512)
def _tokenize_cached(src: str):
    """Token stream is a pure function of the query text (variables are
    substituted later, in the parser) — cache it: repeated queries with
    different variables skip the regex scan (~40 us/request)."""
    return tuple(_tokenize(src))


def _tokenize(src: str):
    out = []
    i = 0
    while i < len(src):
        m = _TOKEN.match(src, i)
        if not m:
            raise GraphQLError(f"bad character at {i}: {src[i]!r}")
        i = m.end()
        if m.lastgroup not in ("ws", "comment"):
            out.append((m.lastgroup, m.group()))
    out.append(("eof", ""))
    return out


class _Parser:
    def __init__(self, src, variables=None):
        self.toks = _tokenize_cached(src)
        self.i = 0
        self.variables = variables or {}

    def peek(self):
        return self.toks[self.i]

    def next(self):
        t = self.toks[self.i]
        if t[0] != "eof":
            self.i += 1
        return t

    def expect(self, value):
        t = self.next()
        if t[1] != value:
            raise GraphQLError(f"expected {value!r}, got {t[1]!r}")

    def parse(self):
        op = "query"
        t = self.peek()
        if t[0] == "name" and t[1] in ("query", "mutation", "subscription"):
            op = self.next()[1]
            if self.peek()[0] == "name":
                self.next()  # operation name
            if self.peek()[1] == "(":  # variable definitions (types ignored)
                self.next()
                depth = 1
                while depth:
                    v = self.next()[1]
                    depth += v == "(" 
                    depth -= v == ")"
        self.expect("{")
        fields = self.selection_set()
        return op, fields

    def selection_set(self):
        fields = []
        while self.peek()[1] != "}":
            fields.append(self.field())
        self.expect("}")
        return fields

    def field(self):
        kind, name = self.next()
        if kind != "name":
            raise GraphQLError(f"expected field name, got {name!r}")
        # alias: name
        alias = None
        if self.peek()[1] == ":":
            self.next()
            alias, name = name, self.next()[1]
        args = {}
        if self.peek()[1] == "(":
            self.next()
            while self.peek()[1] != ")":
                _, aname = self.next()
                self.expect(":")
                args[aname] = self.value()
            self.next()
        sub = None
        if self.peek()[1] == "{":
            self.next()
            sub = self.selection_set()
        return {"name": name, "alias": alias or name, "args": args,
                "fields": sub}

    def value(self):
        kind, v = self.next()
        if kind == "string":
            return json.loads(v)
        if kind == "number":
            return float(v) if "." in v else int(v)
        if kind == "name":
            return {"true": True, "false": False, "null": None}.get(v, v)
        if v == "$":
            _, vn = self.next()
            if vn not in self.variables:
                raise GraphQLError(f"missing variable ${vn}")
            return self.variables[vn]
        if v == "[":
            out = []
            while self.peek()[1] != "]":
                out.append(self.value())
            self.next()
            return out
        if v == "{":
            out = {}
            while self.peek()[1] != "}":
                _, k = self.next()
                self.expect(":")
                out[k] = self.value()
            self.next()
            return out
        raise GraphQLError(f"bad value {v!r}")


class GraphQLEventBroker:
    """Node/relationship lifecycle events for the subscription surface
    (reference resolvers/event_broker.go). Subscribers get dict events
    via bounded queues; server/http.py streams them over SSE."""

    def __init__(self, engine):
        self._subs: List[Tuple[queue.Queue, Optional[set]]] = []
        self._lock = threading.Lock()
        engine.register_callback(self._on_event)

    def _on_event(self, ev, obj):
        from ..storage.types import Edge, Node
        if isinstance(obj, Node):
            kind = {"node_created": "nodeCreated",
                    "node_updated": "nodeUpdated",
                    "node_deleted": "nodeDeleted"}.get(ev)
            payload = {"id": obj.id, "labels": list(obj.labels),
                       "properties": dict(obj.properties)}
            labels = set(obj.labels)
        elif isinstance(obj, Edge):
            kind = {"edge_created": "relationshipCreated",
                    "edge_updated": "relationshipUpdated",
                    "edge_deleted": "relationshipDeleted"}.get(ev)
            payload = {"id": obj.id, "type": obj.type,
                       "startNode": obj.start_node, "endNode": obj.end_node}
            labels = None
        else:
            return
        if kind is None:
            return
        with self._lock:
            for q, want in self._subs:
                if want and labels is not None and not (want & labels):
                    continue
                try:
                    q.put_nowait({"event": kind, "data": payload})
                except queue.Full:
                    pass

    def subscribe(self, labels: Optional[List[str]] = None) -> queue.Queue:
        q: queue.Queue = queue.Queue(maxsize=256)
        with self._lock:
            self._subs.append((q, set(labels) if labels else None))
        return q

    def unsubscribe(self, q: queue.Queue):
        with self._lock:
            self._subs = [(x, w) for x, w in self._subs if x is not q]


class GraphQLExecutor:
    def __init__(self, db):
        self.db = db
        self.broker = GraphQLEventBroker(db.engine)

    def execute(self, query: str, variables: Dict = None) -> Dict[str, Any]:
        try:
            op, fields = _Parser(query, variables).parse()
            data = {}
            for f in fields:
                data[f["alias"]] = self._resolve(op, f)
            return {"data": data}
        except GraphQLError as e:
            return {"errors": [{"message": str(e)}]}
        except Exception as e:
            return {"errors": [{"message": f"{type(e).__name__}: {e}"}]}

    # ---------------------------------------------------------------- util
    def _props_of(self, args, key="properties"):
        props = args.get(key, {})
        if isinstance(props, str):
            props = json.loads(props) if props else {}
        return dict(props or {})

    def _node(self, n, fields):
        if n is None:
            return None
        full = {"id": n.id, "labels": list(n.labels),
                "properties": dict(n.properties)}
        if not fields:
            return full
        out = {}
        for f in fields:
            nm, al = f["name"], f["alias"]
            if nm == "relationships":
                out[al] = [self._rel(e, f["fields"]) for e in
                           self.db.engine.get_out_edges(n.id) +
                           self.db.engine.get_in_edges(n.id)]
            elif nm == "outgoing":
                out[al] = [self._rel(e, f["fields"])
                           for e in self.db.engine.get_out_edges(n.id)]
            elif nm == "incoming":
                out[al] = [self._rel(e, f["fields"])
                           for e in self.db.engine.get_in_edges(n.id)]
            elif nm == "neighbors":
                out[al] = [self._node(self.db.engine.get_node(i), f["fields"])
                           for i in self.db.engine.neighbors(n.id)]
            elif nm in full:
                out[al] = full[nm]
        return out

    def _rel(self, e, fields=None):
        if e is None:
            return None
        full = {"id": e.id, "type": e.type, "startNode": e.start_node,
                "endNode": e.end_node, "properties": dict(e.properties)}
        if not fields:
            return full
        return {f["alias"]: full.get(f["name"]) for f in fields}

    def _plain(self, v):
        from ..storage.types import Edge, Node
        if isinstance(v, Node):
            return self._node(v, None)
        if isinstance(v, Edge):
            return self._rel(v)
        return v

    # ------------------------------------------------------------ resolvers
    def _resolve(self, op, f):
        name, args, fields = f["name"], f["args"], f["fields"]
        eng = self.db.engine
        if op == "mutation":
            return self._mutate(name, args, fields)
        if op == "subscription":
            raise GraphQLError(
                "subscriptions stream via GET /graphql/stream (SSE)")

        if name == "node":
            try:
                return self._node(eng.get_node(args["id"]), fields)
            except Exception:
                return None
        if name == "nodes":
            ids = args.get("ids")
            if ids is not None:
                out = []
                for i in ids:
                    try:
                        out.append(self._node(eng.get_node(i), fields))
                    except Exception:
                        pass
                return out
            # legacy: nodes(label:, limit:) — lazy scan: stop copying
            # once `limit` nodes are taken (Engine.iter_nodes_by_label)
            label = args.get("label")
            limit = int(args.get("limit", 100))
            it = eng.iter_nodes_by_label(label) if label else eng.all_nodes()
            return [self._node(n, fields)
                    for n in itertools.islice(it, limit)]
        if name == "allNodes":
            labels = args.get("labels")
            limit = int(args.get("limit", 100))
            off = int(args.get("offset", 0))
            if labels:
                seen = {}
                for lb in labels:
                    for n in eng.get_nodes_by_label(lb):
                        seen[n.id] = n
                ns = list(seen.values())
            else:
                ns = list(eng.all_nodes())
            return [self._node(n, fields) for n in ns[off:off + limit]]
        if name == "nodesByLabel":
            limit = int(args.get("limit", 100))
            off = int(args.get("offset", 0))
            it = eng.iter_nodes_by_label(args["label"])
            return [self._node(n, fields)
                    for n in itertools.islice(it, off, off + limit)]
        if name == "nodeCount":
            label = args.get("label")
            if label and hasattr(eng, "node_count_by_label"):
                return eng.node_count_by_label(label)
            if label:
                return len(eng.get_nodes_by_label(label))
            return eng.node_count()
        if name == "relationship":
            try:
                return self._rel(eng.get_edge(args["id"]), fields)
            except Exception:
                return None
        if name in ("relationships", "allRelationships"):
            limit = int(args.get("limit", 100))
            return [self._rel(e, fields)
                    for i, e in enumerate(eng.all_edges()) if i < limit]
        if name == "relationshipCount":
            return eng.edge_count()
        if name == "relationshipsByType":
            es = eng.get_edges_by_type(args["type"])
            return [self._rel(e, fields) for e in
                    es[:int(args.get("limit", 100))]]
        if name == "relationshipsBetween":
            a, b = args["from"], args["to"]
            return [self._rel(e, fields) for e in eng.get_out_edges(a)
                    if e.end_node == b]
        if name == "neighbors":
            return [self._node(eng.get_node(i), fields)
                    for i in eng.neighbors(args["id"])]
        if name == "neighborhood":
            depth = int(args.get("depth", 1))
            start = args.get("nodeId", args.get("id"))
            seen = {start}
            frontier = {start}
            for _ in range(depth):
                nxt = set()
                for nid in frontier:
                    for other in eng.neighbors(nid):
                        if other not in seen:
                            seen.add(other)
                            nxt.add(other)
                frontier = nxt
            nodes = [eng.get_node(i) for i in seen]
            return {"nodes": [self._node(n, None) for n in nodes],
                    "relationships": []}
        if name == "shortestPath":
            a = args.get("startNodeId", args.get("from"))
            b = args.get("endNodeId", args.get("to"))
            depth = int(args.get("maxDepth", 10))
            r = self.db.cypher(
                "MATCH (a), (b) WHERE id(a) = $a AND id(b) = $b "
                f"MATCH p = shortestPath((a)-[*1..{depth}]->(b)) RETURN p",
                {"a": a, "b": b})
            if not r.rows:
                return None
            p = r.rows[0][0]
            return {"nodes": [self._node(n, None) for n in p.nodes],
                    "relationships": [self._rel(e) for e in p.edges]}
        if name == "allPaths":
            # schema.graphql allPaths: [[Node!]!]! — lists of node paths
            a = args.get("startNodeId", args.get("from"))
            b = args.get("endNodeId", args.get("to"))
            depth = int(args.get("maxDepth", 5))
            limit = int(args.get("limit", 10))
            r = self.db.cypher(
                "MATCH (a), (b) WHERE id(a) = $a AND id(b) = $b "
                f"MATCH p = (a)-[*1..{depth}]->(b) RETURN p LIMIT $lim",
                {"a": a, "b": b, "lim": limit})
            return [[self._node(n, None) for n in row[0].nodes]
                    for row in r.rows]
        if name == "search":
            qtext = args.get("query") or args.get("text", "")
            k = int(args.get("limit", args.get("k", 10)))
            res = self.db.search.search(query=qtext, k=k)
            return [{"id": r.id, "score": r.score,
                     "node": self._node(r.node, None)} for r in res]
        if name == "similar":
            res = self.db.search.similar_to(args["id"],
                                            k=int(args.get("limit", 10)))
            return [{"id": r.id, "score": r.score,
                     "node": self._node(r.node, None)} for r in res]
        if name == "searchByProperty":
            label = args.get("label")
            prop, val = args["property"], args.get("value")
            ns = (eng.get_nodes_by_label(label) if label
                  else list(eng.all_nodes()))
            return [self._node(n, fields) for n in ns
                    if str(n.properties.get(prop)) == str(val)]
        if name in ("cypher", "executeCypher"):
            r = self.db.cypher(args["query"], args.get("parameters"))
            return {"columns": r.columns,
                    "rows": json.loads(json.dumps(
                        [[self._plain(v) for v in row] for row in r.rows],
                        default=str)),
                    "stats": r.stats}
        raise GraphQLError(f"unknown field {name}")

    def _mutate(self, name, args, fields):
        from ..storage import Edge, Node, new_id
        eng = self.db.engine
        inp = args.get("input", args)
        if name == "createNode":
            n = eng.create_node(Node(
                id=str(inp.get("id") or new_id("n")),
                labels=list(inp.get("labels", [])),
                properties=self._props_of(inp)))
            if self.db.auto_embed:
                self.db.engine.mark_pending_embedding(n.id)
            return self._node(n, fields)
        if name == "updateNode":
            n = eng.get_node(inp["id"])
            if "labels" in inp and inp["labels"] is not None:
                n.labels = list(inp["labels"])
            n.properties.update(self._props_of(inp))
            return self._node(eng.update_node(n), fields)
        if name == "deleteNode":
            eng.detach_delete_node(args.get("id") or inp.get("id"))
            return True
        if name == "mergeNode":
            label = (inp.get("labels") or ["Node"])[0]
            props = self._props_of(inp)
            key = inp.get("mergeKey") or (next(iter(props)) if props else None)
            if key is not None:
                for n in eng.get_nodes_by_label(label):
                    if n.properties.get(key) == props.get(key):
                        n.properties.update(props)
                        return self._node(eng.update_node(n), fields)
            n = eng.create_node(Node(id=new_id("n"),
                                     labels=list(inp.get("labels", [])),
                                     properties=props))
            return self._node(n, fields)
        if name == "createRelationship":
            e = eng.create_edge(Edge(
                id=new_id("e"), type=inp.get("type", "RELATED"),
                start_node=str(inp.get("from") or inp.get("startNode")),
                end_node=str(inp.get("to") or inp.get("endNode")),
                properties=self._props_of(inp)))
            return self._rel(e, fields)
        if name == "updateRelationship":
            e = eng.get_edge(inp["id"])
            e.properties.update(self._props_of(inp))
            return self._rel(eng.update_edge(e), fields)
        if name == "deleteRelationship":
            eng.delete_edge(args.get("id") or inp.get("id"))
            return True
        if name == "mergeRelationship":
            s = str(inp.get("from") or inp.get("startNode"))
            t = str(inp.get("to") or inp.get("endNode"))
            typ = inp.get("type", "RELATED")
            for e in eng.get_out_edges(s):
                if e.end_node == t and e.type == typ:
                    return self._rel(e, fields)
            e = eng.create_edge(Edge(id=new_id("e"), type=typ, start_node=s,
                                     end_node=t,
                                     properties=self._props_of(inp)))
            return self._rel(e, fields)
        if name == "bulkCreateNodes":
            made = []
            for nd in inp.get("nodes", []):
                made.append(eng.create_node(Node(
                    id=str(nd.get("id") or new_id("n")),
                    labels=list(nd.get("labels", [])),
                    properties=self._props_of(nd))))
            return {"count": len(made),
                    "nodes": [self._node(n, None) for n in made]}
        if name == "bulkCreateRelationships":
            made = []
            for rd in inp.get("relationships", []):
                made.append(eng.create_edge(Edge(
                    id=new_id("e"), type=rd.get("type", "RELATED"),
                    start_node=str(rd.get("from") or rd.get("startNode")),
                    end_node=str(rd.get("to") or rd.get("endNode")),
                    properties=self._props_of(rd))))
            return {"count": len(made)}
        if name == "bulkDeleteNodes":
            n = 0
            for i in inp.get("ids", []):
                try:
                    eng.detach_delete_node(i)
                    n += 1
                except Exception:
                    pass
            return {"count": n}
        if name == "bulkDeleteRelationships":
            n = 0
            for i in inp.get("ids", []):
                try:
                    eng.delete_edge(i)
                    n += 1
                except Exception:
                    pass
            return {"count": n}
        if name == "clearAll":
            for n in list(eng.all_nodes()):
                try:
                    eng.detach_delete_node(n.id)
                except Exception:
                    pass
            return True
        if name == "triggerEmbedding":
            nid = args.get("id") or inp.get("id")
            self.db.engine.mark_pending_embedding(nid)
            return True
        raise GraphQLError(f"unknown mutation {name}")
