"""Minimal GraphQL endpoint over the graph.

Parity: reference pkg/graphql (gqlgen-generated schema over nodes /
relationships / search). This is a hand-written executor for the core
query surface (no codegen):

    { nodes(label: "Person", limit: 10) { id labels properties } }
    { node(id: "abc") { id properties relationships { type endNode } } }
    { search(query: "text", limit: 5) { id score } }
    mutation { createNode(labels: ["X"], properties: "{\"k\":1}") { id } }
"""

from __future__ import annotations

import json
import re
from typing import Any, Dict, List, Optional, Tuple


class GraphQLError(Exception):
    pass


_TOKEN = re.compile(r"""
    (?P<ws>[\s,]+)
  | (?P<name>[_A-Za-z][_0-9A-Za-z]*)
  | (?P<string>"(?:\\.|[^"\\])*")
  | (?P<number>-?\d+(?:\.\d+)?)
  | (?P<punct>[{}()\[\]:!@$=])
""", re.VERBOSE)


def _tokenize(src: str):
    out = []
    i = 0
    while i < len(src):
        m = _TOKEN.match(src, i)
        if not m:
            raise GraphQLError(f"bad character at {i}: {src[i]!r}")
        i = m.end()
        if m.lastgroup != "ws":
            out.append((m.lastgroup, m.group()))
    out.append(("eof", ""))
    return out


class _Parser:
    def __init__(self, src):
        self.toks = _tokenize(src)
        self.i = 0

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
        if t[0] == "name" and t[1] in ("query", "mutation"):
            op = self.next()[1]
            if self.peek()[0] == "name":
                self.next()  # operation name
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
        return {"name": name, "args": args, "fields": sub}

    def value(self):
        kind, v = self.next()
        if kind == "string":
            return json.loads(v)
        if kind == "number":
            return float(v) if "." in v else int(v)
        if kind == "name":
            return {"true": True, "false": False, "null": None}.get(v, v)
        if v == "[":
            out = []
            while self.peek()[1] != "]":
                out.append(self.value())
            self.next()
            return out
        raise GraphQLError(f"bad value {v!r}")


class GraphQLExecutor:
    def __init__(self, db):
        self.db = db

    def execute(self, query: str, variables: Dict = None) -> Dict[str, Any]:
        try:
            op, fields = _Parser(query).parse()
            data = {}
            for f in fields:
                data[f["name"]] = self._resolve(op, f)
            return {"data": data}
        except GraphQLError as e:
            return {"errors": [{"message": str(e)}]}
        except Exception as e:
            return {"errors": [{"message": f"{type(e).__name__}: {e}"}]}

    # ---- resolvers ----
    def _resolve(self, op, f):
        name, args = f["name"], f["args"]
        if op == "mutation":
            if name == "createNode":
                from ..storage import Node, new_id
                props = args.get("properties", {})
                if isinstance(props, str):
                    props = json.loads(props)
                n = self.db.engine.create_node(Node(
                    id=new_id("n"), labels=list(args.get("labels", [])),
                    properties=props))
                return self._node(n, f["fields"])
            if name == "createRelationship":
                from ..storage import Edge, new_id
                e = self.db.engine.create_edge(Edge(
                    id=new_id("e"), type=args.get("type", "RELATED"),
                    start_node=args["from"], end_node=args["to"],
                    properties={}))
                return {"id": e.id, "type": e.type}
            if name == "deleteNode":
                self.db.engine.detach_delete_node(args["id"])
                return True
            raise GraphQLError(f"unknown mutation {name}")
        if name == "nodes":
            label = args.get("label")
            limit = int(args.get("limit", 25))
            nodes = (self.db.engine.get_nodes_by_label(label) if label
                     else list(self.db.engine.all_nodes()))
            return [self._node(n, f["fields"]) for n in nodes[:limit]]
        if name == "node":
            n = self.db.engine.get_node(args["id"])
            return self._node(n, f["fields"])
        if name == "search":
            qv = self.db.embedder.embed_query(args["query"])
            res = self.db.search.search(query=args["query"], query_vec=qv,
                                        k=int(args.get("limit", 10)))
            return [{"id": r.id, "score": r.score,
                     "node": self._node(r.node, None)} for r in res]
        if name == "cypher":
            r = self.db.cypher(args["query"])
            return {"columns": r.columns,
                    "rows": json.loads(json.dumps(
                        [[self._plain(v) for v in row] for row in r.rows],
                        default=str))}
        raise GraphQLError(f"unknown field {name}")

    def _node(self, n, fields):
        full = {"id": n.id, "labels": n.labels, "properties": n.properties}
        if fields:
            out = {}
            for f in fields:
                if f["name"] == "relationships":
                    out["relationships"] = [
                        {"id": e.id, "type": e.type, "startNode": e.start_node,
                         "endNode": e.end_node}
                        for e in self.db.engine.get_out_edges(n.id)]
                elif f["name"] in full:
                    out[f["name"]] = full[f["name"]]
            return out
        return full

    def _plain(self, v):
        from ..storage.types import Edge, Node
        if isinstance(v, Node):
            return {"id": v.id, "labels": v.labels, "properties": v.properties}
        if isinstance(v, Edge):
            return {"id": v.id, "type": v.type}
        return v
