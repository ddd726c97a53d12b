"""Cypher executor: clause pipeline over binding rows.

Parity target: reference pkg/cypher/executor.go:490 (Execute) and the
clause executors in match.go/create.go/merge.go/match_with.go/
executor_mutations.go; pattern scans use the storage label index
(reference pkg/storage/label_index_lookup.go) and exact property indexes.

Rows are dicts {var: value}; values are python scalars/lists/maps,
storage.Node / storage.Edge copies, or Path objects.
"""

from __future__ import annotations

import functools
import re
from dataclasses import dataclass, field
from typing import Any, Dict, Iterable, List, Optional, Tuple

from ..storage.types import Edge, Engine, Node, NotFoundError, new_id
from . import ast as A
from .functions import (AGGREGATES, Aggregator, CypherRuntimeError, FUNCTIONS,
                        is_aggregate)
from .lexer import CypherSyntaxError
from .parser import parse


class Path:
    def __init__(self, nodes: List[Node], edges: List[Edge]):
        self.nodes = nodes
        self.edges = edges

    def __len__(self):
        return len(self.edges)

    def __eq__(self, o):
        return isinstance(o, Path) and [n.id for n in self.nodes] == [n.id for n in o.nodes] \
            and [e.id for e in self.edges] == [e.id for e in o.edges]

    def __repr__(self):
        return f"Path({len(self.edges)} rels)"


@dataclass
class Result:
    columns: List[str]
    rows: List[List[Any]]
    stats: Dict[str, int] = field(default_factory=dict)
    profile: Optional[List[Dict[str, Any]]] = None

    def to_dicts(self):
        return [dict(zip(self.columns, r)) for r in self.rows]


def _hkey(v):
    """Hashable identity key for grouping/DISTINCT."""
    if isinstance(v, Node):
        return ("__node__", v.id)
    if isinstance(v, Edge):
        return ("__edge__", v.id)
    if isinstance(v, Path):
        return ("__path__", tuple(n.id for n in v.nodes), tuple(e.id for e in v.edges))
    if isinstance(v, list):
        return ("__list__",) + tuple(_hkey(x) for x in v)
    if isinstance(v, dict):
        return ("__map__",) + tuple(sorted((k, _hkey(x)) for k, x in v.items()))
    return v


_MISSING = object()


class Executor:
    """StorageExecutor equivalent (reference pkg/cypher/executor.go:187)."""

    def __init__(self, engine: Engine, procedures: Dict[str, Any] = None,
                 query_cache=None, schema=None):
        self.engine = engine
        self.procedures = dict(procedures or {})
        self._plan_cache: Dict[str, A.Query] = {}
        self.stats: Dict[str, int] = {}
        self.schema = schema            # storage.SchemaManager (DDL target)
        self._profile_log = None
        self.database_lister = None     # set by DatabaseManager for SHOW DATABASES
        self.database_router = None     # set by DatabaseManager for USE <db>
        self.database_admin = None      # set by DatabaseManager for
                                        # CREATE/DROP DATABASE|ALIAS
        self.current_database = "neo4j"

    # ------------------------------------------------------------------ API
    def execute(self, cypher: str, params: Dict[str, Any] = None,
                bindings: Dict[str, Any] = None) -> Result:
        """bindings: pre-bound row variables (used by apoc.periodic.iterate
        to pass outer-query columns into the inner query)."""
        params = params or {}
        self._initial_bindings = bindings
        q = self._plan_cache.get(cypher)
        if q is None:
            q = parse(cypher)
            if len(self._plan_cache) > 1024:
                self._plan_cache.clear()
            self._plan_cache[cypher] = q
        self.stats = {"nodes_created": 0, "nodes_deleted": 0,
                      "edges_created": 0, "edges_deleted": 0,
                      "properties_set": 0, "labels_added": 0}
        if q.explain:
            return Result(["plan"], [[self._explain(q)]], dict(self.stats))
        if q.profile:
            return self._profile(q, params)
        if q.clauses and isinstance(q.clauses[0], A.UseClause):
            use = q.clauses[0]
            router = getattr(self, "database_router", None)
            if router is None:
                raise CypherRuntimeError(
                    f"USE {use.database}: no database router attached "
                    "(open via DatabaseManager)")
            rest = A.Query(q.clauses[1:], union=q.union)
            if not rest.clauses:
                return Result([], [])
            return router(use.database, rest, params)
        if len(q.clauses) == 1 and isinstance(q.clauses[0], A.SchemaCommand):
            res = self._exec_schema(q.clauses[0], params)
            res.stats = dict(self.stats)
            return res
        fast = self._try_fast_count(q, params)
        if fast is not None:
            fast.stats = dict(self.stats)
            return fast
        res = self._run_query(q, params)
        self._initial_bindings = None
        res.stats = dict(self.stats)
        return res

    @property
    def _colstore(self):
        cs = getattr(self, "_colstore_inst", None)
        if cs is None:
            from .columnar import ColumnStore
            cs = ColumnStore(self.engine)
            self._colstore_inst = cs
        return cs

    def _try_fast_count(self, q: A.Query, params):
        """Fast path for `MATCH (n:L [{props}]) [WHERE ...] RETURN count(..)`
        over the raw label index — no node copies, no row pipeline
        (reference executor.go:994 tryFastPathCompoundQuery +
        storage_fastpaths.go)."""
        if q.union or len(q.clauses) != 2:
            return None
        m, r = q.clauses
        if not (isinstance(m, A.MatchClause) and not m.optional
                and len(m.patterns) == 1):
            return None
        pat = m.patterns[0]
        if pat.var or getattr(pat, "shortest", None) or len(pat.elements) != 1:
            return None
        np_ = pat.elements[0]
        if getattr(np_, "or_labels", False) or getattr(np_, "where", None):
            return None
        if not (isinstance(r, A.ReturnClause) and not r.distinct
                and not r.order_by and r.skip is None and r.limit is None
                and not r.star and len(r.items) == 1):
            return None
        it = r.items[0].expr
        if not (isinstance(it, A.FuncCall) and it.name == "count"
                and not it.distinct):
            return None
        if not it.star:
            if not (len(it.args) == 1 and isinstance(it.args[0], A.Var)
                    and np_.var and it.args[0].name == np_.var):
                return None
        raw_iter = getattr(self.engine, "iter_nodes_raw", None)
        if raw_iter is None:
            return None
        name = r.items[0].alias or self._expr_name(it)

        props = {}
        if np_.props is not None:
            try:
                props = dict(self._eval(np_.props, {}, params) or {})
            except CypherRuntimeError:
                return None
        label = np_.labels[0] if np_.labels else None
        extra_labels = np_.labels[1:]

        # O(1): bare single-label count
        if (label and not extra_labels and not props and m.where is None
                and hasattr(self.engine, "node_count_by_label")):
            return Result([name], [[self.engine.node_count_by_label(label)]])
        if (label is None and not props and m.where is None):
            return Result([name], [[self.engine.node_count()]])

        # indexed equality WHERE: var.prop = <value> -> exact index count
        if (m.where is not None and isinstance(m.where, A.BinOp)
                and m.where.op == "=" and label and not extra_labels
                and not props and np_.var):
            lhs, rhs = m.where.left, m.where.right
            if (isinstance(lhs, A.Prop) and isinstance(lhs.expr, A.Var)
                    and lhs.expr.name == np_.var
                    and isinstance(rhs, (A.Lit, A.Param))):
                lookup = getattr(self.engine, "lookup_property_index", None)
                if lookup is not None:
                    try:
                        val = self._eval(rhs, {}, params)
                        hits = lookup(label, lhs.key, val)
                    except CypherRuntimeError:
                        hits = None
                    if hits is not None:
                        return Result([name], [[len(hits)]])

        # vectorized count over cached numpy columns (columnar.py)
        if (m.where is not None and label and not extra_labels and not props
                and np_.var):
            from .columnar import columnar_filter
            got = columnar_filter(self._colstore, label, np_.var, m.where,
                                  params)
            if got is not None:
                return Result([name], [[int(got[1].sum())]])

        npred = None
        if m.where is not None:
            from .compiler import compile_node_predicate
            npred = compile_node_predicate(m.where, np_.var) if np_.var else None
        count = 0
        try:
            if (m.where is not None and npred is not None
                    and not extra_labels and not props):
                # tightest loop: raw nodes -> compiled predicate
                for n in raw_iter(label):
                    if npred(n, params) is True:
                        count += 1
            else:
                for n in raw_iter(label):
                    if extra_labels and not all(lb in n.labels
                                                for lb in extra_labels):
                        continue
                    if props and not self._props_match(n, props):
                        continue
                    if m.where is not None:
                        if npred is not None:
                            if npred(n, params) is not True:
                                continue
                        else:
                            row = {np_.var: n} if np_.var else {}
                            if self._eval(m.where, row, params) is not True:
                                continue
                    count += 1
        except (CypherRuntimeError, KeyError, TypeError):
            return None  # exotic WHERE -> full pipeline
        return Result([name], [[count]])

    def _explain(self, q: A.Query) -> str:
        lines = []
        for c in q.clauses:
            lines.append(type(c).__name__.replace("Clause", ""))
        return " -> ".join(lines)

    # ------------------------------------------------------------- pipeline
    def _profile(self, q: A.Query, params) -> Result:
        """PROFILE: execute and attach per-clause wall time + row counts
        (clause-granular: the executor pipeline is clause-at-a-time)."""
        self._profile_log = []
        try:
            res = self._run_query(q, params)
        finally:
            plan, self._profile_log = self._profile_log, None
        res.stats = dict(self.stats)
        res.profile = plan
        return res

    def _run_query(self, q: A.Query, params) -> Result:
        cols, rows = self._run_clauses(q.clauses, params)
        if q.union:
            mode, rest = q.union
            r2 = self._run_query(rest, params)
            if r2.columns != cols and cols and r2.columns:
                if len(r2.columns) != len(cols):
                    raise CypherRuntimeError("UNION column count mismatch")
            rows = rows + r2.rows
            if mode == "UNION":
                seen = set()
                uniq = []
                for r in rows:
                    k = tuple(_hkey(v) for v in r)
                    if k not in seen:
                        seen.add(k)
                        uniq.append(r)
                rows = uniq
        return Result(cols, rows)

    def _run_clauses(self, clauses, params) -> Tuple[List[str], List[List[Any]]]:
        init = getattr(self, "_initial_bindings", None)
        rows: List[Dict[str, Any]] = [dict(init) if init else {}]
        out_cols: List[str] = []
        out_rows: List[List[Any]] = []
        i = 0
        n = len(clauses)
        prof = getattr(self, "_profile_log", None)
        while i < n:
            if prof is not None:
                import time as _t
                _pt0 = _t.perf_counter()
            c = clauses[i]
            if isinstance(c, A.MatchClause):
                rows = self._exec_match(c, rows, params)
            elif isinstance(c, A.UnwindClause):
                rows = self._exec_unwind(c, rows, params)
            elif isinstance(c, A.CreateClause):
                rows = self._exec_create(c, rows, params)
            elif isinstance(c, A.MergeClause):
                rows = self._exec_merge(c, rows, params)
            elif isinstance(c, A.SetClause):
                rows = self._exec_set(c.items, rows, params)
            elif isinstance(c, A.RemoveClause):
                rows = self._exec_remove(c, rows, params)
            elif isinstance(c, A.DeleteClause):
                rows = self._exec_delete(c, rows, params)
            elif isinstance(c, A.WithClause):
                rows = self._exec_with(c, rows, params)
            elif isinstance(c, A.ReturnClause):
                out_cols, out_rows = self._exec_return(c, rows, params)
            elif isinstance(c, A.CallClause):
                last = i == n - 1
                rows, call_out = self._exec_call(c, rows, params, standalone=last)
                if call_out is not None:
                    out_cols, out_rows = call_out
            elif isinstance(c, A.SubqueryCallClause):
                rows = self._exec_subquery_call(c, rows, params)
            elif isinstance(c, A.ForeachClause):
                rows = self._exec_foreach(c, rows, params)
            else:
                raise CypherRuntimeError(f"unsupported clause {type(c).__name__}")
            if prof is not None:
                prof.append({
                    "operator": type(c).__name__,
                    "rows": len(out_rows) if isinstance(c, A.ReturnClause)
                            and not isinstance(c, A.WithClause) else len(rows),
                    "time_ms": round((_t.perf_counter() - _pt0) * 1e3, 4)})
            i += 1
        return out_cols, out_rows

    def _exec_schema(self, c: A.SchemaCommand, params) -> Result:
        """Schema DDL (Neo4j 4/5 + 3.x legacy syntax). Requires a
        SchemaManager (wired by the NornicDB facade); raw Executor use
        without one raises."""
        from ..storage.schema import SchemaManager

        if c.op in ("create", "drop") and c.kind in ("database", "alias",
                                                     "composite"):
            if self.database_admin is None:
                raise CypherRuntimeError(
                    f"{c.op.upper()} {c.kind.upper()} needs a DatabaseManager")
            return self.database_admin(c)
        if c.op == "show":
            if c.kind == "aliases":
                if self.database_admin is None:
                    return Result(["name", "database"], [])
                return self.database_admin(c)
            if c.kind == "databases":
                names = (self.database_lister() if self.database_lister
                         else [self.current_database])
                return Result(
                    ["name", "type", "access", "role", "currentStatus",
                     "default", "home"],
                    [[n, "standard", "read-write", "primary", "online",
                      n == "neo4j", n == "neo4j"] for n in names])
            if c.kind == "settings":
                proc = self.procedures.get("dbms.listconfig")
                if proc:
                    cols, rows = proc(self)
                    return Result(cols, rows)
                return Result(["name", "value"], [])
            if c.kind == "transactions":
                # single current transaction (our executor is synchronous)
                return Result(
                    ["database", "transactionId", "currentQuery", "status"],
                    [[self.current_database, "tx-0", "", "Running"]])
            if c.kind == "procedures":
                return Result(["name", "description", "mode"],
                              [[n, (getattr(f, "__doc__", "") or "").strip()
                                .split("\n")[0], "DEFAULT"]
                               for n, f in sorted(self.procedures.items())])
            if c.kind == "functions":
                from .functions import FUNCTIONS
                return Result(["name", "category", "description"],
                              [[n, "builtin", ""] for n in sorted(FUNCTIONS)])
            if self.schema is None:
                return Result(["name"], [])
            if c.kind == "indexes":
                rows = []
                for i, (name, kind, label, props) in enumerate(
                        self.schema.list_indexes()):
                    if c.type_filter and kind.upper() != c.type_filter:
                        continue
                    rows.append([i + 1, name, "ONLINE", 100.0, kind.upper(),
                                 "NODE", [label], props])
                return Result(["id", "name", "state", "populationPercent",
                               "type", "entityType", "labelsOrTypes",
                               "properties"], rows)
            if c.kind == "constraints":
                kindname = {"unique": "UNIQUENESS",
                            "exists": "NODE_PROPERTY_EXISTENCE"}
                return Result(
                    ["id", "name", "type", "entityType", "labelsOrTypes",
                     "properties"],
                    [[i + 1, cc.name, kindname.get(cc.kind, cc.kind.upper()),
                      "NODE", [cc.label], [cc.prop]]
                     for i, cc in enumerate(self.schema.list_constraints())])
            raise CypherRuntimeError(f"cannot SHOW {c.kind}")

        if self.schema is None:
            raise CypherRuntimeError(
                "schema commands need a SchemaManager (open via NornicDB)")

        if c.op == "drop":
            if c.kind == "index":
                if c.name is None:  # legacy DROP INDEX ON :L(p)
                    name = f"index_{c.label}_{'_'.join(c.props)}"
                else:
                    name = c.name
                ok = self.schema.drop_index(name)
                if not ok and not c.if_exists:
                    raise CypherRuntimeError(f"no such index {name!r}")
            else:
                if c.name not in {x.name for x in
                                  self.schema.list_constraints()} \
                        and not c.if_exists:
                    raise CypherRuntimeError(f"no such constraint {c.name!r}")
                self.schema.drop_constraint(c.name)
            return Result([], [])

        # create
        if c.kind == "constraint":
            name = c.name or f"constraint_{c.label}_{'_'.join(c.props)}"
            existing = {x.name for x in self.schema.list_constraints()}
            if name in existing:
                if c.if_not_exists:
                    return Result([], [])
                if c.or_replace:
                    self.schema.drop_constraint(name)
                else:
                    raise CypherRuntimeError(
                        f"constraint {name!r} already exists")
            for p in c.props:
                if c.constraint_kind in ("unique", "node_key"):
                    self.schema.create_unique_constraint(name, c.label, p)
                else:
                    self.schema.create_exists_constraint(name, c.label, p)
                if c.constraint_kind == "node_key":
                    self.schema.create_exists_constraint(
                        name + "_exists", c.label, p)
            return Result([], [])

        name = c.name or f"index_{c.label}_{'_'.join(c.props)}"
        existing = {x[0] for x in self.schema.list_indexes()}
        if name in existing:
            if c.if_not_exists:
                return Result([], [])
            if not c.or_replace:
                raise CypherRuntimeError(f"index {name!r} already exists")
            self.schema.drop_index(name)
        if c.kind == "vector":
            dims, sim = 0, "cosine"
            if c.options is not None:
                try:
                    opts = self._eval(c.options, {}, params) or {}
                    icfg = opts.get("indexConfig", opts)
                    for k, v in icfg.items():
                        lk = k.lower()
                        if "dimension" in lk or lk == "dims":
                            dims = int(v)
                        if "similarity" in lk:
                            sim = str(v)
                except CypherRuntimeError:
                    pass
            self.schema.create_vector_index(name, c.label, c.props[0],
                                            dims, sim)
        else:
            self.schema.create_index(c.label, c.props[0], name=name,
                                     kind=c.kind, props=c.props)
        return Result([], [])

    def _run_subquery(self, q, bindings, params):
        """Execute a subquery AST with the outer row's bindings visible
        (superset of Cypher's WITH-import rule). Returns (cols, rows)."""
        save = getattr(self, "_initial_bindings", None)
        self._initial_bindings = bindings
        try:
            cols, rows = self._run_clauses(q.clauses, params)
            node = q
            while node.union is not None:
                kind, nxt = node.union
                c2, r2 = self._run_clauses(nxt.clauses, params)
                rows = rows + r2
                if kind == "UNION":
                    seen, uniq = set(), []
                    for r in rows:
                        k = tuple(_hkey(v) for v in r)
                        if k not in seen:
                            seen.add(k)
                            uniq.append(r)
                    rows = uniq
                node = nxt
        finally:
            self._initial_bindings = save
        return cols, rows

    def _exec_subquery_call(self, c, rows, params):
        """CALL { ... }: run the inner query once per incoming row; returned
        columns are appended to the row (cartesian with inner results).
        A unit subquery (no RETURN) passes rows through unchanged.
        IN TRANSACTIONS executes eagerly (single-process engine; batching
        only bounds memory, which list processing already does)."""
        out = []
        for row in rows:
            cols, subrows = self._run_subquery(c.query, dict(row), params)
            if cols:
                for sr in subrows:
                    nr = dict(row)
                    nr.update(zip(cols, sr))
                    out.append(nr)
            else:
                out.append(row)
        return out

    # -------------------------------------------------------------- helpers
    def _eval(self, e, row, params):
        if isinstance(e, A.Lit):
            return e.value
        if isinstance(e, A.Param):
            if e.name not in params:
                raise CypherRuntimeError(f"missing parameter ${e.name}")
            return params[e.name]
        if isinstance(e, A.Var):
            v = row.get(e.name, _MISSING)
            if v is _MISSING:
                raise CypherRuntimeError(f"variable `{e.name}` not defined")
            return v
        if isinstance(e, A.Prop):
            base = self._eval(e.expr, row, params)
            if base is None:
                return None
            if isinstance(base, (Node, Edge)):
                return base.properties.get(e.key)
            if isinstance(base, dict):
                return base.get(e.key)
            if hasattr(base, "component"):  # temporal values
                try:
                    return base.component(e.key)
                except KeyError:
                    return None
            raise CypherRuntimeError(f"cannot access .{e.key} on {type(base).__name__}")
        if isinstance(e, A.BinOp):
            return self._eval_binop(e, row, params)
        if isinstance(e, A.UnOp):
            return self._eval_unop(e, row, params)
        if isinstance(e, A.FuncCall):
            return self._eval_func(e, row, params)
        if isinstance(e, A.ListLit):
            return [self._eval(x, row, params) for x in e.items]
        if isinstance(e, A.MapLit):
            return {k: self._eval(v, row, params) for k, v in e.items}
        if isinstance(e, A.Index):
            base = self._eval(e.expr, row, params)
            if base is None:
                return None
            if e.slice is not None:
                lo, hi = e.slice
                lo_v = self._eval(lo, row, params) if lo is not None else None
                hi_v = self._eval(hi, row, params) if hi is not None else None
                return base[lo_v:hi_v]
            idx = self._eval(e.index, row, params)
            if isinstance(base, dict):
                return base.get(idx)
            if isinstance(base, (Node, Edge)):
                return base.properties.get(idx)
            try:
                return base[idx]
            except (IndexError, TypeError):
                return None
        if isinstance(e, A.Case):
            if e.test is not None:
                # Simple form: the operand is evaluated EXACTLY ONCE and
                # bound to the synthetic __case__ variable the parser's
                # desugared WHEN conditions reference (Neo4j semantics for
                # non-deterministic operands like rand()).
                row = dict(row)
                row["__case__"] = self._eval(e.test, row, params)
            for w, r in e.whens:
                if self._eval(w, row, params) is True:
                    return self._eval(r, row, params)
            return self._eval(e.default, row, params) if e.default else None
        if isinstance(e, A.ListComp):
            src = self._eval(e.source, row, params) or []
            out = []
            for item in src:
                r2 = dict(row)
                r2[e.var] = item
                if e.where is not None and self._eval(e.where, r2, params) is not True:
                    continue
                out.append(self._eval(e.projection, r2, params)
                           if e.projection is not None else item)
            return out
        if isinstance(e, A.Quantifier):
            src = self._eval(e.source, row, params) or []
            hits = 0
            for item in src:
                r2 = dict(row)
                r2[e.var] = item
                if self._eval(e.where, r2, params) is True:
                    hits += 1
            if e.kind == "ANY":
                return hits > 0
            if e.kind == "ALL":
                return hits == len(src)
            if e.kind == "NONE":
                return hits == 0
            return hits == 1  # SINGLE
        if isinstance(e, A.PatternPredicate):
            for _ in self._match_path(e.pattern, dict(row), params, limit=1):
                return True
            return False
        if isinstance(e, A.MapProjection):
            base = self._eval(e.expr, row, params)
            if base is None:
                return None
            if hasattr(base, "component"):  # temporal value
                props = {}
                out = {}
                for it in e.items:
                    if it[0] == "prop":
                        try:
                            out[it[1]] = base.component(it[1])
                        except KeyError:
                            out[it[1]] = None
                    elif it[0] == "kv":
                        out[it[1]] = self._eval(it[2], row, params)
                    elif it[0] == "var":
                        out[it[1]] = row.get(it[1])
                return out
            props = (dict(base.properties) if isinstance(base, (Node, Edge))
                     else dict(base or {}))
            out = {}
            for it in e.items:
                if it[0] == "all":
                    out.update(props)
                elif it[0] == "prop":
                    out[it[1]] = props.get(it[1])
                elif it[0] == "kv":
                    out[it[1]] = self._eval(it[2], row, params)
                else:  # var
                    out[it[1]] = row.get(it[1])
            return out
        if isinstance(e, A.PatternComprehension):
            out = []
            for bound in self._match_path(e.pattern, dict(row), params):
                if e.where is not None and \
                        self._eval(e.where, bound, params) is not True:
                    continue
                out.append(self._eval(e.proj, bound, params))
            return out
        if isinstance(e, A.Reduce):
            acc = self._eval(e.init, row, params)
            src_l = self._eval(e.source, row, params)
            for v in (src_l or []):
                r2 = dict(row)
                r2[e.acc] = acc
                r2[e.var] = v
                acc = self._eval(e.expr, r2, params)
            return acc
        if isinstance(e, A.TypePredicate):
            v = self._eval(e.expr, row, params)
            t = e.type_name
            ok = {
                "INTEGER": lambda x: isinstance(x, int) and not isinstance(x, bool),
                "INT": lambda x: isinstance(x, int) and not isinstance(x, bool),
                "FLOAT": lambda x: isinstance(x, float),
                "STRING": lambda x: isinstance(x, str),
                "BOOLEAN": lambda x: isinstance(x, bool),
                "BOOL": lambda x: isinstance(x, bool),
                "LIST": lambda x: isinstance(x, list),
                "MAP": lambda x: isinstance(x, dict),
                "NODE": lambda x: isinstance(x, Node),
                "RELATIONSHIP": lambda x: isinstance(x, Edge),
                "PATH": lambda x: isinstance(x, Path),
                "NULL": lambda x: x is None,
                "NUMBER": lambda x: isinstance(x, (int, float))
                          and not isinstance(x, bool),
            }.get(t, lambda x: False)(v)
            if v is None and t != "NULL":
                ok = False
            return (not ok) if e.negated else ok
        if isinstance(e, A.SubqueryExpr):
            cols, rows2 = self._run_subquery(e.query, dict(row), params)
            if e.kind == "EXISTS":
                return len(rows2) > 0
            if e.kind == "COUNT":
                return len(rows2)
            return [r[0] for r in rows2]  # COLLECT
        raise CypherRuntimeError(f"cannot evaluate {type(e).__name__}")

    def _eval_binop(self, e, row, params):
        op = e.op
        if op in ("AND", "OR", "XOR"):
            l = self._as_bool(self._eval(e.left, row, params))
            if op == "AND":
                if l is False:
                    return False
                r = self._as_bool(self._eval(e.right, row, params))
                if r is False:
                    return False
                return None if (l is None or r is None) else True
            if op == "OR":
                if l is True:
                    return True
                r = self._as_bool(self._eval(e.right, row, params))
                if r is True:
                    return True
                return None if (l is None or r is None) else False
            r = self._as_bool(self._eval(e.right, row, params))
            if l is None or r is None:
                return None
            return l != r
        l = self._eval(e.left, row, params)
        r = self._eval(e.right, row, params)
        if op == "=":
            if l is None or r is None:
                return None
            return self._cy_eq(l, r)
        if op in ("<>", "!="):
            # != accepted as <> (reference executor_mutations.go:995)
            if l is None or r is None:
                return None
            return not self._cy_eq(l, r)
        if op in ("<", ">", "<=", ">="):
            if l is None or r is None:
                return None
            try:
                if op == "<":
                    return l < r
                if op == ">":
                    return l > r
                if op == "<=":
                    return l <= r
                return l >= r
            except TypeError:
                return None
        if op == "+":
            if l is None or r is None:
                return None
            if isinstance(l, list):
                return l + (r if isinstance(r, list) else [r])
            if isinstance(r, list):
                return [l] + r
            if isinstance(l, str) or isinstance(r, str):
                if isinstance(l, str) and isinstance(r, str):
                    return l + r
                return self._to_str(l) + self._to_str(r)
            return l + r
        if op in ("-", "*", "/", "%", "^"):
            if l is None or r is None:
                return None
            if op == "-":
                return l - r
            if op == "*":
                return l * r
            if op == "/":
                if r == 0:
                    if isinstance(l, int) and isinstance(r, int):
                        raise CypherRuntimeError("division by zero")
                    return float("inf") if l > 0 else float("-inf") if l < 0 else float("nan")
                res = l / r
                if isinstance(l, int) and isinstance(r, int):
                    return int(l / r) if (l < 0) == (r < 0) or l % r == 0 else -(-l // r if l < 0 else l // -r)
                return res
            if op == "%":
                return math_fmod(l, r)
            return l ** r
        if op == "IN":
            if r is None:
                return None
            if l is None:
                return None
            return any(self._cy_eq(l, x) for x in r)
        if op == "STARTS WITH":
            if l is None or r is None:
                return None
            return isinstance(l, str) and l.startswith(r)
        if op == "ENDS WITH":
            if l is None or r is None:
                return None
            return isinstance(l, str) and l.endswith(r)
        if op == "CONTAINS":
            if l is None or r is None:
                return None
            return isinstance(l, str) and r in l
        if op == "=~":
            if l is None or r is None:
                return None
            flags = 0
            pat = r
            if pat.startswith("(?i)"):
                flags = re.IGNORECASE
                pat = pat[4:]
            return re.fullmatch(pat, l, flags) is not None
        raise CypherRuntimeError(f"unknown operator {op}")

    @staticmethod
    def _to_str(v):
        if isinstance(v, bool):
            return "true" if v else "false"
        return str(v)

    @staticmethod
    def _cy_eq(l, r):
        if isinstance(l, (Node, Edge)) and isinstance(r, (Node, Edge)):
            return type(l) is type(r) and l.id == r.id
        if isinstance(l, bool) != isinstance(r, bool):
            return False
        try:
            return bool(l == r)
        except Exception:
            return False

    @staticmethod
    def _as_bool(v):
        if v is None or isinstance(v, bool):
            return v
        raise CypherRuntimeError(f"expected boolean, got {type(v).__name__}")

    def _eval_unop(self, e, row, params):
        if e.op == "NOT":
            v = self._as_bool(self._eval(e.expr, row, params))
            return None if v is None else (not v)
        if e.op == "-":
            v = self._eval(e.expr, row, params)
            return None if v is None else -v
        if e.op == "IS NULL":
            return self._eval_null_tolerant(e.expr, row, params) is None
        if e.op == "IS NOT NULL":
            return self._eval_null_tolerant(e.expr, row, params) is not None
        raise CypherRuntimeError(f"unknown unary {e.op}")

    def _eval_null_tolerant(self, e, row, params):
        try:
            return self._eval(e, row, params)
        except CypherRuntimeError:
            return None

    def _eval_func(self, e: A.FuncCall, row, params):
        name = e.name.lower()
        if name == "reduce":
            raise CypherRuntimeError("reduce() requires accumulator syntax")
        if is_aggregate(name):
            raise CypherRuntimeError(
                f"aggregate {name}() only allowed in RETURN/WITH")
        if name in ("startnode", "endnode"):
            rel = self._eval(e.args[0], row, params)
            if rel is None:
                return None
            nid = rel.start_node if name == "startnode" else rel.end_node
            try:
                return self.engine.get_node(nid)
            except NotFoundError:
                return None
        fn = FUNCTIONS.get(name)
        if fn is None:
            raise CypherRuntimeError(f"unknown function {e.name}()")
        args = [self._eval(a, row, params) for a in e.args]
        return fn(*args)

    # ------------------------------------------------------- pattern match
    def _node_candidates(self, np: A.NodePattern, row, params) -> Iterable[Node]:
        if np.var and np.var in row:
            v = row[np.var]
            if v is None:
                return []
            return [v]
        props = {}
        if np.props is not None:
            p = self._eval(np.props, row, params)
            props = dict(p or {})
        # property-index fast path
        if np.labels and props:
            lookup = getattr(self.engine, "lookup_property_index", None)
            if lookup:
                for k, v in props.items():
                    r = lookup(np.labels[0], k, v)
                    if r is not None:
                        return [n for n in r
                                if all(lb in n.labels for lb in np.labels)
                                and self._props_match(n, props)]
        if np.labels and getattr(np, "or_labels", False):
            seen = {}
            for lb in np.labels:
                for n in self.engine.get_nodes_by_label(lb):
                    seen[n.id] = n
            cands = list(seen.values())
        elif np.labels:
            cands = self.engine.get_nodes_by_label(np.labels[0])
            if len(np.labels) > 1:
                cands = [n for n in cands if all(lb in n.labels for lb in np.labels)]
        else:
            cands = self.engine.all_nodes()
        if props:
            cands = [n for n in cands if self._props_match(n, props)]
        if getattr(np, "where", None) is not None and np.var:
            out = []
            for n in cands:
                r2 = dict(row)
                r2[np.var] = n
                if self._eval(np.where, r2, params) is True:
                    out.append(n)
            cands = out
        return cands

    @staticmethod
    def _props_match(obj, props: Dict[str, Any]) -> bool:
        for k, v in props.items():
            if obj.properties.get(k) != v:
                return False
        return True

    def _edges_from(self, node_id: str, rp: A.RelPattern) -> List[Tuple[Edge, str]]:
        """Candidate (edge, other_node_id) respecting direction and types."""
        out: List[Tuple[Edge, str]] = []
        if rp.direction in ("out", "both"):
            for e in self.engine.get_out_edges(node_id):
                if not rp.types or e.type in rp.types:
                    out.append((e, e.end_node))
        if rp.direction in ("in", "both"):
            for e in self.engine.get_in_edges(node_id):
                if not rp.types or e.type in rp.types:
                    out.append((e, e.start_node))
        return out

    def _match_path(self, path: A.PatternPath, row: Dict[str, Any], params,
                    limit: int = 0):
        """Yield extended rows matching the pattern path."""
        shortest = getattr(path, "shortest", None)
        elems = path.elements
        count = [0]

        path_nodes_var = path.var

        def emit(r, nodes, edges):
            if path_nodes_var:
                r = dict(r)
                r[path_nodes_var] = Path(nodes, edges)
            count[0] += 1
            return r

        if shortest:
            yield from self._match_shortest(path, row, params, shortest)
            return

        def walk(i, r, cur_node, nodes, edges, used_edges):
            if limit and count[0] >= limit:
                return
            if i >= len(elems):
                yield emit(r, nodes, edges)
                return
            rp: A.RelPattern = elems[i]
            np: A.NodePattern = elems[i + 1]
            rel_props = None
            if rp.props is not None:
                rel_props = dict(self._eval(rp.props, r, params) or {})

            def try_end(e_list, r2, last_node, mids=()):
                """e_list = edges traversed; mids = intermediate node objects."""
                # bind the end node pattern
                if np.var and np.var in r2:
                    bound = r2[np.var]
                    if bound is None or bound.id != last_node:
                        return
                    end_nodes = [bound]
                else:
                    try:
                        cand = self.engine.get_node(last_node)
                    except NotFoundError:
                        return
                    if np.labels and not all(lb in cand.labels for lb in np.labels):
                        return
                    if np.props is not None:
                        pr = dict(self._eval(np.props, r2, params) or {})
                        if not self._props_match(cand, pr):
                            return
                    end_nodes = [cand]
                for endn in end_nodes:
                    r3 = dict(r2)
                    if np.var:
                        r3[np.var] = endn
                    yield from walk(i + 2, r3, endn.id,
                                    nodes + list(mids) + [endn],
                                    edges + e_list,
                                    used_edges | {e.id for e in e_list})

            if not rp.var_length:
                for e, other in self._edges_from(cur_node, rp):
                    if e.id in used_edges:
                        continue
                    if rel_props and not self._props_match(e, rel_props):
                        continue
                    r2 = dict(r)
                    if rp.var:
                        if rp.var in r2 and r2[rp.var] is not None and r2[rp.var].id != e.id:
                            continue
                        r2[rp.var] = e
                    yield from try_end([e], r2, other)
            else:
                # variable-length BFS/DFS up to max_hops; n_acc carries the
                # intermediate node objects (path var correctness)
                def expand(nid, hops, e_acc, n_acc, visited_edges):
                    if limit and count[0] >= limit:
                        return
                    if hops >= rp.min_hops:
                        r2 = dict(r)
                        if rp.var:
                            r2[rp.var] = list(e_acc)
                        yield from try_end(list(e_acc), r2, nid, n_acc)
                    if hops >= rp.max_hops:
                        return
                    try:
                        nid_node = self.engine.get_node(nid)
                    except NotFoundError:
                        return
                    for e, other in self._edges_from(nid, rp):
                        if e.id in visited_edges or e.id in used_edges:
                            continue
                        if rel_props and not self._props_match(e, rel_props):
                            continue
                        yield from expand(other, hops + 1, e_acc + [e],
                                          n_acc + ([nid_node] if hops > 0 else []),
                                          visited_edges | {e.id})
                # expand() yields the 0-hop case itself when min_hops == 0
                yield from expand(cur_node, 0, [], [], set())

        start_np: A.NodePattern = elems[0]
        for n0 in self._node_candidates(start_np, row, params):
            if limit and count[0] >= limit:
                return
            r0 = dict(row)
            if start_np.var:
                r0[start_np.var] = n0
            if len(elems) == 1:
                yield emit(r0, [n0], [])
            else:
                yield from walk(1, r0, n0.id, [n0], [], set())

    def _match_shortest(self, path: A.PatternPath, row, params, kind):
        """BFS shortest path(s) between the two endpoint patterns."""
        elems = path.elements
        if len(elems) != 3:
            raise CypherRuntimeError("shortestPath needs a single relationship")
        np1, rp, np2 = elems
        starts = list(self._node_candidates(np1, row, params))
        ends = list(self._node_candidates(np2, row, params))
        end_ids = {n.id: n for n in ends}
        max_h = rp.max_hops if rp.var_length else 1
        for s in starts:
            # BFS
            from collections import deque
            q = deque([(s.id, [], [s])])
            seen = {s.id}
            found_len = None
            while q:
                nid, eacc, nacc = q.popleft()
                if found_len is not None and len(eacc) > found_len:
                    break
                if nid in end_ids and len(eacc) >= (rp.min_hops if rp.var_length else 1):
                    found_len = len(eacc)
                    r = dict(row)
                    if np1.var:
                        r[np1.var] = s
                    if np2.var:
                        r[np2.var] = nacc[-1]
                    if rp.var:
                        r[rp.var] = eacc
                    if path.var:
                        r[path.var] = Path(nacc, eacc)
                    yield r
                    if kind == "shortestpath":
                        break
                    continue
                if len(eacc) >= max_h:
                    continue
                for e, other in self._edges_from(nid, rp):
                    if other not in seen:
                        seen.add(other)
                        try:
                            onode = self.engine.get_node(other)
                        except NotFoundError:
                            continue
                        q.append((other, eacc + [e], nacc + [onode]))

    # -------------------------------------------------------------- clauses
    def _columnar_match(self, c: A.MatchClause, rows, params):
        """Vectorized WHERE over a single-label scan: evaluates the
        predicate as numpy masks on cached columns and copies only the
        surviving nodes (columnar.py). Returns rows or None."""
        if (c.optional or c.where is None or len(c.patterns) != 1
                or len(rows) != 1 or rows[0]):
            return None
        pat = c.patterns[0]
        if pat.var or len(pat.elements) != 1:
            return None
        el = pat.elements[0]
        if (not getattr(el, "var", None) or not el.labels
                or getattr(el, "or_labels", False)
                or getattr(el, "where", None) or el.props is not None):
            return None
        from .columnar import columnar_filter
        got = columnar_filter(self._colstore, el.labels[0], el.var, c.where,
                              params)
        if got is None:
            return None
        nodes, mask = got
        extra = el.labels[1:]
        out = []
        import numpy as _np
        for i in _np.nonzero(mask)[0]:
            n = nodes[int(i)]
            if extra and not all(lb in n.labels for lb in extra):
                continue
            out.append({el.var: n.copy()})
        return out

    def _exec_match(self, c: A.MatchClause, rows, params):
        fast = self._columnar_match(c, rows, params)
        if fast is not None:
            return fast
        out = []
        for row in rows:
            matched = [row]
            any_for_row = True
            for pat in c.patterns:
                nxt = []
                for r in matched:
                    nxt.extend(self._match_path(pat, r, params))
                matched = nxt
            if c.where is not None:
                from .compiler import compile_predicate
                fn = compile_predicate(c.where)
                if fn is not None:
                    try:
                        matched = [r for r in matched if fn(r, params) is True]
                    except Exception:
                        matched = [r for r in matched
                                   if self._eval(c.where, r, params) is True]
                else:
                    matched = [r for r in matched
                               if self._eval(c.where, r, params) is True]
            if matched:
                out.extend(matched)
            elif c.optional:
                r2 = dict(row)
                for pat in c.patterns:
                    for el in pat.elements:
                        if getattr(el, "var", None) and el.var not in r2:
                            r2[el.var] = None
                    if pat.var and pat.var not in r2:
                        r2[pat.var] = None
                out.append(r2)
        return out

    def _exec_unwind(self, c: A.UnwindClause, rows, params):
        out = []
        for row in rows:
            v = self._eval(c.expr, row, params)
            if v is None:
                continue
            if not isinstance(v, list):
                v = [v]
            for item in v:
                r2 = dict(row)
                r2[c.alias] = item
                if c.where is not None and \
                        self._eval(c.where, r2, params) is not True:
                    continue
                out.append(r2)
        return out

    def _create_from_pattern(self, pat: A.PatternPath, row, params):
        """CREATE semantics: unbound node vars are created; bound reused."""
        elems = pat.elements
        prev_node: Optional[Node] = None
        nodes, edges = [], []
        r = dict(row)
        for i, el in enumerate(elems):
            if isinstance(el, A.NodePattern):
                if el.var and el.var in r and r[el.var] is not None:
                    node = r[el.var]
                    if not isinstance(node, Node):
                        raise CypherRuntimeError(f"{el.var} is not a node")
                else:
                    props = {}
                    if el.props is not None:
                        props = dict(self._eval(el.props, r, params) or {})
                    node = Node(id=new_id("n"), labels=list(el.labels),
                                properties=props)
                    node = self.engine.create_node(node)
                    self.stats["nodes_created"] += 1
                    if el.var:
                        r[el.var] = node
                nodes.append(node)
                prev_node = node
            else:  # RelPattern
                rp: A.RelPattern = el
                nxt = elems[i + 1]
                # create the next node first (recursion handles chain order)
                # -> handled in loop; stash rel to create after next node
                edges.append(rp)
        # second pass: create relationships between consecutive nodes
        created_edges = []
        for j, rp in enumerate(edges):
            if rp.direction == "both":
                raise CypherRuntimeError("CREATE requires a directed relationship")
            if rp.var_length:
                raise CypherRuntimeError("cannot CREATE variable-length relationship")
            a, b = nodes[j], nodes[j + 1]
            s, t = (a, b) if rp.direction == "out" else (b, a)
            props = {}
            if rp.props is not None:
                props = dict(self._eval(rp.props, r, params) or {})
            if not rp.types:
                raise CypherRuntimeError("CREATE requires a relationship type")
            e = Edge(id=new_id("e"), type=rp.types[0], start_node=s.id,
                     end_node=t.id, properties=props)
            e = self.engine.create_edge(e)
            self.stats["edges_created"] += 1
            if rp.var:
                r[rp.var] = e
            created_edges.append(e)
        if pat.var:
            r[pat.var] = Path(nodes, created_edges)
        return r

    def _exec_create(self, c: A.CreateClause, rows, params):
        out = []
        for row in rows:
            r = row
            for pat in c.patterns:
                r = self._create_from_pattern(pat, r, params)
            out.append(r)
        return out

    def _exec_merge(self, c: A.MergeClause, rows, params):
        out = []
        for row in rows:
            found = list(self._match_path(c.pattern, row, params))
            if found:
                for r in found:
                    if c.on_match:
                        [r] = self._exec_set(c.on_match, [r], params)
                    out.append(r)
            else:
                r = self._create_from_pattern(c.pattern, row, params)
                if c.on_create:
                    [r] = self._exec_set(c.on_create, [r], params)
                out.append(r)
        return out

    def _exec_set(self, items: List[A.SetItem], rows, params):
        for row in rows:
            for it in items:
                if it.op == "label":
                    ent = self._eval(it.target, row, params)
                    if ent is None:
                        continue
                    node = self.engine.get_node(ent.id)
                    for lb in it.labels:
                        if lb not in node.labels:
                            node.labels.append(lb)
                            self.stats["labels_added"] += 1
                    node = self.engine.update_node(node)
                    self._rebind(rows, node)
                elif isinstance(it.target, A.Prop):
                    ent = self._eval(it.target.expr, row, params)
                    if ent is None:
                        continue
                    val = self._eval(it.value, row, params)
                    self._set_prop(ent, it.target.key, val, rows)
                elif isinstance(it.target, A.Var):
                    ent = row.get(it.target.name)
                    if ent is None:
                        continue
                    val = self._eval(it.value, row, params)
                    if isinstance(val, (Node, Edge)):
                        val = dict(val.properties)
                    if not isinstance(val, dict):
                        raise CypherRuntimeError("SET n = value needs a map")
                    if it.op == "+=":
                        newp = dict(ent.properties)
                        newp.update(val)
                    else:
                        newp = dict(val)
                    newp = {k: v for k, v in newp.items() if v is not None}
                    if isinstance(ent, Node):
                        n = self.engine.get_node(ent.id)
                        n.properties = newp
                        n = self.engine.update_node(n)
                        self._rebind(rows, n)
                    else:
                        ed = self.engine.get_edge(ent.id)
                        ed.properties = newp
                        ed = self.engine.update_edge(ed)
                        self._rebind(rows, ed)
                    self.stats["properties_set"] += len(val)
        return rows

    def _set_prop(self, ent, key, val, rows):
        if isinstance(ent, Node):
            n = self.engine.get_node(ent.id)
            if val is None:
                n.properties.pop(key, None)
            else:
                n.properties[key] = val
            n = self.engine.update_node(n)
            self._rebind(rows, n)
        elif isinstance(ent, Edge):
            e = self.engine.get_edge(ent.id)
            if val is None:
                e.properties.pop(key, None)
            else:
                e.properties[key] = val
            e = self.engine.update_edge(e)
            self._rebind(rows, e)
        else:
            raise CypherRuntimeError("SET target must be node or relationship")
        self.stats["properties_set"] += 1

    @staticmethod
    def _rebind(rows, ent):
        """Refresh stale copies of an updated entity in all rows."""
        for r in rows:
            for k, v in r.items():
                if type(v) is type(ent) and getattr(v, "id", None) == ent.id:
                    r[k] = ent

    def _exec_remove(self, c: A.RemoveClause, rows, params):
        for row in rows:
            for it in c.items:
                if isinstance(it, A.SetItem) and it.op == "label":
                    ent = self._eval(it.target, row, params)
                    if ent is None:
                        continue
                    node = self.engine.get_node(ent.id)
                    node.labels = [lb for lb in node.labels if lb not in it.labels]
                    node = self.engine.update_node(node)
                    self._rebind(rows, node)
                elif isinstance(it, A.Prop):
                    ent = self._eval(it.expr, row, params)
                    if ent is None:
                        continue
                    self._set_prop(ent, it.key, None, rows)
        return rows

    def _exec_delete(self, c: A.DeleteClause, rows, params):
        deleted_nodes = set()
        deleted_edges = set()
        for row in rows:
            for e in c.exprs:
                v = self._eval(e, row, params)
                if v is None:
                    continue
                vs = v if isinstance(v, list) else [v]
                for ent in vs:
                    if isinstance(ent, Node) and ent.id not in deleted_nodes:
                        if c.detach:
                            self.engine.detach_delete_node(ent.id)
                        else:
                            self.engine.delete_node(ent.id)
                        deleted_nodes.add(ent.id)
                        self.stats["nodes_deleted"] += 1
                    elif isinstance(ent, Edge) and ent.id not in deleted_edges:
                        try:
                            self.engine.delete_edge(ent.id)
                        except NotFoundError:
                            pass
                        deleted_edges.add(ent.id)
                        self.stats["edges_deleted"] += 1
        return rows

    # ------------------------------------------------------ projection core
    def _project(self, c: A.ReturnClause, rows, params):
        items = list(c.items)
        if c.star:
            seen_names = {it.alias for it in items}
            vars_ = sorted({k for r in rows for k in r.keys()})
            star_items = [A.ReturnItem(A.Var(v), v) for v in vars_
                          if v not in seen_names]
            items = star_items + items

        names = []
        for it in items:
            names.append(it.alias or self._expr_name(it.expr))

        has_agg = any(self._contains_aggregate(it.expr) for it in items)
        pairs = []  # (outrow, binding-for-order-by)
        if has_agg:
            for outrow, src in self._aggregate(items, rows, params):
                binding = dict(src)
                binding.update(zip(names, outrow))
                pairs.append((outrow, binding))
        else:
            for row in rows:
                outrow = [self._eval(it.expr, row, params) for it in items]
                binding = dict(row)
                binding.update(zip(names, outrow))
                pairs.append((outrow, binding))

        if c.distinct:
            seen = set()
            uniq = []
            for p in pairs:
                k = tuple(_hkey(v) for v in p[0])
                if k not in seen:
                    seen.add(k)
                    uniq.append(p)
            pairs = uniq

        if c.order_by:
            # order expressions may reference aliases OR pre-projection vars
            def keys_for(binding):
                out = []
                for expr, asc in c.order_by:
                    nm = self._expr_name(expr)
                    if nm in binding:
                        v = binding[nm]
                    else:
                        try:
                            v = self._eval(expr, binding, params)
                        except CypherRuntimeError:
                            v = None
                    out.append((v, asc))
                return out

            decorated = [(keys_for(b), outrow, b) for outrow, b in pairs]

            import functools as _ft

            def cmp(a, b):
                for (va, asc), (vb, _) in zip(a[0], b[0]):
                    if self._cy_eq(va, vb):
                        continue
                    # NULLs sort last (Neo4j)
                    if va is None:
                        return 1
                    if vb is None:
                        return -1
                    try:
                        lt = va < vb
                    except TypeError:
                        lt = str(type(va)) < str(type(vb))
                    return (-1 if lt else 1) * (1 if asc else -1)
                return 0

            decorated.sort(key=_ft.cmp_to_key(cmp))
            pairs = [(outrow, b) for _, outrow, b in decorated]

        out_rows = [p[0] for p in pairs]
        if c.skip is not None:
            out_rows = out_rows[int(self._eval(c.skip, {}, params)):]
        if c.limit is not None:
            out_rows = out_rows[:int(self._eval(c.limit, {}, params))]
        return names, out_rows

    def _contains_aggregate(self, e) -> bool:
        if isinstance(e, A.FuncCall):
            if is_aggregate(e.name):
                return True
            return any(self._contains_aggregate(a) for a in e.args)
        for attr in ("left", "right", "expr", "index", "test", "default"):
            v = getattr(e, attr, None)
            if v is not None and not isinstance(v, str) and self._contains_aggregate(v):
                return True
        if isinstance(e, A.ListLit):
            return any(self._contains_aggregate(x) for x in e.items)
        if isinstance(e, A.MapLit):
            return any(self._contains_aggregate(v) for _, v in e.items)
        if isinstance(e, A.Case):
            if e.test is not None and self._contains_aggregate(e.test):
                return True
            return any(self._contains_aggregate(w) or self._contains_aggregate(t)
                       for w, t in e.whens)
        return False

    def _aggregate(self, items, rows, params):
        group_items = [it for it in items if not self._contains_aggregate(it.expr)]
        groups: Dict[tuple, Dict] = {}
        order: List[tuple] = []
        for row in rows:
            gkey = tuple(_hkey(self._eval(it.expr, row, params)) for it in group_items)
            g = groups.get(gkey)
            if g is None:
                g = {"row": row, "aggs": {}}
                groups[gkey] = g
                order.append(gkey)
            for idx, it in enumerate(items):
                if self._contains_aggregate(it.expr):
                    self._agg_accumulate(it.expr, row, params, g["aggs"], (idx,))
        if not rows and not group_items:
            # aggregates over empty input produce a single row
            groups[()] = {"row": {}, "aggs": {}}
            order.append(())
        out = []
        for gkey in order:
            g = groups[gkey]
            outrow = []
            for idx, it in enumerate(items):
                if self._contains_aggregate(it.expr):
                    outrow.append(self._agg_finalize(it.expr, g["row"], params,
                                                     g["aggs"], (idx,)))
                else:
                    outrow.append(self._eval(it.expr, g["row"], params))
            out.append((outrow, g["row"]))
        return out

    def _agg_accumulate(self, e, row, params, aggs, path):
        if isinstance(e, A.FuncCall) and is_aggregate(e.name):
            key = path
            agg = aggs.get(key)
            if agg is None:
                agg = Aggregator(e.name, e.distinct)
                aggs[key] = agg
            if e.star:
                agg.add(1)
            elif e.args:
                agg.add(self._eval(e.args[0], row, params))
            return
        for i, sub in enumerate(self._children(e)):
            self._agg_accumulate(sub, row, params, aggs, path + (i,))

    def _agg_finalize(self, e, row, params, aggs, path):
        if isinstance(e, A.FuncCall) and is_aggregate(e.name):
            agg = aggs.get(path)
            if agg is None:
                agg = Aggregator(e.name, e.distinct)
            extra = None
            if e.name.lower().startswith("percentile") and len(e.args) > 1:
                extra = self._eval(e.args[1], row, params)
            return agg.result(extra)
        if isinstance(e, A.BinOp):
            l = self._agg_finalize(e.left, row, params, aggs, path + (0,))
            r = self._agg_finalize(e.right, row, params, aggs, path + (1,))
            return self._eval_binop(A.BinOp(e.op, A.Lit(l), A.Lit(r)), row, params)
        if isinstance(e, A.UnOp):
            v = self._agg_finalize(e.expr, row, params, aggs, path + (0,))
            return self._eval_unop(A.UnOp(e.op, A.Lit(v)), row, params)
        if isinstance(e, A.FuncCall):
            args = [self._agg_finalize(a, row, params, aggs, path + (i,))
                    for i, a in enumerate(e.args)]
            fn = FUNCTIONS.get(e.name.lower())
            if fn is None:
                raise CypherRuntimeError(f"unknown function {e.name}()")
            return fn(*args)
        return self._eval(e, row, params)

    @staticmethod
    def _children(e):
        if isinstance(e, A.BinOp):
            return [e.left, e.right]
        if isinstance(e, A.UnOp):
            return [e.expr]
        if isinstance(e, A.FuncCall):
            return e.args
        if isinstance(e, A.ListLit):
            return e.items
        if isinstance(e, A.MapLit):
            return [v for _, v in e.items]
        if isinstance(e, A.Prop):
            return [e.expr]
        if isinstance(e, A.Index):
            return [e.expr] + ([e.index] if e.index is not None else [])
        return []

    @staticmethod
    def _expr_name(e) -> str:
        if isinstance(e, A.Var):
            return e.name
        if isinstance(e, A.Prop):
            return f"{Executor._expr_name(e.expr)}.{e.key}"
        if isinstance(e, A.FuncCall):
            if e.star:
                return "count(*)"
            inner = ", ".join(Executor._expr_name(a) for a in e.args)
            return f"{e.name}({inner})"
        if isinstance(e, A.Lit):
            return repr(e.value)
        if isinstance(e, A.Param):
            return f"${e.name}"
        if isinstance(e, A.BinOp):
            return f"{Executor._expr_name(e.left)} {e.op} {Executor._expr_name(e.right)}"
        return "expr"

    def _exec_return(self, c: A.ReturnClause, rows, params):
        return self._project(c, rows, params)

    def _exec_with(self, c: A.WithClause, rows, params):
        names, out_rows = self._project(c, rows, params)
        new_rows = [dict(zip(names, r)) for r in out_rows]
        if c.where is not None:
            new_rows = [r for r in new_rows
                        if self._eval(c.where, r, params) is True]
        return new_rows

    def _exec_call(self, c: A.CallClause, rows, params, standalone=False):
        proc = self.procedures.get(c.proc.lower())
        if proc is None:
            raise CypherRuntimeError(f"unknown procedure {c.proc}")
        out_rows = []
        yields = c.yields
        for row in rows:
            args = [self._eval(a, row, params) for a in c.args]
            cols, prows = proc(self, *args)
            if yields and yields[0][0] == "*":
                # YIELD * expands to every procedure column
                yields = [(col, None) for col in cols]
            for pr in prows:
                rec = dict(zip(cols, pr)) if isinstance(pr, (list, tuple)) else dict(pr)
                r2 = dict(row)
                if yields:
                    for (yname, alias) in yields:
                        if yname not in rec:
                            raise CypherRuntimeError(
                                f"procedure {c.proc} does not yield {yname}")
                        r2[alias or yname] = rec[yname]
                else:
                    r2.update(rec)
                if c.where is not None and self._eval(c.where, r2, params) is not True:
                    continue
                out_rows.append(r2)
        if getattr(c, "limit", None) is not None:
            lim = self._eval(c.limit, {}, params)
            out_rows = out_rows[:int(lim)]
        if standalone and not yields:
            # standalone CALL returns all procedure columns
            if out_rows:
                cols = [k for k in out_rows[0].keys()]
                return out_rows, (cols, [[r[k] for k in cols] for r in out_rows])
            return out_rows, ([], [])
        if standalone and yields:
            cols = [alias or y for y, alias in yields]
            return out_rows, (cols, [[r[k] for k in cols] for r in out_rows])
        return out_rows, None

    def _exec_foreach(self, c: A.ForeachClause, rows, params):
        for row in rows:
            src = self._eval(c.source, row, params) or []
            for item in src:
                r2 = dict(row)
                r2[c.var] = item
                sub = [r2]
                for upd in c.updates:
                    if isinstance(upd, A.SetClause):
                        sub = self._exec_set(upd.items, sub, params)
                    elif isinstance(upd, A.CreateClause):
                        sub = self._exec_create(upd, sub, params)
                    elif isinstance(upd, A.MergeClause):
                        sub = self._exec_merge(upd, sub, params)
                    elif isinstance(upd, A.DeleteClause):
                        sub = self._exec_delete(upd, sub, params)
        return rows


def math_fmod(l, r):
    if r == 0:
        raise CypherRuntimeError("modulo by zero")
    v = l - r * int(l / r) if isinstance(l, int) and isinstance(r, int) else None
    if v is None:
        import math
        return math.fmod(l, r)
    return v
