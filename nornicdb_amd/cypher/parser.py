"""Recursive-descent Cypher parser producing the AST in ast.py.

Coverage target: the clause set the reference executes via keyword dispatch
(reference pkg/cypher/executor.go:490 + match.go/create.go/merge.go/
match_with.go/executor_mutations.go/executor_subqueries.go), expressed as a
single grammar instead of string dispatch.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

from . import ast as A
from .lexer import CypherSyntaxError, Token, tokenize


class Parser:
    def __init__(self, text: str):
        self.toks = tokenize(text)
        self.i = 0

    # ---- token helpers ----
    def peek(self, k=0) -> Token:
        return self.toks[min(self.i + k, len(self.toks) - 1)]

    def next(self) -> Token:
        t = self.toks[self.i]
        if t.kind != "EOF":
            self.i += 1
        return t

    def at_kw(self, *kws) -> bool:
        t = self.peek()
        return t.kind == "KW" and t.value in kws

    def at_op(self, *ops) -> bool:
        t = self.peek()
        return t.kind == "OP" and t.value in ops

    def eat_kw(self, kw) -> Token:
        if not self.at_kw(kw):
            raise CypherSyntaxError(f"expected {kw}, got {self.peek().value!r} at {self.peek().pos}")
        return self.next()

    def eat_op(self, op) -> Token:
        if not self.at_op(op):
            raise CypherSyntaxError(f"expected {op!r}, got {self.peek().value!r} at {self.peek().pos}")
        return self.next()

    def try_kw(self, *kws) -> bool:
        if self.at_kw(*kws):
            self.next()
            return True
        return False

    def try_op(self, op) -> bool:
        if self.at_op(op):
            self.next()
            return True
        return False

    def name_part(self) -> str:
        """Function/property name segment: any identifier or keyword."""
        t = self.peek()
        if t.kind == "IDENT":
            return self.next().value
        if t.kind == "KW":
            return self.next().value.lower()
        raise CypherSyntaxError(f"expected name, got {t.value!r} at {t.pos}")

    def label_name(self) -> str:
        """Label / rel-type name: identifiers OR reserved words, original
        case preserved (Neo4j allows `:Order`, `:Match` etc.)."""
        t = self.peek()
        if t.kind == "IDENT":
            return self.next().value
        if t.kind == "KW":
            tok = self.next()
            return getattr(tok, "raw", "") or tok.value
        raise CypherSyntaxError(f"expected label, got {t.value!r} at {t.pos}")

    def ident(self) -> str:
        t = self.peek()
        if t.kind == "IDENT":
            return self.next().value
        # allow non-reserved keywords as identifiers (e.g. count, all)
        if t.kind == "KW" and t.value in ("COUNT", "ALL", "ANY", "NONE", "SINGLE",
                                          "EXISTS", "INDEX", "UNIQUE", "ON", "BY",
                                          "DATABASE", "SHOW"):
            return self.next().value.lower()
        raise CypherSyntaxError(f"expected identifier, got {t.value!r} at {t.pos}")

    # ---- entry ----
    def parse(self) -> A.Query:
        q = self._query()
        if self.peek().kind != "EOF":
            t = self.peek()
            raise CypherSyntaxError(f"unexpected {t.value!r} at {t.pos}")
        return q

    def _query(self) -> A.Query:
        explain = profile = False
        if self.try_kw("EXPLAIN"):
            explain = True
        elif self.try_kw("PROFILE"):
            profile = True
        clauses = []
        while True:
            c = self._clause()
            if c is None:
                break
            clauses.append(c)
            if self.try_op(";"):
                break
        if not clauses:
            raise CypherSyntaxError("empty query")
        q = A.Query(clauses, explain=explain, profile=profile)
        if self.at_kw("UNION"):
            self.next()
            all_ = self.try_kw("ALL")
            rest = self._query()
            q.union = ("UNION ALL" if all_ else "UNION", rest)
        return q

    def _clause(self):
        if self.at_kw("MATCH") or self.at_kw("OPTIONAL"):
            return self._match()
        if self.at_kw("CREATE"):
            return self._create()
        if self.at_kw("MERGE"):
            return self._merge()
        if self.at_kw("SET"):
            return self._set()
        if self.at_kw("REMOVE"):
            return self._remove()
        if self.at_kw("DELETE") or self.at_kw("DETACH"):
            return self._delete()
        if self.at_kw("RETURN"):
            return self._return(A.ReturnClause)
        if self.at_kw("WITH"):
            return self._with()
        if self.at_kw("UNWIND"):
            return self._unwind()
        if self.at_kw("CALL"):
            return self._call()
        if self.at_kw("FOREACH"):
            return self._foreach()
        if self.at_kw("SHOW"):
            return self._show()
        if (self.peek().kind == "IDENT" and self.peek().value.upper() == "USE"
                and self.peek(1).kind in ("IDENT", "KW")):
            self.next()
            name = self.name_part()
            while self.try_op("."):
                name += "." + self.name_part()
            return A.UseClause(name)
        if self.at_kw("DROP") or (self.peek().kind == "IDENT"
                                  and self.peek().value.upper() == "DROP"):
            return self._drop_schema()
        return None

    # ---- clauses ----
    def _match(self):
        optional = self.try_kw("OPTIONAL")
        self.eat_kw("MATCH")
        pats = [self._pattern_path()]
        while self.try_op(","):
            pats.append(self._pattern_path())
        # planner hints: USING INDEX/SCAN/JOIN ... — accepted and ignored
        # (our executor picks indexes itself)
        while self._at_word("USING"):
            self.next()
            t = self.next()  # INDEX | SCAN | JOIN | RANGE etc.
            if t.value.upper() == "JOIN":
                self._eat_word("ON")
                self.ident()
            else:
                if t.value.upper() in ("INDEX", "SCAN") and self._at_word("SEEK"):
                    self.next()
                self.ident()          # variable
                if self.try_op(":"):
                    self.label_name()
                    if self.try_op("("):
                        while not self.at_op(")"):
                            self.next()
                        self.eat_op(")")
        where = None
        if self.try_kw("WHERE"):
            where = self._expr()
        return A.MatchClause(pats, optional=optional, where=where)

    def _create(self):
        self.eat_kw("CREATE")
        or_replace = False
        if self.at_kw("OR") and self.peek(1).kind in ("IDENT", "KW") \
                and self.peek(1).value.upper() == "REPLACE":
            self.next()
            self._eat_word("REPLACE")
            or_replace = True
        kindword = None
        if self._at_ident("VECTOR") or self._at_ident("FULLTEXT") \
                or self._at_ident("TEXT") or self._at_ident("RANGE") \
                or self._at_ident("POINT") or self._at_ident("LOOKUP"):
            nxt = self.peek(1)
            if nxt.kind == "KW" and nxt.value == "INDEX":
                kindword = self.next().value.upper()
        if self.at_kw("INDEX") or kindword is not None:
            return self._create_index(kindword or "RANGE", or_replace)
        if self.at_kw("CONSTRAINT"):
            return self._create_constraint(or_replace)
        if self._at_word("COMPOSITE"):
            self.next()
            self._eat_word("DATABASE")
            name = self.name_part()
            constituents = []
            while self._at_word("ALIAS"):
                self.next()
                alias = self.name_part()
                self._eat_word("FOR")
                self._eat_word("DATABASE")
                target = self.name_part()
                constituents.append((alias, target))
            return A.SchemaCommand("create", "composite", name=name,
                                   props=[f"{a}:{t}" for a, t in constituents])
        if self.at_kw("DATABASE") or self._at_word("DATABASE"):
            self.next()
            ine, _ = (self._if_not_exists()
                      if self._at_word("IF") else (False, False))
            name = self.name_part()
            while self.try_op("."):
                name += "." + self.name_part()
            if self._at_word("IF"):
                ine, _ = self._if_not_exists()
            return A.SchemaCommand("create", "database", name=name,
                                   if_not_exists=ine, or_replace=or_replace)
        if self._at_word("ALIAS"):
            self.next()
            name = self.name_part()
            while self.try_op("."):
                name += "." + self.name_part()
            ine, _ = self._if_not_exists()
            self._eat_word("FOR")
            self._eat_word("DATABASE")
            target = self.name_part()
            while self.try_op("."):
                target += "." + self.name_part()
            return A.SchemaCommand("create", "alias", name=name, label=target,
                                   if_not_exists=ine, or_replace=or_replace)
        if or_replace:
            raise CypherSyntaxError("expected INDEX or CONSTRAINT after "
                                    "CREATE OR REPLACE")
        pats = [self._pattern_path()]
        while self.try_op(","):
            pats.append(self._pattern_path())
        return A.CreateClause(pats)

    # ---- schema DDL ----
    # (FOR/IF/REQUIRE are not reserved words in the lexer; match by value)
    def _at_ident(self, word):
        t = self.peek()
        return t.kind in ("IDENT", "KW") and t.value.upper() == word

    def _at_word(self, word):
        t = self.peek()
        return t.kind in ("IDENT", "KW") and t.value.upper() == word

    def _eat_word(self, word):
        t = self.next()
        if t.value.upper() != word:
            raise CypherSyntaxError(
                f"expected {word}, got {t.value!r} at {t.pos}")

    def _eat_ident(self, word):
        t = self.next()
        if t.value.upper() != word:
            raise CypherSyntaxError(f"expected {word}, got {t.value!r} at {t.pos}")

    def _if_not_exists(self):
        if self._at_word("IF"):
            self.next()
            if self._at_word("NOT"):
                self.next()
                self._eat_word("EXISTS")
                return (True, False)
            self._eat_word("EXISTS")
            return (False, True)
        return (False, False)

    def _create_index(self, kindword, or_replace):
        self.next()  # INDEX
        # legacy 3.x: CREATE INDEX ON :Label(prop)
        if self.at_kw("ON") and self.peek(1).kind == "OP" and self.peek(1).value == ":":
            self.next()
            self.eat_op(":")
            label = self.ident()
            self.eat_op("(")
            props = [self.ident()]
            while self.try_op(","):
                props.append(self.ident())
            self.eat_op(")")
            return A.SchemaCommand("create", "index", label=label, props=props)
        name = None
        if not (self._at_word("IF") or self._at_word("FOR")):
            name = self.name_part()
        ine, _ = self._if_not_exists()
        self._eat_word("FOR")
        self.eat_op("(")
        var = self.ident()
        self.eat_op(":")
        label = self.ident()
        while self.try_op("|"):
            self.ident()  # multi-label fulltext: first label indexed
        self.eat_op(")")
        self._eat_word("ON")
        if self._at_ident("EACH"):   # fulltext: ON EACH [n.p, ...]
            self.next()
            self.eat_op("[")
            props = [self._on_prop(var)]
            while self.try_op(","):
                props.append(self._on_prop(var))
            self.eat_op("]")
        else:
            self.eat_op("(")
            props = [self._on_prop(var)]
            while self.try_op(","):
                props.append(self._on_prop(var))
            self.eat_op(")")
        options = None
        if self._at_ident("OPTIONS"):
            self.next()
            options = self._expr()
        kind = {"VECTOR": "vector", "FULLTEXT": "fulltext"}.get(
            kindword, "index")
        return A.SchemaCommand("create", kind, name=name, label=label,
                               props=props, options=options,
                               if_not_exists=ine, or_replace=or_replace)

    def _on_prop(self, var):
        v = self.ident()
        if v != var:
            raise CypherSyntaxError(f"unknown variable {v!r} in index prop")
        self.eat_op(".")
        return self.name_part()

    def _create_constraint(self, or_replace):
        self.next()  # CONSTRAINT
        name = None
        if not (self._at_word("IF") or self._at_word("FOR")
                or self._at_word("ON")):
            name = self.name_part()
        ine, _ = self._if_not_exists()
        # FOR (n:Label) REQUIRE ...   (5.x)  |  ON (n:Label) ASSERT ... (3.x/4.x)
        if self._at_word("FOR") or self._at_word("ON"):
            self.next()
        else:
            raise CypherSyntaxError("expected FOR or ON")
        self.eat_op("(")
        var = self.ident()
        self.eat_op(":")
        label = self.ident()
        self.eat_op(")")
        if self._at_ident("REQUIRE") or self._at_ident("ASSERT"):
            self.next()
        else:
            raise CypherSyntaxError("expected REQUIRE or ASSERT")
        # exists(n.prop)  (legacy)
        if (self._at_ident("EXISTS") or self.at_kw("EXISTS")) and \
                self.peek(1).kind == "OP" and self.peek(1).value == "(":
            self.next()
            self.eat_op("(")
            prop = self._on_prop(var)
            self.eat_op(")")
            return A.SchemaCommand("create", "constraint", name=name,
                                   label=label, props=[prop],
                                   constraint_kind="exists",
                                   if_not_exists=ine, or_replace=or_replace)
        props = [self._on_prop(var)]
        while self.try_op(","):
            props.append(self._on_prop(var))
        self.eat_kw("IS")
        if self._at_word("UNIQUE"):
            self.next()
            ck = "unique"
        elif self._at_word("NOT"):
            self.next()
            self._eat_word("NULL")
            ck = "exists"
        elif self._at_ident("NODE"):
            self.next()
            self._eat_word("KEY")
            ck = "node_key"
        else:
            raise CypherSyntaxError("expected UNIQUE, NOT NULL or NODE KEY")
        return A.SchemaCommand("create", "constraint", name=name, label=label,
                               props=props, constraint_kind=ck,
                               if_not_exists=ine, or_replace=or_replace)

    def _drop_schema(self):
        self.next()  # DROP
        if self.at_kw("INDEX"):
            self.next()
            # legacy: DROP INDEX ON :Label(prop)
            if self.at_kw("ON"):
                self.next()
                self.eat_op(":")
                label = self.ident()
                self.eat_op("(")
                props = [self.ident()]
                while self.try_op(","):
                    props.append(self.ident())
                self.eat_op(")")
                return A.SchemaCommand("drop", "index", label=label,
                                       props=props)
            name = self.name_part()
            _, ie = self._if_not_exists()
            return A.SchemaCommand("drop", "index", name=name, if_exists=ie)
        if self.at_kw("CONSTRAINT"):
            self.next()
            name = self.name_part()
            _, ie = self._if_not_exists()
            return A.SchemaCommand("drop", "constraint", name=name,
                                   if_exists=ie)
        if self.at_kw("DATABASE") or self._at_word("DATABASE"):
            self.next()
            name = self.name_part()
            _, ie = self._if_not_exists()
            return A.SchemaCommand("drop", "database", name=name,
                                   if_exists=ie)
        if self._at_word("ALIAS"):
            self.next()
            name = self.name_part()
            _, ie = self._if_not_exists()
            if self._at_word("FOR"):
                self.next()
                self._eat_word("DATABASE")
            return A.SchemaCommand("drop", "alias", name=name, if_exists=ie)
        raise CypherSyntaxError(
            "expected INDEX, CONSTRAINT, DATABASE or ALIAS after DROP")

    def _show(self):
        self.eat_kw("SHOW")
        t = self.next()
        word = t.value.upper()
        type_filter = None
        if word in ("VECTOR", "FULLTEXT", "TEXT", "RANGE", "POINT",
                    "LOOKUP", "BTREE"):
            # SHOW VECTOR INDEXES etc. — type-filtered index listing
            type_filter = word
            t = self.next()
            word = t.value.upper()
        kinds = {"INDEX": "indexes", "INDEXES": "indexes",
                 "CONSTRAINT": "constraints", "CONSTRAINTS": "constraints",
                 "DATABASE": "databases", "DATABASES": "databases",
                 "PROCEDURE": "procedures", "PROCEDURES": "procedures",
                 "FUNCTION": "functions", "FUNCTIONS": "functions",
                 "TRANSACTION": "transactions",
                 "TRANSACTIONS": "transactions",
                 "SETTING": "settings", "SETTINGS": "settings",
                 "ALIAS": "aliases", "ALIASES": "aliases"}
        if word not in kinds or (type_filter and kinds[word] != "indexes"):
            raise CypherSyntaxError(f"cannot SHOW {t.value!r}")
        # optional YIELD ... (accepted, ignored: full rows returned)
        if self.try_kw("YIELD"):
            while self.peek().kind in ("IDENT", "KW") and not self.at_kw("RETURN"):
                self.next()
                if not self.try_op(","):
                    break
        return A.SchemaCommand("show", kinds[word],
                               type_filter=type_filter)

    def _merge(self):
        self.eat_kw("MERGE")
        pat = self._pattern_path()
        on_create, on_match = [], []
        while self.at_kw("ON"):
            self.next()
            if self.try_kw("CREATE"):
                self.eat_kw("SET")
                on_create.extend(self._set_items())
            elif self.try_kw("MATCH"):
                self.eat_kw("SET")
                on_match.extend(self._set_items())
            else:
                raise CypherSyntaxError("expected ON CREATE/ON MATCH")
        return A.MergeClause(pat, on_create, on_match)

    def _set(self):
        self.eat_kw("SET")
        return A.SetClause(self._set_items())

    def _set_items(self):
        items = [self._set_item()]
        while self.try_op(","):
            items.append(self._set_item())
        return items

    def _set_item(self):
        target = self._expr_atom_chain()
        if self.at_op(":"):
            labels = []
            while self.try_op(":"):
                labels.append(self.label_name())
            return A.SetItem(target, None, op="label", labels=labels)
        if self.try_op("+="):
            return A.SetItem(target, self._expr(), op="+=")
        self.eat_op("=")
        return A.SetItem(target, self._expr(), op="=")

    def _remove(self):
        self.eat_kw("REMOVE")
        items = []
        while True:
            target = self._expr_atom_chain()
            if self.at_op(":"):
                labels = []
                while self.try_op(":"):
                    labels.append(self.label_name())
                items.append(A.SetItem(target, None, op="label", labels=labels))
            else:
                items.append(target)
            if not self.try_op(","):
                break
        return A.RemoveClause(items)

    def _delete(self):
        detach = self.try_kw("DETACH")
        self.eat_kw("DELETE")
        exprs = [self._expr()]
        while self.try_op(","):
            exprs.append(self._expr())
        return A.DeleteClause(exprs, detach=detach)

    def _return(self, cls):
        self.next()  # RETURN or WITH
        distinct = self.try_kw("DISTINCT")
        star = False
        items = []
        if self.try_op("*"):
            star = True
            if self.try_op(","):
                items = self._return_items()
        else:
            items = self._return_items()
        order, skip, limit = [], None, None
        if self.try_kw("ORDER"):
            self.eat_kw("BY")
            while True:
                e = self._expr()
                asc = True
                if self.try_kw("DESC") or self.try_kw("DESCENDING"):
                    asc = False
                else:
                    self.try_kw("ASC") or self.try_kw("ASCENDING")
                order.append((e, asc))
                if not self.try_op(","):
                    break
        if self.try_kw("SKIP"):
            skip = self._expr()
        if self.try_kw("LIMIT"):
            limit = self._expr()
        c = cls(items=items, star=star, distinct=distinct,
                order_by=order, skip=skip, limit=limit)
        return c

    def _return_items(self):
        items = [self._return_item()]
        while self.try_op(","):
            items.append(self._return_item())
        return items

    def _return_item(self):
        e = self._expr()
        alias = None
        if self.try_kw("AS"):
            t = self.peek()
            if t.kind in ("IDENT",):
                alias = self.next().value
            elif t.kind == "KW":
                alias = self.next().value.lower()
            elif t.kind == "STRING":
                alias = self.next().value
            else:
                raise CypherSyntaxError(f"expected alias at {t.pos}")
        return A.ReturnItem(e, alias)

    def _with(self):
        c = self._return(A.WithClause)
        if self.try_kw("WHERE"):
            c.where = self._expr()
        return c

    def _unwind(self):
        self.eat_kw("UNWIND")
        e = self._expr()
        self.eat_kw("AS")
        alias = self.ident()
        where = self._expr() if self.try_kw("WHERE") else None
        return A.UnwindClause(e, alias, where)

    def _call(self):
        self.eat_kw("CALL")
        if self.at_op("{"):
            q = self._braced_query()
            in_tx = False
            rows_per_tx = 1000
            if self.at_kw("IN"):
                self.next()
                t = self.next()  # TRANSACTIONS (not a reserved word)
                if t.value.upper() != "TRANSACTIONS":
                    raise CypherSyntaxError(
                        f"expected TRANSACTIONS, got {t.value!r} at {t.pos}")
                in_tx = True
                if (self.peek().kind == "IDENT"
                        and self.peek().value.upper() == "OF"):
                    self.next()
                    n = self._expr()
                    t = self.next()
                    if t.value.upper() != "ROWS":
                        raise CypherSyntaxError(
                            f"expected ROWS, got {t.value!r} at {t.pos}")
                    if isinstance(n, A.Lit):
                        rows_per_tx = int(n.value)
            return A.SubqueryCallClause(q, in_transactions=in_tx,
                                        rows_per_tx=rows_per_tx)
        name = self.ident()
        while self.try_op("."):
            name += "." + self.name_part()
        args = []
        if self.try_op("("):
            if not self.at_op(")"):
                # implicit-map call form: CALL p(key: v, key2: v2) — the
                # reference's regex parser accepts bare config entries
                # (gds tests: stream(sourceNode: 'x', topK: 5))
                t0, t1 = self.peek(), self.peek(1)
                if t0.kind in ("IDENT", "KW") and t1.kind == "OP" \
                        and t1.value == ":":
                    items = []
                    while True:
                        k = self.next().value
                        self.eat_op(":")
                        items.append((k, self._expr()))
                        if not self.try_op(","):
                            break
                    args.append(A.MapLit(items))
                else:
                    args.append(self._expr())
                    while self.try_op(","):
                        args.append(self._expr())
            self.eat_op(")")
        yields = []
        where = None
        limit = None
        if self.try_kw("YIELD"):
            if self.try_op("*"):
                yields.append(("*", None))
            else:
                while True:
                    y = self.ident()
                    alias = None
                    if self.try_kw("AS"):
                        alias = self.ident()
                    yields.append((y, alias))
                    if not self.try_op(","):
                        break
            if self.try_kw("WHERE"):
                where = self._expr()
            if self.try_kw("LIMIT"):
                limit = self._expr()
        return A.CallClause(name, args, yields, where, limit)

    def _braced_query(self) -> A.Query:
        """Parse `{ <clauses> }`; a bare pattern (EXISTS shorthand) is
        desugared to MATCH pattern [WHERE ...] RETURN 1."""
        self.eat_op("{")
        t = self.peek()
        if t.kind == "KW" and t.value in (
                "MATCH", "OPTIONAL", "WITH", "UNWIND", "CALL", "CREATE",
                "MERGE", "RETURN", "FOREACH", "SET", "DELETE", "DETACH"):
            q = self._query()
        else:
            pat = self._pattern_path()
            where = self._expr() if self.try_kw("WHERE") else None
            q = A.Query([A.MatchClause([pat], where=where)])
        self.eat_op("}")
        return q

    @staticmethod
    def _ensure_return(q: A.Query) -> A.Query:
        if not any(isinstance(c, A.ReturnClause) for c in q.clauses):
            q.clauses.append(A.ReturnClause([A.ReturnItem(A.Lit(1), "one")]))
        return q

    def _foreach(self):
        self.eat_kw("FOREACH")
        self.eat_op("(")
        var = self.ident()
        self.eat_kw("IN")
        src = self._expr()
        self.eat_op("|")
        updates = []
        while not self.at_op(")"):
            c = self._clause()
            if c is None:
                break
            updates.append(c)
        self.eat_op(")")
        return A.ForeachClause(var, src, updates)

    # ---- patterns ----
    def _pattern_path(self) -> A.PatternPath:
        var = None
        if (self.peek().kind == "IDENT" and self.peek(1).kind == "OP"
                and self.peek(1).value == "="
                and ((self.peek(2).kind == "OP" and self.peek(2).value == "(")
                     or (self.peek(2).kind == "IDENT"
                         and self.peek(2).value.lower() in ("shortestpath", "allshortestpaths")))):
            var = self.next().value
            self.next()  # =
        # shortestPath(...) / allShortestPaths(...)
        if (self.peek().kind == "IDENT"
                and self.peek().value.lower() in ("shortestpath", "allshortestpaths")
                and self.peek(1).kind == "OP" and self.peek(1).value == "("):
            fn = self.next().value.lower()
            self.eat_op("(")
            inner = self._pattern_path()
            self.eat_op(")")
            inner.var = var or inner.var
            # mark via attribute; executor reads it
            inner.shortest = fn  # type: ignore[attr-defined]
            return inner
        elems = [self._node_pattern()]
        while self.at_op("-", "<-", "<"):
            rel = self._rel_pattern()
            node = self._node_pattern()
            elems.append(rel)
            elems.append(node)
        return A.PatternPath(elems, var=var)

    def _node_pattern(self) -> A.NodePattern:
        self.eat_op("(")
        var = None
        t = self.peek()
        if t.kind == "IDENT":
            var = self.next().value
        elif t.kind == "KW" and t.value in ("END", "START", "COUNT", "ALL",
                                            "ANY", "NONE", "SINGLE", "INDEX",
                                            "UNIQUE", "DATABASE", "SHOW",
                                            "BY", "ON", "CONTAINS"):
            # soft keywords usable as variable names (Neo4j allows them)
            var = getattr(t, "raw", "") or t.value.lower()
            self.next()
        labels = []
        or_labels = False
        while self.try_op(":"):
            labels.append(self.label_name())
            while self.try_op("|"):       # :A|B -> OR semantics
                labels.append(self.label_name())
                or_labels = True
        props = None
        if self.at_op("{"):
            props = self._map_lit()
        elif self.peek().kind == "PARAM":
            props = A.Param(self.next().value)
        where = None
        if self._at_word("WHERE"):
            self.next()
            where = self._expr()
        self.eat_op(")")
        return A.NodePattern(var, labels, props, or_labels=or_labels,
                             where=where)

    def _rel_pattern(self) -> A.RelPattern:
        direction = "both"
        if self.try_op("<-"):
            direction = "in"
            left_arrow = True
        elif self.try_op("<"):
            self.eat_op("-")
            direction = "in"
        else:
            self.eat_op("-")
        var, types, props = None, [], None
        min_h, max_h, var_len = 1, 1, False
        if self.try_op("["):
            t = self.peek()
            if t.kind == "IDENT":
                var = self.next().value
            while self.try_op(":"):
                types.append(self.label_name())
                while self.try_op("|"):
                    self.try_op(":")
                    types.append(self.label_name())
            if self.try_op("*"):
                var_len = True
                min_h, max_h = 1, 15
                if self.peek().kind == "INT":
                    min_h = int(self.next().value)
                    max_h = min_h
                if self.try_op(".."):
                    max_h = 15
                    if self.peek().kind == "INT":
                        max_h = int(self.next().value)
            if self.at_op("{"):
                props = self._map_lit()
            elif self.peek().kind == "PARAM":
                props = A.Param(self.next().value)
            self.eat_op("]")
        if self.try_op("->"):
            if direction == "in":
                raise CypherSyntaxError("relationship cannot have two arrows")
            direction = "out"
        else:
            self.eat_op("-")
        return A.RelPattern(var, types, props, direction, min_h, max_h, var_len)

    def _map_lit(self) -> A.MapLit:
        self.eat_op("{")
        items = []
        if not self.at_op("}"):
            while True:
                t = self.peek()
                if t.kind in ("IDENT", "STRING"):
                    k = self.next().value
                elif t.kind == "KW":
                    k = self.next().value.lower()
                else:
                    raise CypherSyntaxError(f"bad map key at {t.pos}")
                # dotted keys: OPTIONS {indexConfig: {vector.dimensions: N}}
                while t.kind != "STRING" and self.at_op("."):
                    self.next()
                    k += "." + self.name_part()
                self.eat_op(":")
                items.append((k, self._expr()))
                if not self.try_op(","):
                    break
        self.eat_op("}")
        return A.MapLit(items)

    # ---- expressions (precedence climbing) ----
    def _expr(self):
        return self._or()

    def _or(self):
        e = self._xor()
        while self.at_kw("OR"):
            self.next()
            e = A.BinOp("OR", e, self._xor())
        return e

    def _xor(self):
        e = self._and()
        while self.at_kw("XOR"):
            self.next()
            e = A.BinOp("XOR", e, self._and())
        return e

    def _and(self):
        e = self._not()
        while self.at_kw("AND"):
            self.next()
            e = A.BinOp("AND", e, self._not())
        return e

    def _not(self):
        if self.try_kw("NOT"):
            return A.UnOp("NOT", self._not())
        return self._comparison()

    def _comparison(self):
        e = self._addsub()
        ops = []
        while True:
            t = self.peek()
            if t.kind == "OP" and t.value in ("=", "<>", "!=", "<", ">",
                                              "<=", ">=", "=~"):
                op = self.next().value
                rhs = self._addsub()
                ops.append((op, rhs))
            elif self.at_kw("IN"):
                self.next()
                ops.append(("IN", self._addsub()))
            elif self.at_kw("STARTS"):
                self.next()
                self.eat_kw("WITH")
                ops.append(("STARTS WITH", self._addsub()))
            elif self.at_kw("ENDS"):
                self.next()
                self.eat_kw("WITH")
                ops.append(("ENDS WITH", self._addsub()))
            elif self.at_kw("CONTAINS"):
                self.next()
                ops.append(("CONTAINS", self._addsub()))
            elif self.at_kw("IS"):
                self.next()
                neg = self.try_kw("NOT")
                if self.at_op("::") or self.at_op(":"):
                    if not self.try_op("::"):
                        self.eat_op(":")
                        self.eat_op(":")
                    tname = self.next().value.upper()
                    if tname == "LOCAL" or tname == "ZONED":
                        tname += " " + self.next().value.upper()
                    e = A.TypePredicate(e, tname, negated=neg)
                    continue
                self.eat_kw("NULL")
                e = A.UnOp("IS NOT NULL" if neg else "IS NULL", e)
                continue
            else:
                break
        if not ops:
            return e
        # chained comparisons: a < b < c  ==  a<b AND b<c
        result = None
        left = e
        for op, rhs in ops:
            cmp_ = A.BinOp(op, left, rhs)
            result = cmp_ if result is None else A.BinOp("AND", result, cmp_)
            left = rhs
        return result

    def _addsub(self):
        e = self._muldiv()
        while self.at_op("+", "-"):
            op = self.next().value
            e = A.BinOp(op, e, self._muldiv())
        return e

    def _muldiv(self):
        e = self._power()
        while self.at_op("*", "/", "%"):
            op = self.next().value
            e = A.BinOp(op, e, self._power())
        return e

    def _power(self):
        e = self._unary()
        if self.at_op("^"):
            self.next()
            return A.BinOp("^", e, self._power())
        return e

    def _unary(self):
        if self.at_op("-"):
            self.next()
            return A.UnOp("-", self._unary())
        if self.at_op("+"):
            self.next()
            return self._unary()
        return self._postfix()

    def _postfix(self):
        e = self._atom()
        while True:
            if self.try_op("."):
                e = A.Prop(e, self.name_part())
            elif self.at_op("["):
                self.next()
                if self.try_op(".."):
                    hi = None if self.at_op("]") else self._expr()
                    e = A.Index(e, None, slice=(None, hi))
                else:
                    idx = self._expr()
                    if self.try_op(".."):
                        hi = None if self.at_op("]") else self._expr()
                        e = A.Index(e, None, slice=(idx, hi))
                    else:
                        e = A.Index(e, idx)
                self.eat_op("]")
            elif self.at_op(":") and isinstance(e, (A.Var,)) and self._label_predicate_ok():
                # n:Label predicate inside expressions
                labels = []
                while self.try_op(":"):
                    labels.append(self.label_name())
                e = A.FuncCall("__haslabels", [e, A.Lit(labels)])
            elif self.at_op("{") and isinstance(e, (A.Var, A.Prop)):
                # map projection n {.name, .*, key: expr, var}
                self.next()
                items = []
                while not self.at_op("}"):
                    if self.try_op("."):
                        if self.try_op("*"):
                            items.append(("all",))
                        else:
                            items.append(("prop", self.name_part()))
                    else:
                        name = self.name_part()
                        if self.try_op(":"):
                            items.append(("kv", name, self._expr()))
                        else:
                            items.append(("var", name))
                    if not self.try_op(","):
                        break
                self.eat_op("}")
                e = A.MapProjection(e, items)
            else:
                return e

    def _label_predicate_ok(self):
        # ':' inside expression context means label predicate only when
        # followed by an identifier (avoid map-literal confusion)
        return self.peek(1).kind in ("IDENT",)

    def _atom(self):
        t = self.peek()
        if t.kind == "INT":
            self.next()
            return A.Lit(int(t.value))
        if t.kind == "FLOAT":
            self.next()
            return A.Lit(float(t.value))
        if t.kind == "STRING":
            self.next()
            return A.Lit(t.value)
        if t.kind == "PARAM":
            self.next()
            return A.Param(t.value)
        if t.kind == "KW":
            if t.value == "NULL":
                self.next()
                return A.Lit(None)
            if t.value == "TRUE":
                self.next()
                return A.Lit(True)
            if t.value == "FALSE":
                self.next()
                return A.Lit(False)
            if t.value == "COUNT" and self.peek(1).kind == "OP" and self.peek(1).value == "{":
                self.next()
                return A.SubqueryExpr(
                    "COUNT", self._ensure_return(self._braced_query()))
            if t.value == "COUNT" and self.peek(1).kind == "OP" and self.peek(1).value == "(":
                self.next()
                self.eat_op("(")
                if self.try_op("*"):
                    self.eat_op(")")
                    return A.FuncCall("count", [], star=True)
                distinct = self.try_kw("DISTINCT")
                arg = self._expr()
                self.eat_op(")")
                return A.FuncCall("count", [arg], distinct=distinct)
            if t.value == "CASE":
                return self._case()
            if t.value == "EXISTS":
                self.next()
                if self.at_op("{"):
                    return A.SubqueryExpr(
                        "EXISTS", self._ensure_return(self._braced_query()))
                self.eat_op("(")
                if self.at_op("("):
                    pat = self._pattern_path()
                    self.eat_op(")")
                    return A.PatternPredicate(pat)
                inner = self._expr()
                self.eat_op(")")
                return A.UnOp("IS NOT NULL", inner)
            if t.value in ("ANY", "ALL", "NONE", "SINGLE"):
                kind = t.value
                if self.peek(1).kind == "OP" and self.peek(1).value == "(":
                    self.next()
                    self.eat_op("(")
                    var = self.ident()
                    self.eat_kw("IN")
                    src = self._expr()
                    self.eat_kw("WHERE")
                    wh = self._expr()
                    self.eat_op(")")
                    return A.Quantifier(kind, var, src, wh)
            if t.value in ("COUNT", "ALL", "ANY", "NONE", "SINGLE", "INDEX",
                           "UNIQUE", "DATABASE", "SHOW", "BY", "ON", "END",
                           "START", "CONTAINS"):
                # soft keyword used as a plain variable name
                self.next()
                return A.Var(getattr(t, "raw", "") or t.value.lower())
        if t.kind == "OP" and t.value == "(":
            # pattern predicate like (n)-[:R]->(m) in boolean context
            save = self.i
            try:
                pat = self._pattern_path()
                if len(pat.elements) > 1:
                    return A.PatternPredicate(pat)
                self.i = save
            except CypherSyntaxError:
                self.i = save
            self.next()
            e = self._expr()
            self.eat_op(")")
            return e
        if t.kind == "OP" and t.value == "[":
            self.next()
            # list comprehension? [x IN src WHERE p | proj]
            if (self.peek().kind == "IDENT" and self.peek(1).kind == "KW"
                    and self.peek(1).value == "IN"):
                var = self.next().value
                self.next()  # IN
                src = self._expr()
                where = None
                proj = None
                if self.try_kw("WHERE"):
                    where = self._expr()
                if self.try_op("|"):
                    proj = self._expr()
                self.eat_op("]")
                return A.ListComp(var, src, where, proj)
            # pattern comprehension [(a)-[:R]->(b) WHERE p | proj]
            if self.at_op("("):
                save = self.i
                try:
                    pat = self._pattern_path()
                    if len(pat.elements) > 1:
                        where = self._expr() if self.try_kw("WHERE") else None
                        self.eat_op("|")
                        proj = self._expr()
                        self.eat_op("]")
                        return A.PatternComprehension(pat, where, proj)
                    self.i = save
                except CypherSyntaxError:
                    self.i = save
            items = []
            if not self.at_op("]"):
                items.append(self._expr())
                while self.try_op(","):
                    items.append(self._expr())
            self.eat_op("]")
            return A.ListLit(items)
        if t.kind == "OP" and t.value == "{":
            return self._map_lit()
        if t.kind == "IDENT":
            # trim([BOTH|LEADING|TRAILING] [ch] FROM s) — SQL-style form
            if (t.value.lower() in ("trim", "ltrim", "rtrim", "btrim")
                    and self.peek(1).kind == "OP" and self.peek(1).value == "("
                    and self.peek(2).kind in ("IDENT", "KW")
                    and self.peek(2).value.upper() in ("BOTH", "LEADING",
                                                       "TRAILING")):
                self.next()
                self.eat_op("(")
                spec = self.next().value.upper()
                ch = None
                if not self._at_word("FROM"):
                    ch = self._expr()
                self._eat_word("FROM")
                s = self._expr()
                self.eat_op(")")
                fname = {"BOTH": "btrim", "LEADING": "ltrim",
                         "TRAILING": "rtrim"}[spec]
                args = [s] + ([ch] if ch is not None else [])
                return A.FuncCall(fname, args)
            # reduce(acc = init, x IN list | expr)
            if (t.value.lower() == "reduce" and self.peek(1).kind == "OP"
                    and self.peek(1).value == "("):
                self.next()
                self.eat_op("(")
                acc = self.ident()
                self.eat_op("=")
                init = self._expr()
                self.eat_op(",")
                var = self.ident()
                self.eat_kw("IN")
                source = self._expr()
                self.eat_op("|")
                body = self._expr()
                self.eat_op(")")
                return A.Reduce(acc, init, var, source, body)
            # COLLECT { ... } subquery (Cypher 5; executor returns the
            # first return column per row)
            if (t.value.lower() == "collect" and self.peek(1).kind == "OP"
                    and self.peek(1).value == "{"):
                self.next()
                return A.SubqueryExpr(
                    "COLLECT", self._ensure_return(self._braced_query()))
            # function call?
            if self.peek(1).kind == "OP" and self.peek(1).value == "(":
                name = self.next().value
                while self.try_op("."):
                    name += "." + self.name_part()
                # qualified name may end before '('
                self.eat_op("(")
                distinct = self.try_kw("DISTINCT")
                args = []
                if not self.at_op(")"):
                    args.append(self._expr())
                    while self.try_op(","):
                        args.append(self._expr())
                self.eat_op(")")
                return A.FuncCall(name.lower(), args, distinct=distinct)
            # dotted function name: ns.fn(...)
            if (self.peek(1).kind == "OP" and self.peek(1).value == "."
                    and self._is_dotted_call()):
                name = self.next().value
                while self.try_op("."):
                    name += "." + self.name_part()
                self.eat_op("(")
                distinct = self.try_kw("DISTINCT")
                args = []
                if not self.at_op(")"):
                    args.append(self._expr())
                    while self.try_op(","):
                        args.append(self._expr())
                self.eat_op(")")
                return A.FuncCall(name.lower(), args, distinct=distinct)
            return A.Var(self.next().value)
        raise CypherSyntaxError(f"unexpected token {t.value!r} at {t.pos}")

    def _is_dotted_call(self):
        """Lookahead: IDENT(.IDENT)+( — distinguishes apoc.coll.max(...) from n.prop"""
        j = self.i
        toks = self.toks
        if toks[j].kind != "IDENT":
            return False
        j += 1
        seen_dot = False
        while (j + 1 < len(toks) and toks[j].kind == "OP" and toks[j].value == "."
               and toks[j + 1].kind in ("IDENT", "KW")):
            seen_dot = True
            j += 2
        return seen_dot and j < len(toks) and toks[j].kind == "OP" and toks[j].value == "("

    def _case(self):
        self.eat_kw("CASE")
        test = None
        if not self.at_kw("WHEN"):
            test = self._expr()
        whens = []
        # Extended simple form (Neo4j 5): each WHEN is a comma-separated
        # list of candidates, each either a value (equality) or a
        # comparison applied to the operand. Desugared to general-form
        # boolean conditions over the synthetic variable __case__, which
        # the executor binds to the operand value evaluated EXACTLY ONCE
        # (Neo4j semantics; a non-deterministic operand like rand() must
        # not be re-evaluated per alternative). Null-semantics match
        # Cypher `=` (WHEN null never matches).
        operand = A.Var("__case__") if test is not None else None
        while self.try_kw("WHEN"):
            if test is not None:
                cond = self._case_alt(operand)
                while self.try_op(","):
                    cond = A.BinOp("OR", cond, self._case_alt(operand))
            else:
                cond = self._expr()
            self.eat_kw("THEN")
            whens.append((cond, self._expr()))
        default = None
        if self.try_kw("ELSE"):
            default = self._expr()
        self.eat_kw("END")
        return A.Case(test, whens, default)

    def _case_alt(self, test):
        """One alternative of an extended simple-form CASE WHEN; `test`
        is the synthetic __case__ variable, never the operand AST."""
        t = self.peek()
        if t.kind == "OP" and t.value in ("=", "<>", "!=", "<", ">",
                                          "<=", ">=", "=~"):
            return A.BinOp(self.next().value, test, self._addsub())
        if self.at_kw("IS"):
            self.next()
            neg = self.try_kw("NOT")
            if self.at_op("::") or self.at_op(":"):
                if not self.try_op("::"):
                    self.eat_op(":")
                    self.eat_op(":")
                tname = self.next().value.upper()
                if tname in ("LOCAL", "ZONED"):
                    tname += " " + self.next().value.upper()
                return A.TypePredicate(test, tname, negated=neg)
            self.eat_kw("NULL")
            return A.UnOp("IS NOT NULL" if neg else "IS NULL", test)
        if self.at_kw("STARTS"):
            self.next()
            self.eat_kw("WITH")
            return A.BinOp("STARTS WITH", test, self._addsub())
        if self.at_kw("ENDS"):
            self.next()
            self.eat_kw("WITH")
            return A.BinOp("ENDS WITH", test, self._addsub())
        if self.at_kw("CONTAINS"):
            self.next()
            return A.BinOp("CONTAINS", test, self._addsub())
        if self.at_kw("IN"):
            self.next()
            return A.BinOp("IN", test, self._addsub())
        return A.BinOp("=", test, self._addsub())

    def _expr_atom_chain(self):
        """Left side of SET: variable with optional property chain."""
        e = A.Var(self.ident())
        while self.try_op("."):
            e = A.Prop(e, self.ident())
        return e


def parse(text: str) -> A.Query:
    return Parser(text).parse()
