"""Predicate/expression compiler: Cypher AST -> python lambda.

Replaces per-row interpreted tree-walks for hot WHERE filters with one
compiled callable (the rebuild's counterpart of the reference's
storage_fastpaths.go / parallel.go filter pool: same goal — make full
scans cheap — achieved by compilation instead of goroutines).

Only the pure, side-effect-free core is compiled (literals, params,
variables, property access, arithmetic, comparisons, boolean 3VL, IN,
string operators, IS NULL). Anything else returns None and the caller
falls back to the interpreter.
"""

from __future__ import annotations

import re
from typing import Callable, Dict, Optional

from ..storage.types import Edge, Node
from . import ast as A

# ---------------------------------------------------------------- runtime
def _prop(base, key):
    if base is None:
        return None
    if isinstance(base, (Node, Edge)):
        return base.properties.get(key)
    if isinstance(base, dict):
        return base.get(key)
    return None


def _cy_eq(l, r):
    if l is None or r is None:
        return None
    if isinstance(l, bool) != isinstance(r, bool):
        return False
    try:
        return bool(l == r)
    except Exception:
        return False


def _cy_ne(l, r):
    e = _cy_eq(l, r)
    return None if e is None else (not e)


def _cmp(op):
    def f(l, r):
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
    return f


_lt, _gt, _le, _ge = _cmp("<"), _cmp(">"), _cmp("<="), _cmp(">=")


def _and(a, b):
    if a is False or b is False:
        return False
    if a is None or b is None:
        return None
    return True


def _or(a, b):
    if a is True or b is True:
        return True
    if a is None or b is None:
        return None
    return False


def _not(a):
    return None if a is None else (not a)


def _in(l, r):
    if l is None or r is None:
        return None
    return any(_cy_eq(l, x) is True for x in r)


def _arith(op):
    def f(l, r):
        if l is None or r is None:
            return None
        try:
            if op == "+":
                if isinstance(l, str) or isinstance(r, str):
                    return str(l) + str(r) if not (isinstance(l, bool) or isinstance(r, bool)) else None
                return l + r
            if op == "-":
                return l - r
            if op == "*":
                return l * r
            if op == "/":
                if r == 0:
                    return None
                v = l / r
                return int(v) if isinstance(l, int) and isinstance(r, int) else v
            if op == "%":
                return l % r if r != 0 else None
        except TypeError:
            return None
    return f


_add, _sub, _mul, _div, _mod = (_arith(o) for o in "+-*/%")


def _starts(l, r):
    if l is None or r is None:
        return None
    return isinstance(l, str) and l.startswith(r)


def _ends(l, r):
    if l is None or r is None:
        return None
    return isinstance(l, str) and l.endswith(r)


def _contains(l, r):
    if l is None or r is None:
        return None
    return isinstance(l, str) and r in l


def _regex(l, r):
    if l is None or r is None:
        return None
    flags = 0
    if r.startswith("(?i)"):
        flags, r = re.IGNORECASE, r[4:]
    return re.fullmatch(r, l, flags) is not None


_NS = {"_prop": _prop, "_cy_eq": _cy_eq, "_cy_ne": _cy_ne, "_lt": _lt,
       "_gt": _gt, "_le": _le, "_ge": _ge, "_and": _and, "_or": _or,
       "_not": _not, "_in": _in, "_add": _add, "_sub": _sub, "_mul": _mul,
       "_div": _div, "_mod": _mod, "_starts": _starts, "_ends": _ends,
       "_contains": _contains, "_regex": _regex}

_BIN = {"=": "_cy_eq", "<>": "_cy_ne", "<": "_lt", ">": "_gt", "<=": "_le",
        ">=": "_ge", "AND": "_and", "OR": "_or", "IN": "_in", "+": "_add",
        "-": "_sub", "*": "_mul", "/": "_div", "%": "_mod",
        "STARTS WITH": "_starts", "ENDS WITH": "_ends",
        "CONTAINS": "_contains", "=~": "_regex"}


def _gen(e, bind_var=None) -> Optional[str]:
    if isinstance(e, A.Lit):
        return repr(e.value)
    if isinstance(e, A.Param):
        return f"P[{e.name!r}]"
    if isinstance(e, A.Var):
        if bind_var is not None:
            return "N" if e.name == bind_var else None
        return f"R[{e.name!r}]"
    if isinstance(e, A.Prop):
        if (bind_var is not None and isinstance(e.expr, A.Var)
                and e.expr.name == bind_var):
            return f"N.properties.get({e.key!r})"
        base = _gen(e.expr, bind_var)
        return None if base is None else f"_prop({base}, {e.key!r})"
    if isinstance(e, A.UnOp):
        inner = _gen(e.expr, bind_var)
        if inner is None:
            return None
        if e.op == "NOT":
            return f"_not({inner})"
        if e.op == "-":
            return f"_sub(0, {inner})"
        if e.op == "IS NULL":
            return f"(({inner}) is None)"
        if e.op == "IS NOT NULL":
            return f"(({inner}) is not None)"
        return None
    if isinstance(e, A.BinOp):
        fn = _BIN.get(e.op)
        if fn is None:
            return None
        l, r = _gen(e.left, bind_var), _gen(e.right, bind_var)
        if l is None or r is None:
            return None
        return f"{fn}({l}, {r})"
    if isinstance(e, A.ListLit):
        parts = [_gen(x, bind_var) for x in e.items]
        if any(p is None for p in parts):
            return None
        return "[" + ", ".join(parts) + "]"
    return None


def compile_predicate(expr) -> Optional[Callable]:
    """Returns f(row_dict, params) -> True/False/None, or None if the
    expression uses features outside the compiled core."""
    cached = getattr(expr, "_compiled", "unset")
    if cached != "unset":
        return cached
    src = _gen(expr)
    fn = None
    if src is not None:
        try:
            fn = eval(f"lambda R, P: {src}", dict(_NS))  # noqa: S307
        except SyntaxError:
            fn = None
    try:
        expr._compiled = fn
    except Exception:
        pass
    return fn


def compile_node_predicate(expr, var: str) -> Optional[Callable]:
    """Specialized form: f(node, params) with the single pattern variable
    bound directly to the node (no row dict) — used by raw label scans."""
    key = f"_compiled_n_{var}"
    cached = getattr(expr, key, "unset")
    if cached != "unset":
        return cached
    src = _gen(expr, bind_var=var)
    fn = None
    if src is not None:
        try:
            fn = eval(f"lambda N, P: {src}", dict(_NS))  # noqa: S307
        except SyntaxError:
            fn = None
    try:
        setattr(expr, key, fn)
    except Exception:
        pass
    return fn
