"""Cypher AST node definitions."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple


# ---- expressions ----
@dataclass
class Lit:
    value: Any


@dataclass
class Param:
    name: str


@dataclass
class Var:
    name: str


@dataclass
class Prop:
    expr: Any
    key: str


@dataclass
class BinOp:
    op: str
    left: Any
    right: Any


@dataclass
class UnOp:
    op: str
    expr: Any


@dataclass
class FuncCall:
    name: str
    args: List[Any]
    distinct: bool = False
    star: bool = False  # count(*)


@dataclass
class ListLit:
    items: List[Any]


@dataclass
class MapLit:
    items: List[Tuple[str, Any]]


@dataclass
class Index:
    expr: Any
    index: Any           # single index
    slice: Optional[Tuple[Any, Any]] = None  # (start, end)


@dataclass
class Case:
    test: Optional[Any]                 # simple CASE operand or None
    whens: List[Tuple[Any, Any]]
    default: Optional[Any]


@dataclass
class ListComp:
    var: str
    source: Any
    where: Optional[Any]
    projection: Optional[Any]


@dataclass
class Quantifier:  # ANY/ALL/NONE/SINGLE(x IN list WHERE p)
    kind: str
    var: str
    source: Any
    where: Any


@dataclass
class PatternPredicate:  # EXISTS((n)-[:R]->()) / bare pattern in WHERE
    pattern: "PatternPath"


@dataclass
class CountSubquery:
    pattern: "PatternPath"
    where: Optional[Any] = None


@dataclass
class MapProjection:
    """n {.name, .age, .*, extra: expr} (Cypher map projection)."""
    expr: Any
    items: List[Any]   # ("prop", name) | ("all",) | ("kv", key, expr) | ("var", name)


@dataclass
class PatternComprehension:
    """[(a)-[:R]->(b) WHERE pred | proj]"""
    pattern: "PatternPath"
    where: Optional[Any]
    proj: Any


@dataclass
class Reduce:
    """reduce(acc = init, x IN list | expr)"""
    acc: str
    init: Any
    var: str
    source: Any
    expr: Any


@dataclass
class TypePredicate:
    """expr IS [NOT] :: TYPE"""
    expr: Any
    type_name: str
    negated: bool = False


@dataclass
class SubqueryExpr:
    """EXISTS { ... } / COUNT { ... } expression subquery (Cypher 5)."""
    kind: str              # "EXISTS" | "COUNT" | "COLLECT"
    query: "Query"


# ---- patterns ----
@dataclass
class NodePattern:
    var: Optional[str]
    labels: List[str]
    props: Optional[Any]  # MapLit / Param
    or_labels: bool = False        # :A|B -> match ANY of labels
    where: Optional[Any] = None    # inline (n:L WHERE expr)


@dataclass
class RelPattern:
    var: Optional[str]
    types: List[str]
    props: Optional[Any]
    direction: str        # "out", "in", "both"
    min_hops: int = 1
    max_hops: int = 1
    var_length: bool = False


@dataclass
class PatternPath:
    elements: List[Any]   # alternating NodePattern, RelPattern
    var: Optional[str] = None  # path variable p = (...)


# ---- clauses ----
@dataclass
class MatchClause:
    patterns: List[PatternPath]
    optional: bool = False
    where: Optional[Any] = None


@dataclass
class CreateClause:
    patterns: List[PatternPath]


@dataclass
class MergeClause:
    pattern: PatternPath
    on_create: List["SetItem"] = field(default_factory=list)
    on_match: List["SetItem"] = field(default_factory=list)


@dataclass
class SetItem:
    target: Any           # Prop or Var
    value: Any
    op: str = "="         # "=", "+=", "label"
    labels: List[str] = field(default_factory=list)


@dataclass
class SetClause:
    items: List[SetItem]


@dataclass
class RemoveClause:
    items: List[Any]      # Prop / (Var, labels)


@dataclass
class DeleteClause:
    exprs: List[Any]
    detach: bool = False


@dataclass
class ReturnItem:
    expr: Any
    alias: Optional[str]


@dataclass
class ReturnClause:
    items: List[ReturnItem]
    star: bool = False
    distinct: bool = False
    order_by: List[Tuple[Any, bool]] = field(default_factory=list)  # (expr, asc)
    skip: Optional[Any] = None
    limit: Optional[Any] = None


@dataclass
class WithClause(ReturnClause):
    where: Optional[Any] = None


@dataclass
class UnwindClause:
    expr: Any
    alias: str
    where: Optional[Any] = None  # reference allows UNWIND ... AS x WHERE p


@dataclass
class CallClause:
    proc: str
    args: List[Any]
    yields: List[Tuple[str, Optional[str]]]  # (name, alias); ("*", None) = all
    where: Optional[Any] = None
    limit: Optional[Any] = None


@dataclass
class UseClause:
    """USE <database> — multi-db routing (executed by DatabaseManager)."""
    database: str


@dataclass
class SchemaCommand:
    """Schema DDL: CREATE/DROP INDEX|CONSTRAINT, SHOW INDEXES|CONSTRAINTS|
    DATABASES|PROCEDURES|FUNCTIONS (Neo4j 4/5 syntax + 3.x legacy)."""
    op: str                      # "create" | "drop" | "show"
    kind: str                    # "index" | "vector" | "fulltext" | "constraint"
                                 # | "indexes" | "constraints" | "databases"
                                 # | "procedures" | "functions"
    name: Optional[str] = None
    label: Optional[str] = None
    props: List[str] = field(default_factory=list)
    constraint_kind: Optional[str] = None   # "unique" | "exists" | "node_key"
    options: Optional[Any] = None           # MapLit AST for OPTIONS {...}
    if_not_exists: bool = False
    type_filter: Optional[str] = None       # SHOW VECTOR|FULLTEXT|... INDEXES
    if_exists: bool = False
    or_replace: bool = False


@dataclass
class SubqueryCallClause:
    """CALL { ... } [IN TRANSACTIONS [OF n ROWS]] clause subquery."""
    query: "Query"
    in_transactions: bool = False
    rows_per_tx: int = 1000


@dataclass
class ForeachClause:
    var: str
    source: Any
    updates: List[Any]


@dataclass
class Query:
    clauses: List[Any]
    union: Optional[Tuple[str, "Query"]] = None  # ("UNION"/"UNION ALL", next)
    explain: bool = False
    profile: bool = False
