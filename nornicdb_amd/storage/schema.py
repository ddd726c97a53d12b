"""Schema management: constraints, property indexes, vector index metadata.

Parity: reference pkg/storage/schema.go + constraint_validation.go
(unique/exists constraints, property indexes, vector index metadata).
Constraints are enforced through engine validators (checked before every
node create/update).
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from .types import ConstraintViolation, Engine, Node


@dataclass
class Constraint:
    name: str
    kind: str          # "unique" | "exists"
    label: str
    prop: str


@dataclass
class VectorIndexMeta:
    name: str
    label: str
    prop: str
    dims: int
    similarity: str = "cosine"


class SchemaManager:
    def __init__(self, engine: Engine):
        self.engine = engine
        self._lock = threading.Lock()
        self.constraints: Dict[str, Constraint] = {}
        self.vector_indexes: Dict[str, VectorIndexMeta] = {}
        self.property_indexes: List[tuple] = []
        self.named_indexes: Dict[str, tuple] = {}  # name -> (kind, label, props)
        if hasattr(engine, "add_validator"):
            engine.add_validator(self._validate)

    # ---- constraints ----
    def create_unique_constraint(self, name: str, label: str, prop: str):
        with self._lock:
            # validate existing data first (reference behavior)
            seen = {}
            for n in self.engine.get_nodes_by_label(label):
                v = n.properties.get(prop)
                if v is None:
                    continue
                key = repr(v)
                if key in seen:
                    raise ConstraintViolation(
                        f"cannot create constraint {name}: duplicate "
                        f"{label}.{prop}={v!r} (nodes {seen[key]}, {n.id})")
                seen[key] = n.id
            self.constraints[name] = Constraint(name, "unique", label, prop)
        # back the constraint with an exact index
        if hasattr(self.engine, "create_property_index"):
            self.engine.create_property_index(label, prop)

    def create_exists_constraint(self, name: str, label: str, prop: str):
        with self._lock:
            for n in self.engine.get_nodes_by_label(label):
                if n.properties.get(prop) is None:
                    raise ConstraintViolation(
                        f"cannot create constraint {name}: node {n.id} "
                        f"missing {label}.{prop}")
            self.constraints[name] = Constraint(name, "exists", label, prop)

    def drop_constraint(self, name: str):
        with self._lock:
            self.constraints.pop(name, None)

    def list_constraints(self) -> List[Constraint]:
        with self._lock:
            return list(self.constraints.values())

    # ---- indexes ----
    def create_index(self, label: str, prop: str, name: str = None,
                     kind: str = "range", props: list = None):
        """Register a property index (optionally named, e.g. from Cypher
        CREATE INDEX). Multi-property indexes index each prop."""
        plist = props or [prop]
        nm = name or f"index_{label}_{'_'.join(plist)}"
        with self._lock:
            self.named_indexes[nm] = (kind, label, list(plist))
        for p in plist:
            self.property_indexes.append((label, p))
            if hasattr(self.engine, "create_property_index"):
                self.engine.create_property_index(label, p)
        return nm

    def drop_index(self, name: str) -> bool:
        with self._lock:
            meta = self.named_indexes.pop(name, None)
            if meta is None:
                return self.vector_indexes.pop(name, None) is not None
        kind, label, plist = meta
        for p in plist:
            try:
                self.property_indexes.remove((label, p))
            except ValueError:
                pass
            if hasattr(self.engine, "drop_property_index"):
                self.engine.drop_property_index(label, p)
        return True

    def list_indexes(self):
        """[(name, kind, label, props)] incl. vector indexes."""
        with self._lock:
            out = [(n, k, lb, ps) for n, (k, lb, ps) in
                   self.named_indexes.items()]
            out += [(v.name, "VECTOR", v.label, [v.prop])
                    for v in self.vector_indexes.values()]
        return out

    def create_vector_index(self, name: str, label: str, prop: str,
                            dims: int, similarity: str = "cosine"):
        with self._lock:
            self.vector_indexes[name] = VectorIndexMeta(name, label, prop,
                                                        dims, similarity)

    # ---- enforcement hook ----
    def _validate(self, node: Node, is_update: bool):
        for c in self.constraints.values():
            if c.label not in node.labels:
                continue
            v = node.properties.get(c.prop)
            if c.kind == "exists" and v is None:
                raise ConstraintViolation(
                    f"constraint {c.name}: {c.label}.{c.prop} must exist")
            if c.kind == "unique" and v is not None:
                lookup = getattr(self.engine, "lookup_property_index", None)
                dupes = lookup(c.label, c.prop, v) if lookup else None
                if dupes is None:
                    dupes = [n for n in self.engine.get_nodes_by_label(c.label)
                             if n.properties.get(c.prop) == v]
                for d in dupes:
                    if d.id != node.id:
                        raise ConstraintViolation(
                            f"constraint {c.name}: duplicate "
                            f"{c.label}.{c.prop}={v!r} (node {d.id})")
