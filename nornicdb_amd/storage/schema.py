"""Schema management: constraints, property indexes, vector index metadata.

Parity: reference pkg/storage/schema.go + constraint_validation.go
(unique/exists constraints, property indexes, vector index metadata).
Constraints are enforced through engine validators (checked before every
node create/update).
"""

from __future__ import annotations

import json
import os
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
        # schema survives restarts: definitions live in a sidecar
        # schema.json next to the engine's store (reference persists
        # schema in badger, pkg/storage/schema.go); keyed by namespace
        # so every logical database keeps its own schema. Memory engines
        # have no data dir -> memory-only, matching their semantics.
        self._load()
        if hasattr(engine, "add_validator"):
            engine.add_validator(self._validate)

    # ---- persistence ----
    def _schema_file(self):
        eng = self.engine
        ns = getattr(eng, "ns", "") or "default"
        inner = eng
        while hasattr(inner, "inner"):
            inner = inner.inner
        d = getattr(inner, "data_dir", None)
        return (os.path.join(d, "schema.json") if d else None), ns

    def _crypt(self):
        """Engine's at-rest cipher, if any — the sidecar must not leak
        plaintext next to an encrypted store."""
        inner = self.engine
        while hasattr(inner, "inner"):
            inner = inner.inner
        kv = getattr(inner, "_kv", None)
        return getattr(kv, "crypt", None)

    def _read_file(self, path):
        with open(path, "rb") as f:
            raw = f.read()
        crypt = self._crypt()
        if crypt is not None:
            raw = crypt.decrypt(raw, aad=b"schema")
        return json.loads(raw.decode())

    def _write_file(self, path, obj):
        raw = json.dumps(obj).encode()
        crypt = self._crypt()
        if crypt is not None:
            raw = crypt.encrypt(raw, aad=b"schema")
        tmp = path + ".tmp"
        with open(tmp, "wb") as f:
            f.write(raw)
            f.flush()
            os.fsync(f.fileno())
        os.replace(tmp, path)

    def _persist(self):
        path, ns = self._schema_file()
        if path is None:
            return
        try:
            all_ = {}
            if os.path.exists(path):
                all_ = self._read_file(path)
            all_[ns] = {
                "constraints": [[c.name, c.kind, c.label, c.prop]
                                for c in self.constraints.values()],
                "vector_indexes": [[v.name, v.label, v.prop, v.dims,
                                    v.similarity]
                                   for v in self.vector_indexes.values()],
                "named_indexes": [[n, k, lb, list(ps)] for n, (k, lb, ps)
                                  in self.named_indexes.items()],
            }
            self._write_file(path, all_)
        except (OSError, ValueError):
            pass

    def _load(self):
        path, ns = self._schema_file()
        if path is None or not os.path.exists(path):
            return
        try:
            payload = self._read_file(path).get(ns)
            if not payload:
                return
            for nm, kind, lb, pr in payload.get("constraints", []):
                self.constraints[nm] = Constraint(nm, kind, lb, pr)
                if kind == "unique" and hasattr(self.engine,
                                                "create_property_index"):
                    self.engine.create_property_index(lb, pr)
            for nm, lb, pr, dims, sim in payload.get("vector_indexes", []):
                self.vector_indexes[nm] = VectorIndexMeta(nm, lb, pr,
                                                          dims, sim)
            for nm, k, lb, ps in payload.get("named_indexes", []):
                self.named_indexes[nm] = (k, lb, list(ps))
                for pp in ps:
                    self.property_indexes.append((lb, pp))
                    if hasattr(self.engine, "create_property_index"):
                        self.engine.create_property_index(lb, pp)
        except (OSError, ValueError, KeyError):
            pass

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
        self._persist()

    def create_exists_constraint(self, name: str, label: str, prop: str):
        with self._lock:
            for n in self.engine.get_nodes_by_label(label):
                if n.properties.get(prop) is None:
                    raise ConstraintViolation(
                        f"cannot create constraint {name}: node {n.id} "
                        f"missing {label}.{prop}")
            self.constraints[name] = Constraint(name, "exists", label, prop)
        self._persist()

    def drop_constraint(self, name: str):
        with self._lock:
            self.constraints.pop(name, None)
        self._persist()

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
        self._persist()
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
        self._persist()
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
        self._persist()

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
