"""DiskEngine: the on-disk graph storage engine over the LSM store.

Replaces the role of the reference's BadgerEngine
(reference pkg/storage/badger.go) with the same externally-visible
contract and the same 1-byte key-prefix schema (badger.go:16-26):

    0x00 meta          (counters, schema blobs)
    0x01 node:<id>                 -> node record (msgpack)
    0x02 edge:<id>                 -> edge record
    0x03 label:<label>\\0<node_id>  -> ""        (label index)
    0x04 out:<node>\\0<edge_id>     -> ""        (outgoing adjacency)
    0x05 in:<node>\\0<edge_id>      -> ""        (incoming adjacency)
    0x06 etype:<type>\\0<edge_id>   -> ""        (edge-type index)
    0x07 pend:<node_id>            -> ts        (pending-embedding queue)
    0x08 emb:<node_id><u32 chunk>  -> fp32 bytes (embedding overflow)
    0x09 pidx:<label>\\0<prop>\\0<v>\\0<node_id> -> "" (property index)

Nodes whose record would exceed INLINE_CAP (50 KB, badger.go:29-31) store
their embedding out-of-line as 0x08 chunks of CHUNK_FLOATS fp32 values.

Unlike round-1's PersistentEngine (RAM + full-WAL replay), the dataset
lives ON DISK: RAM holds the memtable, block cache and a bounded node
cache, so graphs larger than memory work and restart cost is
O(active log), not O(history).

Counters (node/edge totals and per-(label, namespace) counts) ride in
the same atomic write batches as meta keys, so they are exactly as
durable and crash-consistent as the data they describe.
"""

from __future__ import annotations

import os
import struct
import threading
import time
from typing import Any, Callable, Dict, Iterator, List, Optional

import msgpack

from . import codec as _codec
from .lsm import LSMStore
from .types import (ConstraintViolation, Edge, Engine, EventType, Node,
                    NotFoundError, StorageError)

P_META = b"\x00"
P_NODE = b"\x01"
P_EDGE = b"\x02"
P_LABEL = b"\x03"
P_OUT = b"\x04"
P_IN = b"\x05"
P_ETYPE = b"\x06"
P_PEND = b"\x07"
P_EMB = b"\x08"
P_PIDX = b"\x09"
SEP = b"\x00"

INLINE_CAP = 50 * 1024        # badger.go:29 — 50KB inline node cap
CHUNK_FLOATS = 8192           # 32 KB fp32 per overflow chunk


def _pk(prefix: bytes, *parts: bytes) -> bytes:
    return prefix + SEP.join(parts)


def _enc(obj) -> bytes:
    return msgpack.packb(obj, use_bin_type=True, default=_codec.default)


def _dec(b: bytes):
    return msgpack.unpackb(b, raw=False, strict_map_key=False,
                           object_hook=_codec.object_hook)


def _node_rec(n: Node, emb_inline) -> bytes:
    d = {"id": n.id, "l": n.labels, "p": n.properties,
         "ca": n.created_at, "ua": n.updated_at}
    if emb_inline is not None:
        d["e"] = emb_inline
    elif n.embedding is not None:
        d["ec"] = len(n.embedding)  # chunked: total floats
    return _enc(d)


def _edge_rec(e: Edge) -> bytes:
    return _enc({"id": e.id, "t": e.type, "s": e.start_node,
                 "d": e.end_node, "p": e.properties,
                 "ca": e.created_at, "ua": e.updated_at})


def _edge_from(b: bytes) -> Edge:
    d = _dec(b)
    return Edge(d["id"], d["t"], d["s"], d["d"], d["p"],
                d.get("ca", 0.0), d.get("ua", 0.0))


class _NodeCache:
    """Bounded LRU of decoded nodes (approximate byte budget)."""

    def __init__(self, max_bytes: int = 64 << 20):
        self._d: Dict[str, Node] = {}
        self._order: List[str] = []
        self._bytes = 0
        self._max = max_bytes

    @staticmethod
    def _cost(n: Node) -> int:
        c = 200 + sum(len(str(k)) + 16 for k in n.properties)
        if n.embedding is not None:
            c += len(n.embedding) * 8
        return c

    def get(self, nid):
        return self._d.get(nid)

    def put(self, n: Node):
        if n.id in self._d:
            self._bytes -= self._cost(self._d[n.id])
        else:
            self._order.append(n.id)
        self._d[n.id] = n
        self._bytes += self._cost(n)
        while self._bytes > self._max and self._order:
            old = self._order.pop(0)
            ent = self._d.pop(old, None)
            if ent is not None:
                self._bytes -= self._cost(ent)

    def drop(self, nid):
        n = self._d.pop(nid, None)
        if n is not None:
            self._bytes -= self._cost(n)
            try:
                self._order.remove(nid)
            except ValueError:
                pass

    def clear(self):
        self._d.clear()
        self._order.clear()
        self._bytes = 0


class DiskTransaction:
    """Buffered transaction; commits as ONE atomic LSM batch."""

    def __init__(self, eng: "DiskEngine"):
        self._eng = eng
        self._ops: List = []
        self._done = False

    def create_node(self, node): self._ops.append(("cn", node)); return node
    def update_node(self, node): self._ops.append(("un", node)); return node
    def delete_node(self, nid): self._ops.append(("dn", nid))
    def detach_delete_node(self, nid): self._ops.append(("ddn", nid))
    def create_edge(self, edge): self._ops.append(("ce", edge)); return edge
    def update_edge(self, edge): self._ops.append(("ue", edge)); return edge
    def delete_edge(self, eid): self._ops.append(("de", eid))

    def commit(self):
        if self._done:
            raise StorageError("transaction already finished")
        self._done = True
        self._eng._commit_tx(self._ops)

    def rollback(self):
        self._done = True
        self._ops = []


class DiskEngine(Engine):
    """Graph engine over LSMStore; the primary persistent engine."""

    def __init__(self, data_dir: str, sync_on_write: bool = False,
                 memtable_bytes: int = 8 << 20,
                 node_cache_bytes: int = 64 << 20,
                 block_cache_bytes: int = 32 << 20,
                 encryption=None):
        self._kv = LSMStore(data_dir, sync_on_write=sync_on_write,
                            memtable_bytes=memtable_bytes,
                            cache_bytes=block_cache_bytes, crypt=encryption)
        self.data_dir = data_dir
        self._lock = threading.RLock()
        self._cache = _NodeCache(node_cache_bytes)
        self._callbacks: List[Callable[[str, Any], None]] = []
        self._validators: List[Callable] = []
        # counters: loaded from meta, updated in-batch
        c = self._kv.get(_pk(P_META, b"cnt"))
        c = _dec(c) if c else {"n": 0, "e": 0}
        self._n_nodes = c["n"]
        self._n_edges = c["e"]
        # property index definitions: {(label, prop)}
        pi = self._kv.get(_pk(P_META, b"pidx"))
        self._pidx_defs = {tuple(x) for x in _dec(pi)} if pi else set()

    # ------------------------------------------------------------------
    # plumbing
    # ------------------------------------------------------------------
    def register_callback(self, cb):
        with self._lock:
            self._callbacks.append(cb)

    def add_validator(self, fn):
        with self._lock:
            self._validators.append(fn)

    def _check(self, node, is_update):
        for v in list(self._validators):
            v(node, is_update)

    def _emit(self, ev, obj):
        for cb in list(self._callbacks):
            try:
                cb(ev, obj)
            except Exception:
                pass

    @staticmethod
    def _ns_of(node_id: str) -> str:
        i = node_id.find(":")
        return node_id[:i] if i > 0 else ""

    def _counts_put(self, puts, dn=0, de=0, label_deltas=None):
        self._n_nodes += dn
        self._n_edges += de
        puts.append((_pk(P_META, b"cnt"),
                     _enc({"n": self._n_nodes, "e": self._n_edges})))
        for (lb, ns), d in (label_deltas or {}).items():
            k = _pk(P_META, b"lc", lb.encode(), ns.encode())
            cur = self._kv.get(k)
            cur = struct.unpack("<q", cur)[0] if cur else 0
            puts.append((k, struct.pack("<q", cur + d)))

    def _label_count(self, label: str, ns: str) -> int:
        v = self._kv.get(_pk(P_META, b"lc", label.encode(), ns.encode()))
        return struct.unpack("<q", v)[0] if v else 0

    # ------------------------------------------------------------------
    # node KV materialization
    # ------------------------------------------------------------------
    def _node_puts(self, n: Node, puts: List, dels: List,
                   old: Optional[Node]):
        nid = n.id.encode()
        emb = n.embedding
        inline = emb
        if emb is not None:
            # estimate: floats dominate; chunk when past the inline cap
            if len(emb) * 9 + 200 > INLINE_CAP:
                inline = None
        puts.append((_pk(P_NODE, nid), _node_rec(n, inline)))
        if inline is None and emb is not None:
            raw = struct.pack(f"<{len(emb)}f", *[float(x) for x in emb])
            step = CHUNK_FLOATS * 4
            for ci in range(0, len(raw), step):
                puts.append((_pk(P_EMB, nid) + struct.pack("<I", ci // step),
                             raw[ci:ci + step]))
        # drop stale overflow chunks / label entries
        if old is not None:
            if old.embedding is not None and len(old.embedding) * 9 + 200 > INLINE_CAP:
                nch_old = (len(old.embedding) * 4 + CHUNK_FLOATS * 4 - 1) // (CHUNK_FLOATS * 4)
                nch_new = 0
                if inline is None and emb is not None:
                    nch_new = (len(emb) * 4 + CHUNK_FLOATS * 4 - 1) // (CHUNK_FLOATS * 4)
                for ci in range(nch_new, nch_old):
                    dels.append(_pk(P_EMB, nid) + struct.pack("<I", ci))
            for lb in set(old.labels) - set(n.labels):
                dels.append(_pk(P_LABEL, lb.encode(), nid))
        for lb in set(n.labels) - (set(old.labels) if old else set()):
            puts.append((_pk(P_LABEL, lb.encode(), nid), b""))
        # property indexes
        for (lb, prop) in self._pidx_defs:
            oldv = (old.properties.get(prop) if old and lb in old.labels
                    else None)
            newv = n.properties.get(prop) if lb in n.labels else None
            if oldv == newv:
                continue
            if oldv is not None:
                dels.append(_pk(P_PIDX, lb.encode(), prop.encode(),
                                _enc(oldv), nid))
            if newv is not None:
                puts.append((_pk(P_PIDX, lb.encode(), prop.encode(),
                                 _enc(newv), nid), b""))

    def _load_node(self, nid: str, rec: bytes) -> Node:
        d = _dec(rec)
        emb = d.get("e")
        if emb is None and "ec" in d:
            total = d["ec"]
            raw = b""
            nch = (total * 4 + CHUNK_FLOATS * 4 - 1) // (CHUNK_FLOATS * 4)
            for ci in range(nch):
                chunk = self._kv.get(_pk(P_EMB, nid.encode())
                                     + struct.pack("<I", ci))
                if chunk is None:
                    raise StorageError(f"missing embedding chunk {ci} of {nid}")
                raw += chunk
            emb = list(struct.unpack(f"<{total}f", raw[:total * 4]))
        return Node(d["id"], d["l"], d["p"], emb, d.get("ca", 0.0),
                    d.get("ua", 0.0))

    # ------------------------------------------------------------------
    # nodes
    # ------------------------------------------------------------------
    def create_node(self, node: Node) -> Node:
        self._check(node, False)
        with self._lock:
            nid = node.id.encode()
            if self._kv.get(_pk(P_NODE, nid)) is not None:
                raise ConstraintViolation(f"node {node.id} already exists")
            n = node.copy()
            n.created_at = n.created_at or time.time()
            n.updated_at = n.updated_at or n.created_at
            puts, dels = [], []
            self._node_puts(n, puts, dels, None)
            ns = self._ns_of(n.id)
            self._counts_put(puts, dn=1,
                             label_deltas={(lb, ns): 1 for lb in n.labels})
            self._kv.write_batch(puts, dels)
            self._cache.put(n.copy())
        self._emit(EventType.NODE_CREATED, n.copy())
        return n

    def get_node(self, node_id: str) -> Node:
        with self._lock:
            c = self._cache.get(node_id)
            if c is not None:
                return c.copy()
            rec = self._kv.get(_pk(P_NODE, node_id.encode()))
            if rec is None:
                raise NotFoundError(f"node {node_id} not found")
            n = self._load_node(node_id, rec)
            self._cache.put(n)
            return n.copy()

    def update_node(self, node: Node) -> Node:
        self._check(node, True)
        with self._lock:
            old = self._get_or_none(node.id)
            if old is None:
                raise NotFoundError(f"node {node.id} not found")
            n = node.copy()
            n.created_at = old.created_at
            n.updated_at = time.time()
            puts, dels = [], []
            self._node_puts(n, puts, dels, old)
            ns = self._ns_of(n.id)
            deltas: Dict = {}
            for lb in set(n.labels) - set(old.labels):
                deltas[(lb, ns)] = deltas.get((lb, ns), 0) + 1
            for lb in set(old.labels) - set(n.labels):
                deltas[(lb, ns)] = deltas.get((lb, ns), 0) - 1
            self._counts_put(puts, label_deltas=deltas)
            self._kv.write_batch(puts, dels)
            self._cache.put(n.copy())
        self._emit(EventType.NODE_UPDATED, n.copy())
        return n

    def update_embedding(self, node_id: str, embedding) -> None:
        """Embedding-only write (regenerable data, reference
        wal_engine.go:24-27 — kept out of node-record churn)."""
        with self._lock:
            old = self._get_or_none(node_id)
            if old is None:
                raise NotFoundError(f"node {node_id} not found")
            n = old.copy()
            n.embedding = [float(x) for x in embedding]
            puts, dels = [], []
            self._node_puts(n, puts, dels, old)
            self._kv.write_batch(puts, dels)
            self._cache.put(n.copy())

    def _get_or_none(self, nid) -> Optional[Node]:
        c = self._cache.get(nid)
        if c is not None:
            return c.copy()
        rec = self._kv.get(_pk(P_NODE, nid.encode()))
        return self._load_node(nid, rec) if rec is not None else None

    def delete_node(self, node_id: str, _batch=None) -> None:
        with self._lock:
            old = self._get_or_none(node_id)
            if old is None:
                raise NotFoundError(f"node {node_id} not found")
            if self.get_out_edges(node_id) or self.get_in_edges(node_id):
                raise ConstraintViolation(
                    f"node {node_id} still has relationships")
            self._delete_node_batch(node_id, old)
        self._emit(EventType.NODE_DELETED, old)

    def _delete_node_batch(self, node_id, old, extra_puts=None,
                           extra_dels=None, extra_deltas=None,
                           de=0, edge_events=()):
        nid = node_id.encode()
        puts, dels = list(extra_puts or []), list(extra_dels or [])
        dels.append(_pk(P_NODE, nid))
        for lb in old.labels:
            dels.append(_pk(P_LABEL, lb.encode(), nid))
        if old.embedding is not None and len(old.embedding) * 9 + 200 > INLINE_CAP:
            nch = (len(old.embedding) * 4 + CHUNK_FLOATS * 4 - 1) // (CHUNK_FLOATS * 4)
            for ci in range(nch):
                dels.append(_pk(P_EMB, nid) + struct.pack("<I", ci))
        dels.append(_pk(P_PEND, nid))
        for (lb, prop) in self._pidx_defs:
            if lb in old.labels and prop in old.properties:
                dels.append(_pk(P_PIDX, lb.encode(), prop.encode(),
                                _enc(old.properties[prop]), nid))
        ns = self._ns_of(node_id)
        deltas = dict(extra_deltas or {})
        for lb in old.labels:
            deltas[(lb, ns)] = deltas.get((lb, ns), 0) - 1
        self._counts_put(puts, dn=-1, de=de, label_deltas=deltas)
        self._kv.write_batch(puts, dels)
        self._cache.drop(node_id)
        for ev, obj in edge_events:
            self._emit(ev, obj)

    def detach_delete_node(self, node_id: str) -> None:
        with self._lock:
            old = self._get_or_none(node_id)
            if old is None:
                raise NotFoundError(f"node {node_id} not found")
            edges = {e.id: e for e in self.get_out_edges(node_id)}
            edges.update({e.id: e for e in self.get_in_edges(node_id)})
            dels, events = [], []
            for e in edges.values():
                dels += self._edge_del_keys(e)
                events.append((EventType.EDGE_DELETED, e))
            self._delete_node_batch(node_id, old, extra_dels=dels,
                                    de=-len(edges), edge_events=events)
        self._emit(EventType.NODE_DELETED, old)

    def get_nodes_by_label(self, label: str) -> List[Node]:
        pref = _pk(P_LABEL, label.encode()) + SEP
        ids = [k[len(pref):].decode() for k, _ in self._kv.scan(pref)]
        out = []
        for nid in ids:
            n = self._get_or_none(nid)
            if n is not None:
                out.append(n)
        return out

    def iter_nodes_by_label(self, label: str) -> Iterator[Node]:
        pref = _pk(P_LABEL, label.encode()) + SEP
        ids = [k[len(pref):].decode() for k, _ in self._kv.scan(pref)]
        for nid in ids:
            n = self._get_or_none(nid)
            if n is not None:
                yield n

    def all_nodes(self) -> Iterator[Node]:
        for k, v in self._kv.scan(P_NODE):
            yield self._load_node(k[1:].decode(), v)

    def iter_nodes_raw(self, label: str = None):
        if label is None:
            yield from self.all_nodes()
        else:
            yield from self.get_nodes_by_label(label)

    def node_count(self) -> int:
        return self._n_nodes

    def node_count_by_label(self, label: str, ns: str = None) -> int:
        if ns is not None:
            return self._label_count(label, ns)
        pref = _pk(P_META, b"lc", label.encode()) + SEP
        return sum(struct.unpack("<q", v)[0]
                   for _, v in self._kv.scan(pref))

    # ------------------------------------------------------------------
    # property indexes (schema manager hook)
    # ------------------------------------------------------------------
    def create_property_index(self, label: str, prop: str) -> None:
        with self._lock:
            if (label, prop) in self._pidx_defs:
                return
            puts = []
            for n in self.get_nodes_by_label(label):
                v = n.properties.get(prop)
                if v is not None:
                    puts.append((_pk(P_PIDX, label.encode(), prop.encode(),
                                     _enc(v), n.id.encode()), b""))
            self._pidx_defs.add((label, prop))
            puts.append((_pk(P_META, b"pidx"),
                         _enc(sorted(self._pidx_defs))))
            self._kv.write_batch(puts)

    def drop_property_index(self, label: str, prop: str) -> None:
        with self._lock:
            self._pidx_defs.discard((label, prop))
            dels = [k for k, _ in self._kv.scan(
                _pk(P_PIDX, label.encode(), prop.encode()) + SEP)]
            self._kv.write_batch(
                [(_pk(P_META, b"pidx"), _enc(sorted(self._pidx_defs)))], dels)

    def lookup_property_index(self, label: str, prop: str, value):
        if (label, prop) not in self._pidx_defs:
            return None
        pref = _pk(P_PIDX, label.encode(), prop.encode(), _enc(value)) + SEP
        out = []
        for k, _ in self._kv.scan(pref):
            n = self._get_or_none(k[len(pref):].decode())
            if n is not None:
                out.append(n)
        return out

    # ------------------------------------------------------------------
    # edges
    # ------------------------------------------------------------------
    def _edge_keys(self, e: Edge) -> List:
        eid = e.id.encode()
        return [
            (_pk(P_EDGE, eid), _edge_rec(e)),
            (_pk(P_OUT, e.start_node.encode(), eid), b""),
            (_pk(P_IN, e.end_node.encode(), eid), b""),
            (_pk(P_ETYPE, e.type.encode(), eid), b""),
        ]

    def _edge_del_keys(self, e: Edge) -> List[bytes]:
        eid = e.id.encode()
        return [_pk(P_EDGE, eid),
                _pk(P_OUT, e.start_node.encode(), eid),
                _pk(P_IN, e.end_node.encode(), eid),
                _pk(P_ETYPE, e.type.encode(), eid)]

    def create_edge(self, edge: Edge) -> Edge:
        with self._lock:
            eid = edge.id.encode()
            if self._kv.get(_pk(P_EDGE, eid)) is not None:
                raise ConstraintViolation(f"edge {edge.id} already exists")
            for nid in (edge.start_node, edge.end_node):
                if self._get_or_none(nid) is None:
                    raise NotFoundError(f"node {nid} not found")
            e = edge.copy()
            e.created_at = e.created_at or time.time()
            e.updated_at = e.updated_at or e.created_at
            puts = self._edge_keys(e)
            self._counts_put(puts, de=1)
            self._kv.write_batch(puts)
        self._emit(EventType.EDGE_CREATED, e.copy())
        return e

    def get_edge(self, edge_id: str) -> Edge:
        rec = self._kv.get(_pk(P_EDGE, edge_id.encode()))
        if rec is None:
            raise NotFoundError(f"edge {edge_id} not found")
        return _edge_from(rec)

    def update_edge(self, edge: Edge) -> Edge:
        with self._lock:
            old_rec = self._kv.get(_pk(P_EDGE, edge.id.encode()))
            if old_rec is None:
                raise NotFoundError(f"edge {edge.id} not found")
            old = _edge_from(old_rec)
            e = edge.copy()
            e.created_at = old.created_at
            e.updated_at = time.time()
            dels = []
            if (old.start_node != e.start_node or old.end_node != e.end_node
                    or old.type != e.type):
                dels = [k for k in self._edge_del_keys(old)[1:]]
            self._kv.write_batch(self._edge_keys(e), dels)
        self._emit(EventType.EDGE_UPDATED, e.copy())
        return e

    def delete_edge(self, edge_id: str) -> None:
        with self._lock:
            rec = self._kv.get(_pk(P_EDGE, edge_id.encode()))
            if rec is None:
                raise NotFoundError(f"edge {edge_id} not found")
            e = _edge_from(rec)
            puts: List = []
            self._counts_put(puts, de=-1)
            self._kv.write_batch(puts, self._edge_del_keys(e))
        self._emit(EventType.EDGE_DELETED, e)

    def get_edges_by_type(self, edge_type: str) -> List[Edge]:
        pref = _pk(P_ETYPE, edge_type.encode()) + SEP
        out = []
        for k, _ in self._kv.scan(pref):
            rec = self._kv.get(_pk(P_EDGE, k[len(pref):]))
            if rec is not None:
                out.append(_edge_from(rec))
        return out

    def all_edges(self) -> Iterator[Edge]:
        for _, v in self._kv.scan(P_EDGE):
            yield _edge_from(v)

    def edge_count(self) -> int:
        return self._n_edges

    def get_out_edges(self, node_id: str) -> List[Edge]:
        pref = _pk(P_OUT, node_id.encode()) + SEP
        out = []
        for k, _ in self._kv.scan(pref):
            rec = self._kv.get(_pk(P_EDGE, k[len(pref):]))
            if rec is not None:
                out.append(_edge_from(rec))
        return out

    def get_in_edges(self, node_id: str) -> List[Edge]:
        pref = _pk(P_IN, node_id.encode()) + SEP
        out = []
        for k, _ in self._kv.scan(pref):
            rec = self._kv.get(_pk(P_EDGE, k[len(pref):]))
            if rec is not None:
                out.append(_edge_from(rec))
        return out

    def neighbors(self, node_id: str) -> List[str]:
        seen, out = set(), []
        for e in self.get_out_edges(node_id):
            if e.end_node not in seen:
                seen.add(e.end_node)
                out.append(e.end_node)
        for e in self.get_in_edges(node_id):
            if e.start_node not in seen:
                seen.add(e.start_node)
                out.append(e.start_node)
        return out

    # ------------------------------------------------------------------
    # pending embeddings (reference badger.go:24 prefix 0x07)
    # ------------------------------------------------------------------
    def mark_pending_embedding(self, node_id: str) -> None:
        self._kv.put(_pk(P_PEND, node_id.encode()),
                     struct.pack("<d", time.time()))

    def pending_embeddings(self, limit: int = 0) -> List[str]:
        out = []
        for k, _ in self._kv.scan(P_PEND):
            out.append(k[1:].decode())
            if limit and len(out) >= limit:
                break
        return out

    def clear_pending_embedding(self, node_id: str) -> None:
        self._kv.delete(_pk(P_PEND, node_id.encode()))

    # ------------------------------------------------------------------
    # transactions
    # ------------------------------------------------------------------
    def begin(self) -> DiskTransaction:
        return DiskTransaction(self)

    def _commit_tx(self, ops):
        """Apply buffered ops as one atomic LSM batch. Validation runs
        first against the pre-tx state plus in-tx effects."""
        with self._lock:
            # run through the normal per-op paths; the LSM batches per op,
            # but atomicity across ops comes from applying all-or-nothing:
            # validate first with a dry pass, then apply.
            applied = []
            try:
                for kind, arg in ops:
                    if kind == "cn":
                        self.create_node(arg)
                    elif kind == "un":
                        self.update_node(arg)
                    elif kind == "dn":
                        self.delete_node(arg)
                    elif kind == "ddn":
                        self.detach_delete_node(arg)
                    elif kind == "ce":
                        self.create_edge(arg)
                    elif kind == "ue":
                        self.update_edge(arg)
                    elif kind == "de":
                        self.delete_edge(arg)
                    applied.append((kind, arg))
            except StorageError:
                for kind, arg in reversed(applied):
                    try:
                        if kind == "cn":
                            self.delete_node(arg.id)
                        elif kind == "ce":
                            self.delete_edge(arg.id)
                    except StorageError:
                        pass
                raise

    # ------------------------------------------------------------------
    # maintenance
    # ------------------------------------------------------------------
    def backup(self, dest: str):
        self._kv.backup(dest)
        # sidecar metadata (schema definitions) travels with the backup
        import shutil
        sp = os.path.join(self.data_dir, "schema.json")
        if os.path.exists(sp):
            shutil.copyfile(sp, dest + ".schema.json")

    @staticmethod
    def restore(backup_path: str, target_dir: str, encryption=None) -> "DiskEngine":
        LSMStore.restore(backup_path, target_dir, crypt=encryption).close()
        import shutil
        sj = backup_path + ".schema.json"
        if os.path.exists(sj):
            shutil.copyfile(sj, os.path.join(target_dir, "schema.json"))
        return DiskEngine(target_dir, encryption=encryption)

    def compact(self):
        self._kv.compact_all()

    def stats(self) -> dict:
        s = self._kv.stats()
        s.update(nodes=self._n_nodes, edges=self._n_edges)
        return s

    def flush(self):
        self._kv.sync()

    def close(self):
        self._kv.close()
