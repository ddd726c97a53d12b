"""Embedded database facade.

Parity: reference pkg/nornicdb/db.go — Open (:750), the agent-memory API
Store/Recall/Remember/Link/Neighbors/Forget (:1365-1776), memory tiers
(:157-290), embed queue (embed_queue.go), per-DB search services
(search_services.go), storage->search event wiring (:994-1035) — and
pkg/multidb DatabaseManager (manager.go:117).
"""

from __future__ import annotations

import os
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence

from .cypher import Executor
from .cypher.procedures import build_procedures
from .embed import CachedEmbedder, Embedder, MockEmbedder, chunk_text
from .search import SearchService
from .storage import (AsyncEngine, Engine, MemoryEngine, NamespacedEngine,
                      Node, Edge, NotFoundError, PersistentEngine, new_id)

# memory tiers (reference pkg/decay: episodic 7d / semantic 69d / procedural 693d)
TIER_EPISODIC = "episodic"
TIER_SEMANTIC = "semantic"
TIER_PROCEDURAL = "procedural"
TIERS = (TIER_EPISODIC, TIER_SEMANTIC, TIER_PROCEDURAL)


@dataclass
class Memory:
    id: str = ""
    content: str = ""
    title: str = ""
    memory_type: str = TIER_EPISODIC
    importance: float = 0.5
    tags: List[str] = field(default_factory=list)
    metadata: Dict[str, Any] = field(default_factory=dict)
    created_at: float = 0.0
    access_count: int = 0
    last_accessed: float = 0.0


class EmbedQueue:
    """Background embedding workers (reference pkg/nornicdb/embed_queue.go:
    N pull-based workers over pending_embed; debounced k-means trigger)."""

    def __init__(self, db: "NornicDB", workers: int = 2, batch: int = 16,
                 poll_interval: float = 0.2, recluster_debounce: float = 30.0):
        self.db = db
        self.batch = batch
        self.poll = poll_interval
        self.debounce = recluster_debounce
        self._stop = threading.Event()
        self._threads = [threading.Thread(target=self._loop, daemon=True)
                         for _ in range(workers)]
        self._last_empty = 0.0
        self._pending_recluster = False
        self._lock = threading.Lock()

    def start(self):
        self._started = True
        for t in self._threads:
            t.start()

    def stop(self):
        self._stop.set()
        if getattr(self, "_started", False):
            for t in self._threads:
                t.join(timeout=1)

    def drain(self, timeout: float = 30.0):
        """Synchronously embed everything pending (for tests/CLI)."""
        t0 = time.time()
        while self.db.engine.pending_embeddings(1) and time.time() - t0 < timeout:
            self._process_batch()
        return not self.db.engine.pending_embeddings(1)

    def _loop(self):
        while not self._stop.wait(self.poll):
            try:
                worked = self._process_batch()
                if not worked:
                    with self._lock:
                        if (self._pending_recluster
                                and time.time() - self._last_empty > self.debounce):
                            self._pending_recluster = False
                            self.db.search.recluster()
            except Exception:
                pass

    def _process_batch(self) -> bool:
        ids = self.db.engine.pending_embeddings(self.batch)
        if not ids:
            return False
        texts, keep = [], []
        for nid in ids:
            try:
                node = self.db.engine.get_node(nid)
            except NotFoundError:
                self.db.engine.clear_pending_embedding(nid)
                continue
            text = node.properties.get("content") or node.properties.get("title") or ""
            chunks = chunk_text(text) or [""]
            texts.append(chunks[0])  # head chunk embeds the node itself
            keep.append(nid)
        if not keep:
            return True
        vecs = self.db.embedder.embed_batch(texts)
        for nid, v in zip(keep, vecs):
            try:
                if hasattr(self.db.engine, "update_embedding"):
                    self.db.engine.update_embedding(nid, [float(x) for x in v])
                else:
                    n = self.db.engine.get_node(nid)
                    n.embedding = [float(x) for x in v]
                    self.db.engine.update_node(n)
            except NotFoundError:
                pass
            self.db.engine.clear_pending_embedding(nid)
        with self._lock:
            self._last_empty = time.time()
            self._pending_recluster = True
        return True


class NornicDB:
    """One logical database: engine + executor + search + embed queue."""

    def __init__(self, engine: Engine, name: str = "neo4j",
                 embedder: Embedder = None, dims: int = None,
                 device: str = None, auto_embed: bool = True,
                 embed_workers: int = 0):
        self.name = name
        self.engine = engine
        self.embedder = embedder or MockEmbedder(dims or 64)
        self.dims = dims or self.embedder.dims
        self.search = SearchService(engine, dims=self.dims, device=device,
                                    embedder=self.embedder)
        # reopening a persisted store: rebuild the GPU/embedding indexes
        # from stored embeddings (reference pkg/gpu auto-sync on boot /
        # search_services BuildIndexes) — without this, vector search
        # starts empty after a restart
        try:
            self.search.build_indexes()   # no-op scan on a fresh store
        except Exception:
            pass
        from .apoc import build_apoc_procedures
        from .storage import SchemaManager
        procs = build_procedures(self)
        procs.update(build_apoc_procedures(self))
        self.schema = SchemaManager(engine)
        self.executor = Executor(engine, procedures=procs, schema=self.schema)
        self.executor.current_database = name
        # read-query result cache (reference pkg/cache/query_cache.go),
        # invalidated by any storage write event
        from .utils.cache import QueryCache
        self.query_cache = QueryCache(capacity=512, ttl=60.0)
        engine.register_callback(lambda ev, obj: self.query_cache.invalidate())
        self.auto_embed = auto_embed
        # apoc.trigger registry: fired after mutating cypher statements
        self.triggers: Dict[str, Dict[str, Any]] = {}
        self.embed_queue = EmbedQueue(self, workers=max(embed_workers, 1))
        if embed_workers > 0:
            self.embed_queue.start()
        # cognitive pipeline (reference db.go:957-973 + inference.go OnStore):
        # access tracking feeds co-access/temporal link sources, session
        # boundaries scope evidence counting, query load adapts decay.
        from .cognitive import AccessTracker, InferenceConfig, InferenceEngine
        from .cognitive.patterns import PatternDetector, QueryLoadPredictor
        self.tracker = AccessTracker()
        self.patterns = PatternDetector()
        self.query_load = QueryLoadPredictor()
        self.auto_link = False  # NORNICDB_AUTO_TLP-equivalent toggle
        self.inference = InferenceEngine(
            engine, search_service=self.search, tracker=self.tracker,
            config=InferenceConfig(evidence_required=2))
        # background decay recalculation (reference pkg/decay Manager.Start,
        # hourly ticker) — enabled by the edge_decay feature flag or
        # NORNICDB_DECAY_INTERVAL_S; stopped by close()
        from .cognitive import DecayManager
        self.decay = DecayManager(engine)
        import os as _os
        _di = _os.environ.get("NORNICDB_DECAY_INTERVAL_S")
        if _di:
            self.decay.start(float(_di))

    def begin_tx(self):
        """Explicit transaction: (executor, recorder). Statements run on
        the returned executor apply immediately (read-your-writes) while
        the recorder captures inverses; recorder.rollback() undoes them
        (reference pkg/cypher/transaction.go semantics; isolation is
        read-uncommitted — see storage/txrecorder.py)."""
        from .cypher import Executor as _Ex
        from .storage.txrecorder import TxRecorder
        rec = TxRecorder(self.engine)
        ex = _Ex(rec, procedures=self.executor.procedures,
                 schema=self.schema)
        ex.current_database = self.name
        ex.database_lister = self.executor.database_lister
        ex.database_router = self.executor.database_router
        ex.database_admin = getattr(self.executor, "database_admin", None)
        return ex, rec

    # ---- cypher ----
    # queries containing these are never served from the result cache;
    # "call" is included because procedures may mutate state the storage
    # event stream doesn't see (triggers, schema, periodic jobs, logs)
    _WRITE_KEYWORDS = ("create", "merge", "set ", "delete", "remove", "drop",
                       "detach", "foreach", "call ")

    def cypher(self, query: str, params: Dict[str, Any] = None):
        q = query.strip().lower()
        cacheable = not any(kw in q for kw in self._WRITE_KEYWORDS)
        key = None
        if cacheable:
            try:
                key = (query, repr(sorted((params or {}).items())))
            except Exception:
                key = None
            if key is not None:
                hit = self.query_cache.get(key)
                if hit is not None:
                    return hit
        res = self.executor.execute(query, params)
        if key is not None and not any(res.stats.values()):
            self.query_cache.put(key, res)
        elif not cacheable:
            # schema DDL (CREATE/DROP INDEX...) mutates no engine state, so
            # no storage event fires — invalidate cached reads explicitly
            self.query_cache.invalidate()
        if self.triggers and any(res.stats.values()):
            for name, t in list(self.triggers.items()):
                if t.get("paused"):
                    continue
                try:
                    self.executor.execute(t["statement"], {})
                except Exception:
                    pass
        return res

    execute_cypher = cypher

    # ---- memory API (reference db.go:1365-1776) ----
    def store(self, content: str, title: str = "", memory_type: str = TIER_EPISODIC,
              importance: float = 0.5, tags: Sequence[str] = (),
              metadata: Dict[str, Any] = None, embed: bool = None) -> Memory:
        if memory_type not in TIERS:
            memory_type = TIER_EPISODIC
        mid = new_id("m")
        now = time.time()
        props = {"content": content, "title": title, "importance": importance,
                 "memory_type": memory_type, "tags": list(tags),
                 "access_count": 0, "created_at": now, "last_accessed": now}
        if metadata:
            props["metadata"] = dict(metadata)
        node = Node(id=mid, labels=["Memory", memory_type.capitalize()],
                    properties=props)
        self.engine.create_node(node)
        if embed if embed is not None else self.auto_embed:
            self.engine.mark_pending_embedding(mid)
        self.tracker.record(mid)
        self.patterns.record_access(mid)
        if self.auto_link:
            try:
                self.inference.on_store(node, self.tracker.session_id)
            except Exception:
                pass
        return Memory(id=mid, content=content, title=title,
                      memory_type=memory_type, importance=importance,
                      tags=list(tags), metadata=metadata or {}, created_at=now)

    def recall(self, query: str, limit: int = 10,
               memory_type: str = None) -> List[Memory]:
        self.query_load.record_query()
        qv = self.embedder.embed_query(query)
        labels = [memory_type.capitalize()] if memory_type else ["Memory"]
        res = self.search.search(query=query, query_vec=qv, k=limit, labels=labels)
        out = []
        for r in res:
            m = self._to_memory(r.node)
            self._touch(r.node.id)
            self.tracker.record(r.node.id)
            self.patterns.record_access(r.node.id)
            out.append(m)
        return out

    def remember(self, memory_id: str) -> Memory:
        node = self.engine.get_node(memory_id)
        self._touch(memory_id)
        return self._to_memory(node)

    def link(self, from_id: str, to_id: str, rel_type: str = "RELATES_TO",
             confidence: float = 1.0) -> Edge:
        e = Edge(id=new_id("e"), type=rel_type, start_node=from_id,
                 end_node=to_id, properties={"confidence": confidence})
        return self.engine.create_edge(e)

    def neighbors(self, memory_id: str, depth: int = 1) -> List[Memory]:
        seen = {memory_id}
        frontier = [memory_id]
        out = []
        for _ in range(depth):
            nxt = []
            for nid in frontier:
                for nb in self.engine.neighbors(nid):
                    if nb not in seen:
                        seen.add(nb)
                        nxt.append(nb)
                        try:
                            out.append(self._to_memory(self.engine.get_node(nb)))
                        except NotFoundError:
                            pass
            frontier = nxt
        return out

    def forget(self, memory_id: str) -> bool:
        try:
            self.engine.detach_delete_node(memory_id)
            return True
        except NotFoundError:
            return False

    def _touch(self, node_id: str):
        try:
            n = self.engine.get_node(node_id)
            n.properties["access_count"] = n.properties.get("access_count", 0) + 1
            n.properties["last_accessed"] = time.time()
            self.engine.update_node(n)
        except NotFoundError:
            pass

    @staticmethod
    def _to_memory(node: Node) -> Memory:
        p = node.properties
        return Memory(id=node.id, content=p.get("content", ""),
                      title=p.get("title", ""),
                      memory_type=p.get("memory_type", TIER_EPISODIC),
                      importance=p.get("importance", 0.5),
                      tags=list(p.get("tags", [])),
                      metadata=dict(p.get("metadata", {})),
                      created_at=p.get("created_at", 0.0),
                      access_count=p.get("access_count", 0),
                      last_accessed=p.get("last_accessed", 0.0))

    def close(self):
        self.decay.stop()
        self.embed_queue.stop()
        self.engine.flush()


class DatabaseManager:
    """Multi-database manager over one shared base engine.

    Parity: reference pkg/multidb/manager.go:117 (CREATE/DROP DATABASE,
    aliases, metadata) using NamespacedEngine isolation.
    """

    SYSTEM = "system"
    DEFAULT = "neo4j"

    def __init__(self, base_engine: Engine, embedder: Embedder = None,
                 dims: int = None, device: str = None):
        self._base = base_engine
        self._embedder = embedder
        self._dims = dims
        self._device = device
        self._lock = threading.Lock()
        self._dbs: Dict[str, NornicDB] = {}
        self._aliases: Dict[str, str] = {}
        for name in (self.DEFAULT, self.SYSTEM):
            self._open(name)
        # database registry persists in the system namespace (_Database /
        # _DbAlias nodes — Neo4j keeps db metadata in the system db;
        # reference pkg/multidb/manager.go persists the registry): without
        # this, user-created databases vanish from the manager on restart
        # while their namespaced data stays orphaned on disk
        try:
            sys_eng = self._dbs[self.SYSTEM].engine
            for n in sys_eng.get_nodes_by_label("_Database"):
                nm = n.properties.get("name")
                if nm and nm not in self._dbs:
                    self._open(nm)
            for n in sys_eng.get_nodes_by_label("_DbAlias"):
                a, t = n.properties.get("alias"), n.properties.get("target")
                if a and t:
                    self._aliases[a] = t
        except Exception:
            pass

    def _open(self, name: str) -> NornicDB:
        eng = NamespacedEngine(self._base, name)
        db = NornicDB(eng, name=name, embedder=self._embedder,
                      dims=self._dims, device=self._device)
        db.executor.database_lister = lambda: sorted(self._dbs)

        def _route(target, rest_query, params):
            tdb = self.get(target)
            ex = tdb.executor
            ex.stats = {"nodes_created": 0, "nodes_deleted": 0,
                        "edges_created": 0, "edges_deleted": 0,
                        "properties_set": 0, "labels_added": 0}
            res = ex._run_query(rest_query, params)
            res.stats = dict(ex.stats)
            return res

        db.executor.database_router = _route

        def _admin(cmd):
            from .cypher.executor import Result
            if cmd.op == "create" and cmd.kind == "database":
                try:
                    self.create(cmd.name)
                except ValueError:
                    if not cmd.if_not_exists and not cmd.or_replace:
                        raise
                return Result([], [])
            if cmd.op == "drop" and cmd.kind == "database":
                try:
                    self.drop(cmd.name)
                except (KeyError, ValueError):
                    if not cmd.if_exists:
                        raise
                return Result([], [])
            if cmd.op == "create" and cmd.kind == "composite":
                from .storage import CompositeEngine, NamespacedEngine
                parts = {}
                default = None
                for spec in cmd.props:
                    alias, target = spec.split(":", 1)
                    if target not in self._dbs:
                        self._open(target)
                    parts[alias] = self._dbs[target].engine
                    default = default or alias
                if not parts:
                    raise ValueError("composite database needs constituents")
                eng = CompositeEngine(parts, default)
                self._dbs[cmd.name] = NornicDB(
                    eng, name=cmd.name, embedder=self._embedder,
                    dims=self._dims, device=self._device)
                self._dbs[cmd.name].executor.database_lister = \
                    lambda: sorted(self._dbs)
                return Result([], [])
            if cmd.op == "create" and cmd.kind == "alias":
                if cmd.name in self._aliases and not (cmd.if_not_exists
                                                      or cmd.or_replace):
                    raise ValueError(f"alias {cmd.name!r} exists")
                self._aliases[cmd.name] = cmd.label
                return Result([], [])
            if cmd.op == "drop" and cmd.kind == "alias":
                if cmd.name not in self._aliases and not cmd.if_exists:
                    raise KeyError(cmd.name)
                self._aliases.pop(cmd.name, None)
                return Result([], [])
            if cmd.op == "show":
                return Result(["name", "database"],
                              [[a, t] for a, t in sorted(self._aliases.items())])
            raise ValueError(f"unsupported admin op {cmd.op} {cmd.kind}")

        db.executor.database_admin = _admin
        self._dbs[name] = db
        return db

    def get(self, name: str = None) -> NornicDB:
        name = self._aliases.get(name or self.DEFAULT, name or self.DEFAULT)
        with self._lock:
            db = self._dbs.get(name)
            if db is None:
                raise KeyError(f"database {name} does not exist")
            return db

    def create(self, name: str) -> NornicDB:
        with self._lock:
            if name in self._dbs:
                raise ValueError(f"database {name} already exists")
            db = self._open(name)
        try:
            from .storage.types import Node as _N
            self._dbs[self.SYSTEM].engine.create_node(_N(
                id=f"_db:{name}", labels=["_Database"],
                properties={"name": name}))
        except Exception:
            pass
        return db

    def drop(self, name: str):
        if name in (self.SYSTEM, self.DEFAULT):
            raise ValueError(f"cannot drop {name}")
        with self._lock:
            db = self._dbs.pop(name, None)
            if db is None:
                raise KeyError(f"database {name} does not exist")
            for node in list(db.engine.all_nodes()):
                try:
                    db.engine.detach_delete_node(node.id)
                except NotFoundError:
                    pass
            db.close()
        try:
            self._dbs[self.SYSTEM].engine.delete_node(f"_db:{name}")
        except Exception:
            pass

    def alias(self, alias: str, target: str):
        with self._lock:
            self._aliases[alias] = target
        try:
            from .storage.types import Node as _N
            eng = self._dbs[self.SYSTEM].engine
            n = _N(id=f"_dbalias:{alias}", labels=["_DbAlias"],
                   properties={"alias": alias, "target": target})
            try:
                eng.update_node(n)
            except Exception:
                eng.create_node(n)
        except Exception:
            pass

    def list(self) -> List[str]:
        with self._lock:
            return sorted(self._dbs)

    def close(self):
        for db in self._dbs.values():
            db.close()
        self._base.close()


def open_db(data_dir: Optional[str] = None, embedder: Embedder = None,
            dims: int = None, device: str = None, durable_sync: bool = False,
            engine: str = "disk", encryption_passphrase: str = "",
            **kw) -> DatabaseManager:
    """Open a NornicDB instance (reference nornicdb.Open, db.go:750).

    data_dir=None -> in-memory. Otherwise `engine` picks the stack:
      "disk" (default) -> DiskEngine: LSM on-disk store (Badger-parity;
              datasets may exceed RAM, restart cost O(active log));
      "wal"  -> PersistentEngine: RAM + WAL + snapshots (round-1 engine,
              fastest when the working set fits memory).
    encryption_passphrase != "" seals every block/log/backup on disk
    (reference pkg/nornicdb/db.go:775-808 Badger at-rest encryption).
    """
    if data_dir:
        crypt = None
        if encryption_passphrase:
            from .utils.encryption import EncryptionManager
            import hashlib
            import os as _os
            # deterministic per-datadir salt, persisted next to the store
            _os.makedirs(data_dir, exist_ok=True)
            salt_p = _os.path.join(data_dir, "SALT")
            if _os.path.exists(salt_p):
                salt = open(salt_p, "rb").read()
            else:
                salt = _os.urandom(16)
                with open(salt_p, "wb") as f:
                    f.write(salt)
            crypt = EncryptionManager(encryption_passphrase, salt=salt)
        if engine == "wal":
            if crypt is not None:
                raise ValueError(
                    "encryption_passphrase requires the disk engine")
            base = PersistentEngine(data_dir, sync_on_write=durable_sync)
        else:
            from .storage import DiskEngine
            base = DiskEngine(data_dir, sync_on_write=durable_sync,
                              encryption=crypt)
    else:
        base = MemoryEngine()
    return DatabaseManager(base, embedder=embedder, dims=dims, device=device)
