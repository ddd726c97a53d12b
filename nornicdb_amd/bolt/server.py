"""Bolt protocol server (TCP :7687), asyncio.

Parity: reference pkg/bolt/server.go — handshake (:867), message dispatch
HELLO/RUN/PULL/DISCARD/BEGIN/COMMIT/ROLLBACK/RESET/ROUTE (:1033-1736),
bookmarks (:1617), per-database executor routing (:1916). Supports Bolt
4.x (3-field Node structs) and 5.x (element ids).
"""

from __future__ import annotations

import asyncio
import struct
from typing import Any, Callable, Dict, List, Optional

from ..cypher import CypherRuntimeError, CypherSyntaxError
from ..cypher.executor import Path, Result
from ..storage.types import Edge, Node
from . import packstream as ps

BOLT_MAGIC = 0x6060B017

# message tags
M_HELLO = 0x01
M_GOODBYE = 0x02
M_RESET = 0x0F
M_RUN = 0x10
M_BEGIN = 0x11
M_COMMIT = 0x12
M_ROLLBACK = 0x13
M_DISCARD = 0x2F
M_PULL = 0x3F
M_ROUTE = 0x66
M_LOGON = 0x6A
M_LOGOFF = 0x6B
M_TELEMETRY = 0x54

R_SUCCESS = 0x70
R_RECORD = 0x71
R_IGNORED = 0x7E
R_FAILURE = 0x7F

SUPPORTED = {(5, m) for m in range(0, 9)} | {(4, m) for m in range(1, 5)}


class BoltSession:
    _tx_wrote = False

    def __init__(self, server: "BoltServer", reader, writer):
        self.server = server
        self.reader = reader
        self.writer = writer
        self.version = (0, 0)
        self.authenticated = False
        self.db = ""
        self.ids = ps.IdMap()
        self.pending: Optional[Result] = None
        self.pending_pos = 0
        self.failed = False
        self.in_tx = False
        self._txpair = None   # (executor, recorder) when tx_factory is set
        self.tx_queries: List = []

    @property
    def bolt5(self):
        return self.version[0] >= 5

    # ---- wire ----
    async def read_chunked(self) -> Optional[bytes]:
        buf = bytearray()
        while True:
            hdr = await self.reader.readexactly(2)
            size = struct.unpack(">H", hdr)[0]
            if size == 0:
                if buf:
                    return bytes(buf)
                continue  # noop keep-alive
            buf += await self.reader.readexactly(size)

    def write_message(self, tag: int, *fields):
        data = ps.pack(ps.Structure(tag, list(fields)))
        out = bytearray()
        for i in range(0, len(data), 0xFFFF):
            chunk = data[i:i + 0xFFFF]
            out += struct.pack(">H", len(chunk)) + chunk
        out += b"\x00\x00"
        self.writer.write(bytes(out))

    async def handshake(self) -> bool:
        try:
            hdr = await self.reader.readexactly(20)
        except (asyncio.IncompleteReadError, ConnectionError):
            return False
        magic = struct.unpack(">I", hdr[:4])[0]
        if magic != BOLT_MAGIC:
            return False
        chosen = (0, 0)
        for off in range(4, 20, 4):
            _, rng, minor, major = hdr[off:off + 4]
            for m in range(minor, max(minor - rng, 0) - 1, -1):
                if (major, m) in SUPPORTED:
                    chosen = (major, m)
                    break
            if chosen != (0, 0):
                break
        self.version = chosen
        self.writer.write(struct.pack(">I", (chosen[1] << 8) | chosen[0]))
        await self.writer.drain()
        return chosen != (0, 0)

    # ---- conversion ----
    def to_bolt(self, v):
        if isinstance(v, Node):
            return ps.node_struct(v, self.ids, self.bolt5)
        if isinstance(v, Edge):
            return ps.rel_struct(v, self.ids, self.bolt5)
        if isinstance(v, Path):
            return ps.path_struct(v, self.ids, self.bolt5)
        if isinstance(v, list):
            return [self.to_bolt(x) for x in v]
        if isinstance(v, dict):
            return {k: self.to_bolt(x) for k, x in v.items()}
        if isinstance(v, float) and v != v:  # NaN
            return None
        ts = ps.temporal_struct(v, self.bolt5)
        if ts is not None:
            return ts
        pt = ps.point_struct(v)
        if pt is not None:
            return pt
        return v

    # ---- message handlers ----
    async def serve(self):
        if not await self.handshake():
            self.writer.close()
            return
        try:
            while True:
                try:
                    raw = await self.read_chunked()
                except (asyncio.IncompleteReadError, ConnectionError):
                    break
                if raw is None:
                    break
                msg = ps.unpack(raw)
                if not isinstance(msg, ps.Structure):
                    break
                stop = await self.dispatch(msg)
                await self.writer.drain()
                if stop:
                    break
        finally:
            try:
                self.writer.close()
            except Exception:
                pass

    async def dispatch(self, msg: ps.Structure) -> bool:
        tag = msg.tag
        if tag == M_HELLO:
            return self.on_hello(msg.fields[0] if msg.fields else {})
        if tag == M_LOGON:
            return self.on_logon(msg.fields[0] if msg.fields else {})
        if tag == M_LOGOFF:
            self.authenticated = False
            self.write_message(R_SUCCESS, {})
            return False
        if tag == M_GOODBYE:
            return True
        if tag == M_RESET:
            self.failed = False
            self.pending = None
            if self.in_tx and self._txpair is not None:
                self._txpair[1].rollback()
            self._txpair = None
            self.in_tx = False
            self.write_message(R_SUCCESS, {})
            return False
        if tag == M_TELEMETRY:
            self.write_message(R_SUCCESS, {})
            return False
        if self.failed and tag in (M_RUN, M_PULL, M_DISCARD):
            self.write_message(R_IGNORED, {})
            return False
        if tag == M_RUN:
            return await self.on_run(msg.fields)
        if tag == M_PULL:
            return self.on_pull(msg.fields[0] if msg.fields else {})
        if tag == M_DISCARD:
            self.pending = None
            self.write_message(R_SUCCESS, {})
            return False
        if tag == M_BEGIN:
            meta = msg.fields[0] if msg.fields else {}
            self.db = meta.get("db", self.db)
            if not await self.server.await_bookmarks(
                    self.db, meta.get("bookmarks")):
                self.write_message(R_FAILURE, {
                    "code": "Neo.TransientError.Transaction.BookmarkTimeout",
                    "message": "bookmark version not reached"})
                return False
            self.in_tx = True
            self._tx_wrote = False
            self._txpair = self.server.make_tx(self.db)
            self.write_message(R_SUCCESS, {})
            return False
        if tag == M_COMMIT:
            self.in_tx = False
            if self._txpair is not None:
                self._txpair[1].commit()
                self._txpair = None
            if self._tx_wrote:
                self.server.bump_version(self.db)
                self._tx_wrote = False
            self.write_message(R_SUCCESS,
                               {"bookmark": self.server.bookmark(self.db)})
            return False
        if tag == M_ROLLBACK:
            self.in_tx = False
            if self._txpair is not None:
                # undo every applied statement (reference
                # pkg/cypher/transaction.go handleRollback)
                self._txpair[1].rollback()
                self._txpair = None
            self.write_message(R_SUCCESS, {})
            return False
        if tag == M_ROUTE:
            host = f"{self.server.host}:{self.server.port}"
            if self.bolt5:
                rt = {"ttl": 300, "db": self.db or "neo4j", "servers": [
                    {"addresses": [host], "role": "WRITE"},
                    {"addresses": [host], "role": "READ"},
                    {"addresses": [host], "role": "ROUTE"}]}
                self.write_message(R_SUCCESS, {"rt": rt})
            else:
                self.write_message(R_SUCCESS, {"rt": {
                    "ttl": 300, "servers": [
                        {"addresses": [host], "role": "WRITE"},
                        {"addresses": [host], "role": "READ"},
                        {"addresses": [host], "role": "ROUTE"}]}})
            return False
        self.write_message(R_FAILURE, {"code": "Neo.ClientError.Request.Invalid",
                                       "message": f"unknown message 0x{tag:02x}"})
        return False

    def on_hello(self, extra: Dict[str, Any]) -> bool:
        agent = extra.get("user_agent", "")
        creds_inline = "scheme" in extra
        meta = {"server": f"NornicDB-AMD/{self.server.version_str}",
                "connection_id": f"bolt-{id(self):x}"}
        if self.version >= (4, 1):
            meta["hints"] = {}
        if creds_inline or self.version < (5, 1):
            ok, err = self.server.authenticate(extra)
            if not ok:
                self.write_message(R_FAILURE, {
                    "code": "Neo.ClientError.Security.Unauthorized",
                    "message": err or "authentication failure"})
                return True
            self.authenticated = True
        self.write_message(R_SUCCESS, meta)
        return False

    def on_logon(self, auth: Dict[str, Any]) -> bool:
        ok, err = self.server.authenticate(auth)
        if not ok:
            self.write_message(R_FAILURE, {
                "code": "Neo.ClientError.Security.Unauthorized",
                "message": err or "authentication failure"})
            return True
        self.authenticated = True
        self.write_message(R_SUCCESS, {})
        return False

    def from_bolt(self, v):
        """Convert inbound parameter values: temporal structures ->
        cypher temporal values; recurse containers."""
        if isinstance(v, ps.Structure):
            t = ps.temporal_from_struct(v)
            if t is not None:
                return t
            p = ps.point_from_struct(v)
            return p if p is not None else v
        if isinstance(v, list):
            return [self.from_bolt(x) for x in v]
        if isinstance(v, dict):
            return {k: self.from_bolt(x) for k, x in v.items()}
        return v

    async def on_run(self, fields) -> bool:
        query = fields[0] if fields else ""
        params = self.from_bolt(fields[1] if len(fields) > 1 else {})
        extra = fields[2] if len(fields) > 2 else {}
        db = extra.get("db") or self.db
        if not self.in_tx and not await self.server.await_bookmarks(
                db, extra.get("bookmarks")):
            self.failed = True
            self.write_message(R_FAILURE, {
                "code": "Neo.TransientError.Transaction.BookmarkTimeout",
                "message": "bookmark version not reached"})
            return False
        try:
            if self.in_tx and self._txpair is not None:
                result = self._txpair[0].execute(query, params or {})
            else:
                result = self.server.execute(db, query, params or {})
        except (CypherSyntaxError,) as e:
            self.failed = True
            self.write_message(R_FAILURE, {
                "code": "Neo.ClientError.Statement.SyntaxError",
                "message": str(e)})
            return False
        except Exception as e:  # runtime errors
            self.failed = True
            self.write_message(R_FAILURE, {
                "code": "Neo.ClientError.Statement.ExecutionFailed",
                "message": str(e)})
            return False
        self.pending = result
        self.pending_pos = 0
        wrote = any((result.stats or {}).get(k) for k in
                    ("nodes_created", "nodes_deleted", "edges_created",
                     "edges_deleted", "properties_set", "labels_added"))
        if wrote:
            if self.in_tx:
                self._tx_wrote = True
            else:
                self.server.bump_version(db)
        self.write_message(R_SUCCESS, {"fields": result.columns,
                                       "t_first": 0,
                                       "qid": 0})
        return False

    def on_pull(self, extra: Dict[str, Any]) -> bool:
        n = extra.get("n", -1)
        if self.pending is None:
            self.write_message(R_SUCCESS, {"has_more": False})
            return False
        rows = self.pending.rows
        end = len(rows) if n < 0 else min(len(rows), self.pending_pos + n)
        for i in range(self.pending_pos, end):
            self.write_message(R_RECORD, [self.to_bolt(v) for v in rows[i]])
        self.pending_pos = end
        if end >= len(rows):
            stats = self.pending.stats or {}
            counters = {}
            mapping = {"nodes_created": "nodes-created",
                       "nodes_deleted": "nodes-deleted",
                       "edges_created": "relationships-created",
                       "edges_deleted": "relationships-deleted",
                       "properties_set": "properties-set",
                       "labels_added": "labels-added"}
            for k, bk in mapping.items():
                if stats.get(k):
                    counters[bk] = stats[k]
            meta = {"type": "rw" if counters else "r", "t_last": 0,
                    "db": self.db or "neo4j"}
            if counters:
                meta["stats"] = counters
            if not self.in_tx:
                # autocommit summary carries the causal bookmark
                meta["bookmark"] = self.server.bookmark(self.db)
            self.pending = None
            self.write_message(R_SUCCESS, meta)
        else:
            self.write_message(R_SUCCESS, {"has_more": True})
        return False


class BoltServer:
    """TCP server exposing executors per database name."""

    def version_of(self, db: str) -> int:
        return self._versions.get(db or "neo4j", 0)

    def bump_version(self, db: str) -> int:
        db = db or "neo4j"
        self._versions[db] = self._versions.get(db, 0) + 1
        return self._versions[db]

    def bookmark(self, db: str) -> str:
        db = db or "neo4j"
        return f"ndb:{db}:{self.version_of(db)}"

    @staticmethod
    def parse_bookmark(bm: str):
        """-> (db, version) or None for foreign/legacy bookmarks."""
        parts = bm.split(":")
        if len(parts) == 3 and parts[0] == "ndb":
            try:
                return parts[1], int(parts[2])
            except ValueError:
                return None
        return None

    async def await_bookmarks(self, db: str, bookmarks) -> bool:
        """Causal consistency: block until this server has applied at
        least the bookmarked version (satisfied immediately on a
        single instance; bounded wait covers replicated catch-up)."""
        want = 0
        for bm in bookmarks or ():
            parsed = self.parse_bookmark(bm)
            if parsed and parsed[0] == (db or "neo4j"):
                want = max(want, parsed[1])
        if want <= self.version_of(db):
            return True
        deadline = asyncio.get_event_loop().time() + self.BOOKMARK_WAIT_TIMEOUT
        while asyncio.get_event_loop().time() < deadline:
            await asyncio.sleep(0.01)
            if want <= self.version_of(db):
                return True
        return False

    BOOKMARK_WAIT_TIMEOUT = 3.0

    def __init__(self, executor_for_db: Callable[[str], Any],
                 host: str = "127.0.0.1", port: int = 7687,
                 authenticator=None, version_str: str = "0.1.0",
                 ssl_context=None, log_queries: bool = False,
                 tx_factory: Callable[[str], Any] = None):
        # causal bookmarks: per-db monotonically increasing commit
        # version; bookmark = "ndb:<db>:<version>" (reference
        # server.go:1617-1650 bookmark lifecycle)
        self._versions: Dict[str, int] = {}
        self.executor_for_db = executor_for_db
        self.ssl_context = ssl_context
        self.host = host
        self.port = port
        self.authenticator = authenticator
        self.version_str = version_str
        # reference pkg/bolt/server.go:440 LogQueries — stdout query log
        self.log_queries = log_queries
        # tx_factory(db) -> (executor, recorder): real BEGIN/ROLLBACK
        # support (undo-recording); None keeps implicit-apply semantics
        self.tx_factory = tx_factory
        self._server: Optional[asyncio.AbstractServer] = None

    def authenticate(self, auth: Dict[str, Any]):
        if self.authenticator is None:
            return True, None
        scheme = auth.get("scheme", "none")
        if scheme == "none":
            return (False, "credentials required")
        user = auth.get("principal", "")
        pw = auth.get("credentials", "")
        try:
            self.authenticator.login(user, pw)
            return True, None
        except Exception as e:
            return False, str(e)

    def make_tx(self, db: str):
        if self.tx_factory is None:
            return None
        try:
            return self.tx_factory(db or "neo4j")
        except Exception:
            return None

    def execute(self, db: str, query: str, params: Dict[str, Any]):
        ex = self.executor_for_db(db or "neo4j")
        if not self.log_queries:
            return ex.execute(query, params)
        import time as _t
        t0 = _t.perf_counter()
        try:
            r = ex.execute(query, params)
        except Exception as e:
            print(f"[query] db={db or 'neo4j'} FAILED "
                  f"({(_t.perf_counter() - t0) * 1e3:.1f} ms): "
                  f"{query!r} err={e}", flush=True)
            raise
        print(f"[query] db={db or 'neo4j'} "
              f"{(_t.perf_counter() - t0) * 1e3:.1f} ms rows={len(r.rows)}: "
              f"{query!r}", flush=True)
        return r

    async def start(self):
        self._server = await asyncio.start_server(
            self._on_conn, self.host, self.port, ssl=self.ssl_context)
        if self.port == 0 and self._server.sockets:
            self.port = self._server.sockets[0].getsockname()[1]
        return self

    async def _on_conn(self, reader, writer):
        await BoltSession(self, reader, writer).serve()

    async def serve_forever(self):
        await self.start()
        async with self._server:
            await self._server.serve_forever()

    def close(self):
        if self._server:
            self._server.close()
