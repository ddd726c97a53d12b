"""Bolt server tests: packstream round trips and a minimal in-process
Bolt 4.4 client exercising handshake + HELLO/RUN/PULL/RESET over TCP.

Models reference pkg/bolt tests (packstream_bench_test.go, integration_test.go).
"""

import asyncio
import struct

import pytest

from nornicdb_amd.bolt import packstream as ps
from nornicdb_amd.bolt.server import (BOLT_MAGIC, BoltServer, M_HELLO, M_PULL,
                                      M_RUN, R_FAILURE, R_RECORD, R_SUCCESS)
from nornicdb_amd.cypher import Executor
from nornicdb_amd.storage import MemoryEngine


class TestPackStream:
    @pytest.mark.parametrize("v", [
        None, True, False, 0, 1, -1, 127, -16, -17, 128, 32767, -32768,
        2 ** 31, -2 ** 31 - 1, 1.5, -0.25, "", "hello", "x" * 300,
        [1, [2, 3], "a"], {"k": 1, "nested": {"x": [True, None]}},
        b"\x00\x01\x02", list(range(20)),
        {"m": "x" * 70000},
    ])
    def test_roundtrip(self, v):
        assert ps.unpack(ps.pack(v)) == v

    def test_struct_roundtrip(self):
        s = ps.Structure(0x4E, [1, ["A"], {"k": "v"}])
        assert ps.unpack(ps.pack(s)) == s

    def test_int_boundaries(self):
        for v in (-9223372036854775808, 9223372036854775807):
            assert ps.unpack(ps.pack(v)) == v


class _Client:
    """Minimal Bolt 4.4 test client."""

    def __init__(self, reader, writer):
        self.reader = reader
        self.writer = writer

    async def handshake(self):
        self.writer.write(struct.pack(">I", BOLT_MAGIC))
        self.writer.write(bytes([0, 0, 4, 4, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0]))
        await self.writer.drain()
        resp = await self.reader.readexactly(4)
        return resp[3], resp[2]  # major, minor

    async def send(self, tag, *fields):
        data = ps.pack(ps.Structure(tag, list(fields)))
        self.writer.write(struct.pack(">H", len(data)) + data + b"\x00\x00")
        await self.writer.drain()

    async def recv(self):
        buf = bytearray()
        while True:
            size = struct.unpack(">H", await self.reader.readexactly(2))[0]
            if size == 0:
                if buf:
                    return ps.unpack(bytes(buf))
                continue
            buf += await self.reader.readexactly(size)


@pytest.fixture
def bolt_server_port(unused_tcp_port_factory=None):
    return 0  # unused; we pick ephemeral below


async def _start_server():
    eng = MemoryEngine()
    ex = Executor(eng)
    srv = BoltServer(lambda db: ex, host="127.0.0.1", port=0)
    await srv.start()
    port = srv._server.sockets[0].getsockname()[1]
    return srv, port


def test_bolt_end_to_end():
    async def run():
        srv, port = await _start_server()
        try:
            reader, writer = await asyncio.open_connection("127.0.0.1", port)
            c = _Client(reader, writer)
            major, minor = await c.handshake()
            assert (major, minor) == (4, 4)

            await c.send(M_HELLO, {"user_agent": "test/1.0", "scheme": "none"})
            resp = await c.recv()
            assert resp.tag == R_SUCCESS
            assert "NornicDB-AMD" in resp.fields[0]["server"]

            await c.send(M_RUN, "CREATE (n:Person {name: $n}) RETURN n.name",
                         {"n": "Ada"}, {})
            resp = await c.recv()
            assert resp.tag == R_SUCCESS
            assert resp.fields[0]["fields"] == ["n.name"]

            await c.send(M_PULL, {"n": -1})
            rec = await c.recv()
            assert rec.tag == R_RECORD
            assert rec.fields[0] == ["Ada"]
            done = await c.recv()
            assert done.tag == R_SUCCESS
            assert done.fields[0]["stats"]["nodes-created"] == 1

            # node struct returned for entity values
            await c.send(M_RUN, "MATCH (n:Person) RETURN n", {}, {})
            await c.recv()
            await c.send(M_PULL, {"n": -1})
            rec = await c.recv()
            node = rec.fields[0][0]
            assert isinstance(node, ps.Structure) and node.tag == 0x4E
            assert node.fields[1] == ["Person"]
            assert node.fields[2]["name"] == "Ada"
            await c.recv()

            # syntax error -> FAILURE, then RUN ignored until RESET
            await c.send(M_RUN, "MATCH (n RETURN", {}, {})
            fail = await c.recv()
            assert fail.tag == R_FAILURE
            assert "SyntaxError" in fail.fields[0]["code"]
            await c.send(M_RUN, "RETURN 1", {}, {})
            ign = await c.recv()
            assert ign.tag == 0x7E  # IGNORED
            await c.send(0x0F)  # RESET
            ok = await c.recv()
            assert ok.tag == R_SUCCESS
            await c.send(M_RUN, "RETURN 1 AS one", {}, {})
            assert (await c.recv()).tag == R_SUCCESS
            await c.send(M_PULL, {"n": -1})
            assert (await c.recv()).fields[0] == [1]
            await c.recv()

            writer.close()
        finally:
            srv.close()

    asyncio.get_event_loop_policy().new_event_loop().run_until_complete(
        asyncio.wait_for(run(), timeout=15))


def test_bolt_auth_required():
    class Auth:
        def login(self, user, pw):
            if (user, pw) != ("neo4j", "secret"):
                raise PermissionError("bad credentials")

    async def run():
        eng = MemoryEngine()
        ex = Executor(eng)
        srv = BoltServer(lambda db: ex, host="127.0.0.1", port=0,
                         authenticator=Auth())
        await srv.start()
        port = srv._server.sockets[0].getsockname()[1]
        try:
            reader, writer = await asyncio.open_connection("127.0.0.1", port)
            c = _Client(reader, writer)
            await c.handshake()
            await c.send(M_HELLO, {"scheme": "basic", "principal": "neo4j",
                                   "credentials": "wrong"})
            resp = await c.recv()
            assert resp.tag == R_FAILURE
            writer.close()

            reader, writer = await asyncio.open_connection("127.0.0.1", port)
            c = _Client(reader, writer)
            await c.handshake()
            await c.send(M_HELLO, {"scheme": "basic", "principal": "neo4j",
                                   "credentials": "secret"})
            resp = await c.recv()
            assert resp.tag == R_SUCCESS
            writer.close()
        finally:
            srv.close()

    asyncio.get_event_loop_policy().new_event_loop().run_until_complete(
        asyncio.wait_for(run(), timeout=15))


class TestNativeCodec:
    def test_native_pack_matches_python(self):
        if not ps._HAS_NATIVE_PS:
            pytest.skip("native codec not built")
        vals = [None, True, 42, -300, 2 ** 40, 1.5, "hi", b"xy",
                [1, [2], "a"], {"k": {"n": None}},
                ps.Structure(0x4E, [1, ["A"], {"x": 1}]), "y" * 300,
                list(range(20)), {"m": "x" * 70000}]
        for v in vals:
            assert ps.pack(v) == ps.pack_py(v)
            assert ps.unpack_py(ps.pack(v)) == v


class TestTemporalParams:
    """Inbound Bolt temporal structures in parameters convert to cypher
    temporal values (packstream.temporal_from_struct); outbound values
    convert back (temporal_struct) — full driver roundtrip."""

    def test_struct_roundtrips(self):
        from nornicdb_amd.bolt import packstream as ps
        from nornicdb_amd.cypher import temporal as tp
        for v in (tp.make_date("2026-09-12"),
                  tp.make_duration("P1Y2M3DT4H5M6S"),
                  tp.make_datetime("2026-09-12T08:00:00+02:00"),
                  tp.make_datetime("2026-09-12T08:00:00", local=True)):
            s = ps.temporal_struct(v, bolt5=True)
            back = ps.temporal_from_struct(s)
            assert type(back) is type(v)
            assert str(back)[:19] == str(v)[:19]

    def test_bolt4_legacy_datetime(self):
        from nornicdb_amd.bolt import packstream as ps
        from nornicdb_amd.cypher import temporal as tp
        v = tp.make_datetime("2026-09-12T08:00:00+02:00")
        s4 = ps.temporal_struct(v, bolt5=False)
        assert s4.tag == ps.DATETIME_LEGACY_TAG
        assert str(ps.temporal_from_struct(s4))[:19] == str(v)[:19]


def test_causal_bookmarks():
    """Bookmarks are causal tokens (ndb:<db>:<version>), not fabricated
    ids: commits advance a per-db version, the
    autocommit PULL summary carries the bookmark, and RUN waits for (or
    times out on) future versions."""
    async def run():
        srv, port = await _start_server()
        srv.BOOKMARK_WAIT_TIMEOUT = 0.15
        try:
            reader, writer = await asyncio.open_connection("127.0.0.1", port)
            c = _Client(reader, writer)
            await c.handshake()
            await c.send(M_HELLO, {"user_agent": "t", "scheme": "none"})
            await c.recv()

            async def run_pull(q, extra=None):
                await c.send(M_RUN, q, {}, extra or {})
                r1 = await c.recv()
                if r1.tag == R_FAILURE:
                    return r1.fields[0], None, []
                await c.send(M_PULL, {"n": -1})
                recs = []
                while True:
                    m = await c.recv()
                    if m.tag == R_RECORD:
                        recs.append(m.fields[0])
                    else:
                        return None, m.fields[0], recs

            # write advances the version; summary carries the bookmark
            _, s1, _ = await run_pull("CREATE (:B1)")
            bm1 = s1["bookmark"]
            assert bm1.startswith("ndb:neo4j:")
            v1 = int(bm1.split(":")[2])
            assert v1 >= 1
            _, s2, _ = await run_pull("CREATE (:B2)")
            assert int(s2["bookmark"].split(":")[2]) == v1 + 1
            # reads do not advance
            _, s3, _ = await run_pull("RETURN 1")
            assert int(s3["bookmark"].split(":")[2]) == v1 + 1
            # presenting a satisfied bookmark proceeds
            _, s4, recs = await run_pull("MATCH (n:B1) RETURN count(n)",
                                         {"bookmarks": [s2["bookmark"]]})
            assert recs[0][0] == 1
            # a future bookmark times out with a transient error
            fail, _, _ = await run_pull("RETURN 1",
                                        {"bookmarks": ["ndb:neo4j:999999"]})
            assert fail and "Bookmark" in fail["code"]
            await c.send(0x0F)  # RESET
            await c.recv()
        finally:
            srv.close()
    asyncio.run(run())


def test_explicit_tx_commit_bookmark():
    async def run():
        srv, port = await _start_server()
        try:
            reader, writer = await asyncio.open_connection("127.0.0.1", port)
            c = _Client(reader, writer)
            await c.handshake()
            await c.send(M_HELLO, {"user_agent": "t", "scheme": "none"})
            await c.recv()
            await c.send(0x11, {})              # BEGIN
            assert (await c.recv()).tag == R_SUCCESS
            await c.send(M_RUN, "CREATE (:TX1)", {}, {})
            await c.recv()
            await c.send(M_PULL, {"n": -1})
            while (await c.recv()).tag == R_RECORD:
                pass
            await c.send(0x12)                  # COMMIT
            done = await c.recv()
            assert done.tag == R_SUCCESS
            bm = done.fields[0]["bookmark"]
            assert bm.startswith("ndb:neo4j:") and int(bm.split(":")[2]) == 1
        finally:
            srv.close()
    asyncio.run(run())


def test_log_queries_stdout(capsys):
    """--log-queries / cfg.log_queries: every executed query is logged
    with duration and row count (reference pkg/bolt/server.go LogQueries)."""
    from nornicdb_amd.bolt.server import BoltServer
    from nornicdb_amd.cypher import Executor
    from nornicdb_amd.storage.memory import MemoryEngine

    ex = Executor(MemoryEngine())
    srv = BoltServer(lambda db: ex, log_queries=True)
    r = srv.execute("neo4j", "RETURN 1 AS one", {})
    assert r.rows == [[1]]
    out = capsys.readouterr().out
    assert "[query]" in out and "RETURN 1 AS one" in out and "rows=1" in out


def test_explicit_tx_rollback_undoes_writes():
    """BEGIN .. RUN(CREATE) .. ROLLBACK leaves no state (reference
    pkg/cypher/transaction.go handleRollback — real rollback, not
    implicit-apply); COMMIT keeps it; RESET mid-tx also rolls back."""
    import asyncio

    from nornicdb_amd.bolt.server import BoltServer
    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder

    mgr = open_db(embedder=MockEmbedder(8), dims=8)
    srv = BoltServer(lambda db: mgr.get(db).executor,
                     tx_factory=lambda name: mgr.get(name).begin_tx())

    async def run():
        import struct

        from nornicdb_amd.bolt import packstream as ps
        await srv.start()
        r, w = await asyncio.open_connection("127.0.0.1", srv.port)
        w.write(struct.pack(">I", 0x6060B017) + bytes([0, 0, 4, 4]) +
                bytes(12))
        await r.read(4)

        async def send(tag, *fields):
            data = ps.pack(ps.Structure(tag, list(fields)))
            out = b""
            for i in range(0, len(data), 65535):
                chunk = data[i:i + 65535]
                out += struct.pack(">H", len(chunk)) + chunk
            w.write(out + b"\x00\x00")
            await w.drain()

        async def recv():
            buf = b""
            while True:
                hdr = await r.readexactly(2)
                size = struct.unpack(">H", hdr)[0]
                if size == 0:
                    break
                buf += await r.readexactly(size)
            return ps.unpack(buf)

        await send(0x01, {"scheme": "none"})
        await recv()
        # tx 1: create then ROLLBACK
        await send(0x11, {})
        await recv()
        await send(0x10, "CREATE (:TxB {x: 1})", {}, {})
        await recv()
        await send(0x3F, {"n": -1})
        await recv()
        await send(0x13)   # ROLLBACK
        await recv()
        # tx 2: create then COMMIT
        await send(0x11, {})
        await recv()
        await send(0x10, "CREATE (:TxB {x: 2})", {}, {})
        await recv()
        await send(0x3F, {"n": -1})
        await recv()
        await send(0x12)   # COMMIT
        await recv()
        # tx 3: create then RESET (must roll back too)
        await send(0x11, {})
        await recv()
        await send(0x10, "CREATE (:TxB {x: 3})", {}, {})
        await recv()
        await send(0x3F, {"n": -1})
        await recv()
        await send(0x0F)   # RESET
        await recv()
        w.close()
        await srv._server.wait_closed() if False else None

    asyncio.new_event_loop().run_until_complete(run())
    rows = mgr.get().cypher("MATCH (n:TxB) RETURN n.x ORDER BY n.x").rows
    assert rows == [[2]], rows


def test_point_values_over_bolt():
    """RETURN point(...) encodes as Bolt Point2D/Point3D structures and
    inbound point parameters decode back (regression: returning a point
    crashed the session — native packer rejected CypherPoint)."""
    import asyncio
    import struct

    from nornicdb_amd.bolt import packstream as ps
    from nornicdb_amd.bolt.server import BoltServer
    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder

    mgr = open_db(embedder=MockEmbedder(8), dims=8)
    srv = BoltServer(lambda db: mgr.get(db).executor)

    async def run():
        await srv.start()
        r, w = await asyncio.open_connection("127.0.0.1", srv.port)
        w.write(struct.pack(">I", 0x6060B017) + bytes([0, 0, 4, 4]) +
                bytes(12))
        await r.read(4)

        async def send(tag, *fields):
            data = ps.pack(ps.Structure(tag, list(fields)))
            out = b""
            for i in range(0, len(data), 65535):
                ch = data[i:i + 65535]
                out += struct.pack(">H", len(ch)) + ch
            w.write(out + b"\x00\x00")
            await w.drain()

        async def recv():
            buf = b""
            while True:
                size = struct.unpack(">H", await r.readexactly(2))[0]
                if size == 0:
                    break
                buf += await r.readexactly(size)
            return ps.unpack(buf)

        await send(0x01, {"scheme": "none"})
        await recv()
        # round-trip: a Point2D parameter comes back out
        await send(0x10, "RETURN $p AS p, point({longitude: 9.0, "
                         "latitude: 48.0, height: 10.0}) AS g",
                   {"p": ps.Structure(0x58, [7203, 1.5, 2.5])}, {})
        await recv()
        await send(0x3F, {"n": -1})
        rec = await recv()
        assert rec.tag == 0x71, rec
        p, g = rec.fields[0]
        assert isinstance(p, ps.Structure) and p.tag == 0x58
        assert p.fields == [7203, 1.5, 2.5]
        assert isinstance(g, ps.Structure) and g.tag == 0x59
        assert g.fields[0] == 4979 and g.fields[1] == 9.0
        await recv()   # summary
        w.close()

    asyncio.new_event_loop().run_until_complete(run())
