"""Cross-protocol throughput benchmark (loopback).

Mirrors the reference's testing/e2e/endpoints_bench_test.go: the same
indexed read query over Bolt (real TCP, sync clients in threads),
Neo4j HTTP tx + GraphQL (real uvicorn server), and the embedded API.
Reference comparison points (Windows, concurrency 16):
Bolt 2489 ops/s p50 0.53 ms | HTTP 4082 ops/s | GraphQL 3200 ops/s.

Usage: python scripts/bench_protocols.py [--n 2000] [--concurrency 16]
"""
import argparse
import asyncio
import os
import socket
import struct
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

Q = "MATCH (n:Person {name: $n}) RETURN n.name, n.age"


def pct(lat, p):
    s = sorted(lat)
    return s[min(int(p * len(s)), len(s) - 1)] * 1000


class BoltClient:
    def __init__(self, port):
        self.s = socket.create_connection(("127.0.0.1", port))
        self.s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        self.buf = b""
        from nornicdb_amd.bolt import packstream as ps
        self.ps = ps
        self.s.sendall(struct.pack(">I", 0x6060B017) + bytes([0, 0, 4, 4]) + bytes(12))
        self.s.recv(4)
        self.send(0x01, {"scheme": "none"})
        self.recv()

    def send(self, tag, *fields):
        data = self.ps.pack(self.ps.Structure(tag, list(fields)))
        self.s.sendall(struct.pack(">H", len(data)) + data + b"\x00\x00")

    def recv(self):
        msg = b""
        while True:
            while len(self.buf) < 2:
                self.buf += self.s.recv(65536)
            size = struct.unpack(">H", self.buf[:2])[0]
            self.buf = self.buf[2:]
            if size == 0:
                if msg:
                    return self.ps.unpack(msg)
                continue
            while len(self.buf) < size:
                self.buf += self.s.recv(65536)
            msg += self.buf[:size]
            self.buf = self.buf[size:]

    def run(self, query, params):
        self.send(0x10, query, params, {})
        self.send(0x3F, {"n": -1})
        self.recv()
        while self.recv().tag != 0x70:
            pass


def _bolt_proc(port, per, q):
    c = BoltClient(port)
    lat = []
    for i in range(per):
        t = time.perf_counter()
        c.run(Q, {"n": f"p{i % 100}"})
        lat.append(time.perf_counter() - t)
    q.put(lat)


def _http_post_loop(port, per, path, payload_fn):
    import http.client
    import json as J
    conn = http.client.HTTPConnection("127.0.0.1", port)
    lat = []
    for i in range(per):
        t = time.perf_counter()
        conn.request("POST", path, body=J.dumps(payload_fn(i)),
                     headers={"Content-Type": "application/json"})
        resp = conn.getresponse()
        resp.read()
        assert resp.status == 200
        lat.append(time.perf_counter() - t)
    return lat


def _http_tx_proc(port, per, q):
    q.put(_http_post_loop(port, per, "/db/neo4j/tx/commit", lambda i: {
        "statements": [{"statement": Q, "parameters": {"n": f"p{i % 100}"}}]}))


def _graphql_proc(port, per, q):
    q.put(_http_post_loop(port, per, "/graphql", lambda i: {
        "query": '{ nodes(label: "Person", limit: 1) { id properties } }'}))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=2000)
    ap.add_argument("--concurrency", type=int, default=16)
    args = ap.parse_args()

    import uvicorn
    from nornicdb_amd.bolt import BoltServer
    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder
    from nornicdb_amd.server import create_app

    mgr = open_db(embedder=MockEmbedder(16), dims=16)
    db = mgr.get()
    for i in range(100):
        db.cypher("CREATE (:Person {name: $n, age: $a})",
                  {"n": f"p{i}", "a": 20 + i % 50})
    db.engine.create_property_index("Person", "name")

    # servers in background threads
    ports = {}

    def bolt_thread():
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)
        srv = BoltServer(lambda name: mgr.get(name).executor,
                         host="127.0.0.1", port=0)
        loop.run_until_complete(srv.start())
        ports["bolt"] = srv.port
        loop.run_forever()

    app = create_app(mgr)
    use_uvicorn = os.environ.get("NORNICDB_HTTP_SERVER") == "uvicorn"
    if use_uvicorn:
        http_cfg = uvicorn.Config(app, host="127.0.0.1", port=0,
                                  log_level="error")
        http_srv = uvicorn.Server(http_cfg)

        def http_thread():
            asyncio.new_event_loop()
            http_srv.run()
    else:
        from nornicdb_amd.server.fasthttp import start_http_server

        def http_thread():
            loop = asyncio.new_event_loop()
            asyncio.set_event_loop(loop)
            srv = loop.run_until_complete(
                start_http_server(app, "127.0.0.1", 0))
            ports["http"] = srv.sockets[0].getsockname()[1]
            loop.run_forever()

    threading.Thread(target=bolt_thread, daemon=True).start()
    threading.Thread(target=http_thread, daemon=True).start()
    t0 = time.time()

    def _http_up():
        return http_srv.started if use_uvicorn else "http" in ports

    while ("bolt" not in ports or not _http_up()) and time.time() - t0 < 15:
        time.sleep(0.05)
    http_port = (http_srv.servers[0].sockets[0].getsockname()[1]
                 if use_uvicorn else ports["http"])

    results = {}

    # --- embedded (query cache disabled: direct executor) ---
    lat = []
    for i in range(args.n):
        t1 = time.perf_counter()
        db.executor.execute(Q, {"n": f"p{i % 100}"})
        lat.append(time.perf_counter() - t1)
    results["embedded"] = (lat, sum(lat))
    print("embedded done", flush=True)

    # --- client load generated from separate PROCESSES (the server is a
    # single python process; in-process clients would share its GIL and
    # measure the client, not the server) ---
    import multiprocessing as mp
    ctx = mp.get_context("fork")
    per = args.n // args.concurrency

    def run_procs(target, arg):
        q = ctx.Queue()
        procs = [ctx.Process(target=target, args=(arg, per, q))
                 for _ in range(args.concurrency)]
        t1 = time.perf_counter()
        [p.start() for p in procs]
        lat = []
        for _ in procs:
            lat.extend(q.get(timeout=300))
        [p.join() for p in procs]
        return lat, time.perf_counter() - t1

    results["bolt"] = run_procs(_bolt_proc, ports["bolt"])
    print("bolt done", flush=True)
    results["http_tx"] = run_procs(_http_tx_proc, http_port)
    results["graphql"] = run_procs(_graphql_proc, http_port)

    print(f"{'endpoint':<10} {'ops/s':>8} {'p50 ms':>8} {'p95 ms':>8}", flush=True)
    for name, (lat, wall) in results.items():
        print(f"{name:<10} {len(lat)/wall:>8.0f} {pct(lat, 0.5):>8.2f} "
              f"{pct(lat, 0.95):>8.2f}", flush=True)
    sys.stdout.flush()
    os._exit(0)  # daemon servers


if __name__ == "__main__":
    main()
