"""server/fasthttp.py: the in-repo asyncio HTTP/1.1 server over a real
socket — keep-alive, chunked request bodies, streamed (SSE) responses,
static console, 404s. The ASGI app is the full create_app surface."""

import asyncio
import json
import socket
import threading
import time

import pytest


@pytest.fixture(scope="module")
def server():
    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder
    from nornicdb_amd.server import create_app
    from nornicdb_amd.server.fasthttp import start_http_server

    mgr = open_db(embedder=MockEmbedder(16), dims=16)
    db = mgr.get()
    db.cypher("CREATE (:Person {name: 'alice', age: 30})")
    app = create_app(mgr)
    ports = {}
    loops = {}

    def run():
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)
        loops["loop"] = loop
        srv = loop.run_until_complete(start_http_server(app, "127.0.0.1", 0))
        ports["http"] = srv.sockets[0].getsockname()[1]
        loop.run_forever()

    t = threading.Thread(target=run, daemon=True)
    t.start()
    t0 = time.time()
    while "http" not in ports and time.time() - t0 < 10:
        time.sleep(0.02)
    assert "http" in ports
    yield ports["http"]
    loops["loop"].call_soon_threadsafe(loops["loop"].stop)


def _req(port, raw: bytes, keep_sock=None, read_until_close=False):
    s = keep_sock or socket.create_connection(("127.0.0.1", port), timeout=5)
    s.sendall(raw)
    buf = b""
    s.settimeout(5)
    while b"\r\n\r\n" not in buf:
        buf += s.recv(65536)
    head, _, rest = buf.partition(b"\r\n\r\n")
    headers = {}
    for ln in head.split(b"\r\n")[1:]:
        k, _, v = ln.partition(b":")
        headers[k.strip().lower()] = v.strip()
    status = int(head.split(b" ", 2)[1])
    if b"content-length" in headers:
        want = int(headers[b"content-length"])
        while len(rest) < want:
            rest += s.recv(65536)
        body = rest[:want]
    elif read_until_close:
        try:
            while True:
                chunk = s.recv(65536)
                if not chunk:
                    break
                rest += chunk
        except socket.timeout:
            pass
        body = rest
    else:
        body = rest
    return status, headers, body, s


def test_get_health_and_keepalive(server):
    raw = (b"GET /health HTTP/1.1\r\nhost: t\r\n\r\n")
    st, hd, body, s = _req(server, raw)
    assert st == 200 and b"ok" in body
    # second request on the SAME socket (keep-alive)
    st2, _, body2, _ = _req(server, raw, keep_sock=s)
    assert st2 == 200 and b"ok" in body2
    s.close()


def test_post_tx_commit(server):
    payload = json.dumps({"statements": [{
        "statement": "MATCH (n:Person {name: $n}) RETURN n.name, n.age",
        "parameters": {"n": "alice"}}]}).encode()
    raw = (b"POST /db/neo4j/tx/commit HTTP/1.1\r\nhost: t\r\n"
           b"content-type: application/json\r\n"
           b"content-length: " + str(len(payload)).encode() + b"\r\n\r\n"
           + payload)
    st, _, body, s = _req(server, raw)
    s.close()
    assert st == 200
    out = json.loads(body)
    assert out["results"][0]["data"][0]["row"] == ["alice", 30]


def test_chunked_request_body(server):
    payload = json.dumps({"statements": [{
        "statement": "RETURN 1 AS one"}]}).encode()
    half = len(payload) // 2
    raw = (b"POST /db/neo4j/tx/commit HTTP/1.1\r\nhost: t\r\n"
           b"content-type: application/json\r\n"
           b"transfer-encoding: chunked\r\n\r\n"
           + b"%x\r\n" % half + payload[:half] + b"\r\n"
           + b"%x\r\n" % (len(payload) - half) + payload[half:] + b"\r\n"
           + b"0\r\n\r\n")
    st, _, body, s = _req(server, raw)
    s.close()
    assert st == 200
    assert json.loads(body)["results"][0]["data"][0]["row"] == [1]


def test_query_string_and_404(server):
    st, _, _, s = _req(server, b"GET /nope/nothing HTTP/1.1\r\nhost: t\r\n\r\n")
    s.close()
    assert st == 404


def test_console_served(server):
    st, hd, body, s = _req(server, b"GET / HTTP/1.1\r\nhost: t\r\n\r\n")
    s.close()
    assert st == 200 and b"<html" in body.lower()


def test_streaming_sse_chunked(server):
    # /api/bifrost/events streams without content-length -> the server
    # must emit transfer-encoding: chunked and terminate the stream
    raw = b"GET /api/bifrost/events HTTP/1.1\r\nhost: t\r\nconnection: close\r\n\r\n"
    st, hd, body, s = _req(server, raw, read_until_close=True)
    s.close()
    assert st == 200
    assert hd.get(b"transfer-encoding") == b"chunked"
    assert b"data:" in body


def test_graphql_roundtrip(server):
    payload = json.dumps({"query":
        'query { nodes(label: "Person", limit: 3) { id labels } }'}).encode()
    raw = (b"POST /graphql HTTP/1.1\r\nhost: t\r\n"
           b"content-type: application/json\r\n"
           b"content-length: " + str(len(payload)).encode() + b"\r\n\r\n"
           + payload)
    st, _, body, s = _req(server, raw)
    s.close()
    assert st == 200
    out = json.loads(body)
    assert out["data"]["nodes"] and out["data"]["nodes"][0]["labels"] == ["Person"]


def test_https_serving(tmp_path):
    """fasthttp over TLS: self-signed pair + ssl context, GET /health."""
    import ssl
    import urllib.request

    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder
    from nornicdb_amd.server import create_app
    from nornicdb_amd.server.fasthttp import start_http_server
    from nornicdb_amd.utils.tls import ensure_self_signed, make_ssl_context

    cert, key = ensure_self_signed(str(tmp_path))
    sctx = make_ssl_context(cert, key)
    mgr = open_db(embedder=MockEmbedder(8), dims=8)
    app = create_app(mgr, auth=None)
    ports = {}
    loops = {}

    def run():
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)
        loops["loop"] = loop
        srv = loop.run_until_complete(
            start_http_server(app, "127.0.0.1", 0, ssl_context=sctx))
        ports["p"] = srv.sockets[0].getsockname()[1]
        loop.run_forever()

    t = threading.Thread(target=run, daemon=True)
    t.start()
    t0 = time.time()
    while "p" not in ports and time.time() - t0 < 10:
        time.sleep(0.02)
    cctx = ssl.create_default_context()
    cctx.check_hostname = False
    cctx.verify_mode = ssl.CERT_NONE
    with urllib.request.urlopen(f"https://127.0.0.1:{ports['p']}/health",
                                context=cctx, timeout=5) as r:
        assert r.status == 200 and b"ok" in r.read()
    loops["loop"].call_soon_threadsafe(loops["loop"].stop)
