"""End-to-end `python -m nornicdb_amd serve` integration: real process,
disk engine, Bolt + HTTP + console, clean shutdown."""

import json
import os
import socket
import struct
import subprocess
import sys
import time
import urllib.request

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_ports(n):
    socks = []
    for _ in range(n):
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        socks.append(s)
    ports = [s.getsockname()[1] for s in socks]
    for s in socks:
        s.close()
    return ports


@pytest.mark.timeout(120)
def test_serve_boots_disk_engine_and_answers(tmp_path):
    bolt_port, http_port = _free_ports(2)
    env = dict(os.environ, PYTHONPATH=REPO)
    proc = subprocess.Popen(
        [sys.executable, "-m", "nornicdb_amd", "serve",
         "--data-dir", str(tmp_path / "data"),
         "--bolt-port", str(bolt_port), "--http-port", str(http_port)],
        cwd=REPO, env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    try:
        # wait for HTTP
        deadline = time.time() + 60
        up = False
        while time.time() < deadline:
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{http_port}/health", timeout=1) as r:
                    if r.status == 200:
                        up = True
                        break
            except Exception:
                time.sleep(0.2)
        assert up, proc.stdout.read(4000)

        # HTTP tx write + read
        body = json.dumps({"statements": [
            {"statement": "CREATE (:Boot {ok: true}) RETURN 1 AS one"}]}).encode()
        req = urllib.request.Request(
            f"http://127.0.0.1:{http_port}/db/neo4j/tx/commit", data=body,
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=5) as r:
            out = json.loads(r.read())
        assert out["results"][0]["data"][0]["row"] == [1]

        # console served
        with urllib.request.urlopen(
                f"http://127.0.0.1:{http_port}/", timeout=5) as r:
            assert b"NornicDB-AMD Console" in r.read()

        # Bolt handshake + RUN over the same data
        from nornicdb_amd.bolt import packstream as ps
        s = socket.create_connection(("127.0.0.1", bolt_port), timeout=5)
        s.sendall(struct.pack(">I", 0x6060B017)
                  + bytes([0, 0, 4, 4]) + bytes(12))
        assert s.recv(4)[3] == 4

        def send(tag, *fields):
            data = ps.pack(ps.Structure(tag, list(fields)))
            s.sendall(struct.pack(">H", len(data)) + data + b"\x00\x00")

        buf = b""

        def recv():
            nonlocal buf
            msg = b""
            while True:
                while len(buf) < 2:
                    buf += s.recv(65536)
                size = struct.unpack(">H", buf[:2])[0]
                buf = buf[2:]
                if size == 0:
                    if msg:
                        return ps.unpack(msg)
                    continue
                while len(buf) < size:
                    buf += s.recv(65536)
                msg += buf[:size]
                buf = buf[size:]

        send(0x01, {"user_agent": "it", "scheme": "none"})
        assert recv().tag == 0x70
        send(0x10, "MATCH (b:Boot) RETURN count(b)", {}, {})
        send(0x3F, {"n": -1})
        assert recv().tag == 0x70
        rec = recv()
        assert rec.tag == 0x71 and rec.fields[0] == [1]
        s.close()
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()

    # data survived on disk: reopen offline and check
    from nornicdb_amd.db import open_db
    mgr = open_db(str(tmp_path / "data"))
    r = mgr.get("neo4j").execute_cypher("MATCH (b:Boot) RETURN count(b)")
    assert r.rows == [[1]]
    mgr.close()


@pytest.mark.timeout(120)
def test_serve_auth_login_and_protected_routes(tmp_path):
    """--auth: boot prints the initial admin password; login yields a
    token; protected routes 401 without it and work with it — all over
    the fasthttp server."""
    bolt_port, http_port = _free_ports(2)
    env = dict(os.environ, PYTHONPATH=REPO,
               NORNICDB_INITIAL_ADMIN_PASSWORD="s3cret-pw")
    proc = subprocess.Popen(
        [sys.executable, "-m", "nornicdb_amd", "serve", "--auth",
         "--data-dir", str(tmp_path / "data"),
         "--bolt-port", str(bolt_port), "--http-port", str(http_port)],
        cwd=REPO, env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    try:
        deadline = time.time() + 60
        up = False
        while time.time() < deadline:
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{http_port}/health", timeout=1) as r:
                    if r.status == 200:
                        up = True
                        break
            except Exception:
                time.sleep(0.2)
        assert up, proc.stdout.read(4000)

        base = f"http://127.0.0.1:{http_port}"
        # protected route without a token -> 401
        req = urllib.request.Request(base + "/admin/stats")
        try:
            urllib.request.urlopen(req, timeout=5)
            assert False, "expected 401"
        except urllib.error.HTTPError as e:
            assert e.code == 401

        # login
        body = json.dumps({"username": "neo4j",
                           "password": "s3cret-pw"}).encode()
        req = urllib.request.Request(
            base + "/auth/login", data=body,
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=5) as r:
            out = json.loads(r.read())
        token = out.get("token") or out.get("access_token")
        assert token, out

        hdr = {"Authorization": f"Bearer {token}",
               "Content-Type": "application/json"}
        req = urllib.request.Request(base + "/auth/me", headers=hdr)
        with urllib.request.urlopen(req, timeout=5) as r:
            me = json.loads(r.read())
        assert me.get("username") == "neo4j" or me.get("user"), me

        # authenticated tx round-trip (full FastAPI path: the fast path
        # is disabled when auth is on)
        body = json.dumps({"statements": [
            {"statement": "RETURN 42 AS v"}]}).encode()
        req = urllib.request.Request(
            base + "/db/neo4j/tx/commit", data=body, headers=hdr)
        with urllib.request.urlopen(req, timeout=5) as r:
            out = json.loads(r.read())
        assert out["results"][0]["data"][0]["row"] == [42]
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
