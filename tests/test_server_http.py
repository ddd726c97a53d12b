"""HTTP + MCP server tests via FastAPI TestClient.

Models reference pkg/server + pkg/mcp e2e tests."""

import pytest
from fastapi.testclient import TestClient

from nornicdb_amd.db import open_db
from nornicdb_amd.embed import MockEmbedder
from nornicdb_amd.server import create_app


@pytest.fixture
def client():
    mgr = open_db(embedder=MockEmbedder(32), dims=32)
    app = create_app(mgr)
    with TestClient(app) as c:
        yield c
    mgr.close()


class TestCore:
    def test_health_status(self, client):
        assert client.get("/health").json()["status"] == "ok"
        s = client.get("/status").json()
        assert "neo4j" in s["databases"]

    def test_metrics_prometheus(self, client):
        text = client.get("/metrics").text
        assert "nornicdb_nodes" in text and "# TYPE" in text


class TestTxAPI:
    def test_cypher_roundtrip(self, client):
        r = client.post("/db/neo4j/tx/commit", json={"statements": [
            {"statement": "CREATE (n:City {name: $n}) RETURN n",
             "parameters": {"n": "Oslo"}}]}).json()
        assert r["errors"] == []
        res = r["results"][0]
        assert res["columns"] == ["n"]
        assert res["data"][0]["row"][0]["properties"]["name"] == "Oslo"
        assert res["stats"]["nodes_created"] == 1

    def test_syntax_error_reported(self, client):
        r = client.post("/db/neo4j/tx/commit", json={"statements": [
            {"statement": "MATCH (n RETURN"}]}).json()
        assert r["errors"] and "SyntaxError" in r["errors"][0]["code"]

    def test_unknown_db_404(self, client):
        assert client.post("/db/nope/tx/commit",
                           json={"statements": []}).status_code == 404

    def test_multi_statement(self, client):
        r = client.post("/db/neo4j/tx/commit", json={"statements": [
            {"statement": "CREATE (:X {v: 1})"},
            {"statement": "MATCH (n:X) RETURN n.v"}]}).json()
        assert r["results"][1]["data"][0]["row"] == [1]


class TestNornicRoutes:
    def test_store_search_similar(self, client):
        rid = client.post("/nornicdb/store", json={
            "content": "the quick brown fox", "title": "fox"}).json()["id"]
        # drain embed queue synchronously through the manager
        mgr = client.app.state.manager
        mgr.get().embed_queue.drain()
        res = client.post("/nornicdb/search",
                          json={"query": "quick brown fox"}).json()["results"]
        assert res and res[0]["id"] == rid
        sim = client.get(f"/nornicdb/similar/{rid}").json()
        assert "results" in sim

    def test_embed_endpoint(self, client):
        r = client.post("/nornicdb/embed", json={"texts": ["a", "b"]}).json()
        assert len(r["embeddings"]) == 2
        assert len(r["embeddings"][0]) == 32

    def test_decay_run(self, client):
        client.post("/nornicdb/store", json={"content": "m"})
        r = client.post("/nornicdb/decay/run").json()
        assert r["scored"] >= 1


class TestAdmin:
    def test_db_management(self, client):
        assert client.post("/admin/databases/t1").status_code == 200
        assert "t1" in client.get("/admin/databases").json()["databases"]
        assert client.delete("/admin/databases/t1").status_code == 200

    def test_gdpr(self, client):
        client.post("/db/neo4j/tx/commit", json={"statements": [
            {"statement": "CREATE (:P {subject: 'alice', d: 1})"}]})
        exp = client.get("/gdpr/export/alice").json()
        assert len(exp["nodes"]) == 1
        dele = client.delete("/gdpr/delete/alice").json()
        assert dele["deleted"] == 1


class TestMCP:
    def _rpc(self, client, method, params=None, id=1):
        return client.post("/mcp", json={
            "jsonrpc": "2.0", "id": id, "method": method,
            "params": params or {}}).json()

    def test_initialize_and_list(self, client):
        r = self._rpc(client, "initialize")
        assert r["result"]["serverInfo"]["name"] == "nornicdb-amd"
        tools = self._rpc(client, "tools/list")["result"]["tools"]
        names = {t["name"] for t in tools}
        assert names == {"store", "recall", "discover", "link", "task", "tasks"}

    def test_store_recall_link_flow(self, client):
        import json as J
        r1 = self._rpc(client, "tools/call", {
            "name": "store", "arguments": {"content": "mcp memory one"}})
        id1 = J.loads(r1["result"]["content"][0]["text"])["id"]
        r2 = self._rpc(client, "tools/call", {
            "name": "store", "arguments": {"content": "mcp memory two"}})
        id2 = J.loads(r2["result"]["content"][0]["text"])["id"]
        client.app.state.manager.get().embed_queue.drain()
        rec = self._rpc(client, "tools/call", {
            "name": "recall", "arguments": {"query": "mcp memory one"}})
        found = J.loads(rec["result"]["content"][0]["text"])
        assert any(m["id"] == id1 for m in found)
        self._rpc(client, "tools/call", {
            "name": "link", "arguments": {"from": id1, "to": id2}})
        disc = self._rpc(client, "tools/call", {
            "name": "discover", "arguments": {"id": id1}})
        linked = J.loads(disc["result"]["content"][0]["text"])
        assert any(m["id"] == id2 for m in linked)

    def test_tasks(self, client):
        import json as J
        self._rpc(client, "tools/call", {
            "name": "task", "arguments": {"title": "write tests"}})
        r = self._rpc(client, "tools/call", {"name": "tasks", "arguments": {}})
        tasks = J.loads(r["result"]["content"][0]["text"])
        assert tasks and tasks[0]["title"] == "write tests"

    def test_unknown_method(self, client):
        r = self._rpc(client, "nope/nope")
        assert r["error"]["code"] == -32601


def test_console_served(client):
    r = client.get("/")
    assert r.status_code == 200
    assert "NornicDB-AMD console" in r.text or "console" in r.text


class TestExplicitTransactions:
    """Explicit tx lifecycle endpoints (reference server_db.go:381 — same
    simplified semantics: eager execution, bookmark on commit,
    acknowledged rollback)."""

    def test_open_execute_commit(self, client):
        r = client.post("/db/neo4j/tx", json={"statements": [
            {"statement": "CREATE (:TX {v: 1})"}]})
        assert r.status_code == 200
        loc = r.headers["Location"]
        txid = loc.rstrip("/").rsplit("/", 1)[-1]
        r2 = client.post(f"/db/neo4j/tx/{txid}", json={"statements": [
            {"statement": "MATCH (t:TX) RETURN t.v"}]})
        assert r2.json()["results"][0]["data"][0]["row"] == [1]
        r3 = client.post(f"/db/neo4j/tx/{txid}/commit", json={})
        assert r3.json()["lastBookmarks"]
        # closed tx rejects further statements
        r4 = client.post(f"/db/neo4j/tx/{txid}", json={"statements": []})
        assert r4.status_code == 404

    def test_rollback_acknowledges(self, client):
        r = client.post("/db/neo4j/tx", json={"statements": []})
        txid = r.headers["Location"].rstrip("/").rsplit("/", 1)[-1]
        r2 = client.delete(f"/db/neo4j/tx/{txid}")
        assert r2.status_code == 200


class TestAdminAuthRoutes:
    """Reference server_router.go surface: /auth/*, /admin/*, /mcp REST
    aliases, embed/index admin, /graphql/playground."""

    def _client(self, with_auth=False):
        from fastapi.testclient import TestClient
        from nornicdb_amd.db import DatabaseManager
        from nornicdb_amd.storage.memory import MemoryEngine
        from nornicdb_amd.server.http import create_app
        mgr = DatabaseManager(MemoryEngine())
        auth = None
        if with_auth:
            from nornicdb_amd.auth import Authenticator
            auth = Authenticator(mgr.get().engine)
            auth.ensure_admin(password="secret123")
        return TestClient(create_app(mgr, auth=auth)), mgr

    def test_admin_and_gpu_status(self):
        c, mgr = self._client()
        mgr.get().cypher("CREATE (:Doc {t: 'x'})")
        assert c.get("/admin/stats").json()["nodes"] == 1
        assert "bolt_host" in c.get("/admin/config").json()
        gpu = c.get("/admin/gpu/status").json()
        assert "available" in gpu and "backend" in gpu
        assert c.post("/admin/gpu/disable").json()["enabled"] is False
        assert c.post("/admin/gpu/enable").json()["enabled"] is True
        # no GPU in CI: test endpoint reports gracefully
        assert c.post("/admin/gpu/test").status_code == 200

    def test_embed_and_rebuild(self):
        c, mgr = self._client()
        mgr.get().cypher("CREATE (:Doc {title: 'hello'})")
        assert c.get("/nornicdb/embed/stats").status_code == 200
        assert c.post("/nornicdb/search/rebuild").json()["status"] == "rebuilt"
        assert c.post("/nornicdb/embed/clear").status_code == 200
        assert c.get("/nornicdb/decay").status_code == 200

    def test_mcp_rest_aliases(self):
        c, _ = self._client()
        assert c.get("/mcp/health").json()["status"] == "ok"
        tools = c.get("/mcp/tools/list").json()["tools"]
        assert any(t["name"] == "recall" for t in tools)
        init = c.post("/mcp/initialize").json()
        assert init["serverInfo"]["name"] == "nornicdb-amd"

    def test_graphql_playground(self):
        c, _ = self._client()
        r = c.get("/graphql/playground")
        assert r.status_code == 200 and "GraphQL" in r.text

    def test_auth_user_lifecycle(self):
        c, _ = self._client(with_auth=True)
        tok = c.post("/auth/login", json={
            "username": "neo4j", "password": "secret123"}).json()["token"]
        H = {"Authorization": f"Bearer {tok}"}
        me = c.get("/auth/me", headers=H).json()
        assert me == {"username": "neo4j", "role": "admin"}
        assert c.post("/auth/users", headers=H, json={
            "username": "bob", "password": "pw1234567",
            "role": "reader"}).status_code == 200
        users = {u["username"] for u in
                 c.get("/auth/users", headers=H).json()["users"]}
        assert users == {"neo4j", "bob"}
        assert c.post("/auth/password", headers=H, json={
            "current": "secret123", "password": "newpass99"}).status_code == 200
        assert c.post("/auth/login", json={
            "username": "neo4j", "password": "newpass99"}).status_code == 200
        assert c.post("/auth/logout", headers=H).status_code == 200
        cfg = c.get("/auth/config").json()
        assert cfg["enabled"] is True

    def test_auth_me_requires_token(self):
        c, _ = self._client(with_auth=True)
        assert c.get("/auth/me").status_code == 401

    def test_bifrost_status_alias(self):
        c, _ = self._client()
        assert "available" in c.get("/api/bifrost/status").json()


def test_admin_console_served_and_endpoints(client):
    """The single-file admin SPA (reference ui/src/ parity) is served at /
    and every endpoint it calls exists."""
    html = client.get("/").text
    assert "NornicDB-AMD Console" in html
    for tab in ("tab-query", "tab-search", "tab-memory", "tab-databases",
                "tab-admin"):
        assert tab in html
    for path in ("/admin/databases", "/admin/stats", "/nornicdb/embed/stats",
                 "/nornicdb/decay", "/status"):
        assert client.get(path).status_code == 200, path


def test_http_tx_rollback_undoes_writes():
    """POST /db/x/tx (open, run CREATE) then DELETE (rollback) leaves
    no state; the commit path keeps it."""
    from fastapi.testclient import TestClient

    from nornicdb_amd.db import open_db
    from nornicdb_amd.embed import MockEmbedder
    from nornicdb_amd.server import create_app

    mgr = open_db(embedder=MockEmbedder(8), dims=8)
    app = create_app(mgr, auth=None)
    c = TestClient(app.app if hasattr(app, "app") else app)

    r = c.post("/db/neo4j/tx", json={"statements": [
        {"statement": "CREATE (:HT {x: 1})"}]})
    txid = r.headers["Location"].rsplit("/", 1)[1]
    # visible inside the tx (read-your-writes), then rolled back
    r2 = c.post(f"/db/neo4j/tx/{txid}", json={"statements": [
        {"statement": "MATCH (n:HT) RETURN count(n)"}]})
    assert r2.json()["results"][0]["data"][0]["row"] == [1]
    c.delete(f"/db/neo4j/tx/{txid}")
    assert mgr.get().cypher("MATCH (n:HT) RETURN count(n)").rows == [[0]]

    r = c.post("/db/neo4j/tx", json={"statements": [
        {"statement": "CREATE (:HT {x: 2})"}]})
    txid = r.headers["Location"].rsplit("/", 1)[1]
    c.post(f"/db/neo4j/tx/{txid}/commit", json={})
    assert mgr.get().cypher("MATCH (n:HT) RETURN count(n)").rows == [[1]]
