"""HTTP API server (FastAPI).

Parity: reference pkg/server/server_router.go — Neo4j HTTP tx API
(/db/{name}/tx/commit, server_db.go), /nornicdb/* memory+search routes
(server_nornicdb.go), auth routes, admin + health + Prometheus /metrics
(:104-108), GDPR endpoints (server_gdpr.go), MCP mount (/mcp).
"""

from __future__ import annotations

import json
import time
from typing import Any, Dict, List, Optional

from fastapi import Depends, FastAPI, HTTPException, Request, Response
from starlette.responses import StreamingResponse
from pydantic import BaseModel

from ..auth import AuthError
from ..cypher import CypherRuntimeError, CypherSyntaxError
from ..cypher.executor import Path
from ..db import DatabaseManager
from ..storage.types import Edge, Node
from .metrics import MetricsRegistry
from .mcp import MCPServer

START_TIME = time.time()


def _jsonable(v):
    if isinstance(v, Node):
        return {"id": v.id, "labels": v.labels, "properties": v.properties}
    if isinstance(v, Edge):
        return {"id": v.id, "type": v.type, "startNode": v.start_node,
                "endNode": v.end_node, "properties": v.properties}
    if isinstance(v, Path):
        return {"nodes": [_jsonable(n) for n in v.nodes],
                "relationships": [_jsonable(e) for e in v.edges]}
    if isinstance(v, list):
        return [_jsonable(x) for x in v]
    if isinstance(v, dict):
        return {k: _jsonable(x) for k, x in v.items()}
    if isinstance(v, float) and (v != v or v in (float("inf"), float("-inf"))):
        return None
    if hasattr(v, "component"):  # temporal values -> ISO-8601 strings
        return str(v)
    return v


class Statement(BaseModel):
    statement: str
    parameters: Dict[str, Any] = {}


class TxRequest(BaseModel):
    statements: List[Statement] = []


class StoreRequest(BaseModel):
    content: str
    title: str = ""
    memory_type: str = "episodic"
    importance: float = 0.5
    tags: List[str] = []
    metadata: Dict[str, Any] = {}


class SearchRequest(BaseModel):
    query: str
    limit: int = 10
    labels: Optional[List[str]] = None
    mmr: bool = False


class LoginRequest(BaseModel):
    username: str
    password: str


class _TxFastPath:
    """Raw-ASGI fast path for POST /db/{name}/tx/commit — the cross-
    protocol hot route. Skips FastAPI routing/DI/validation (~0.5 ms per
    request of the reference-gap ASGI tax): parse the
    body, run the shared statement runner, emit json. Everything else
    (auth present, other paths, errors) falls through to the FastAPI app.
    """

    def __init__(self, app, auth):
        self.app = app
        self.auth = auth

    def __getattr__(self, item):  # delegate state/router/etc. for tests
        return getattr(self.app, item)

    async def __call__(self, scope, receive, send):
        if (scope.get("type") == "http" and scope.get("method") == "POST"
                and self.auth is None):
            path = scope.get("path", "")
            if path.startswith("/db/") and path.endswith("/tx/commit"):
                db_name = path[4:-len("/tx/commit")]
                if "/" not in db_name:
                    await self._fast(db_name, receive, send)
                    return
            if path == "/graphql":
                await self._fast_gql(receive, send)
                return
        await self.app(scope, receive, send)

    async def _fast(self, db_name, receive, send):
        chunks = []
        while True:
            ev = await receive()
            chunks.append(ev.get("body", b""))
            if not ev.get("more_body"):
                break
        status = 200
        try:
            body = json.loads(b"".join(chunks) or b"{}")
            out = self.app._tx_commit_body(db_name, body)
        except KeyError:
            status, out = 404, {"detail": f"database {db_name} not found"}
        except Exception as e:
            out = {"results": [], "errors": [
                {"code": "Neo.ClientError.Request.InvalidFormat",
                 "message": str(e)}]}
        await self._respond(send, status, out)

    async def _fast_gql(self, receive, send):
        chunks = []
        while True:
            ev = await receive()
            chunks.append(ev.get("body", b""))
            if not ev.get("more_body"):
                break
        try:
            body = json.loads(b"".join(chunks) or b"{}")
            out = self.app._gql_for(None).execute(
                body.get("query", ""), body.get("variables"))
            status = 200
        except Exception as e:
            status, out = 200, {"data": None, "errors": [{"message": str(e)}]}
        await self._respond(send, status, out)

    @staticmethod
    async def _respond(send, status, out):
        payload = json.dumps(out, default=str).encode()
        await send({"type": "http.response.start", "status": status,
                    "headers": [(b"content-type", b"application/json"),
                                (b"content-length",
                                 str(len(payload)).encode())]})
        await send({"type": "http.response.body", "body": payload})


def create_app(mgr: DatabaseManager, auth=None, version: str = "0.1.0") -> FastAPI:
    app = FastAPI(title="NornicDB-AMD", version=version)
    metrics = MetricsRegistry()
    mcp = MCPServer(mgr)
    app.state.manager = mgr
    app.state.metrics = metrics

    def check_auth(request: Request):
        if auth is None:
            return None
        hdr = request.headers.get("authorization", "")
        if hdr.lower().startswith("bearer "):
            try:
                return auth.validate_token(hdr[7:])
            except Exception:
                raise HTTPException(401, "invalid token")
        if hdr.lower().startswith("basic "):
            import base64
            try:
                user, pw = base64.b64decode(hdr[6:]).decode().split(":", 1)
                auth.login(user, pw)
                return user
            except HTTPException:
                raise
            except Exception:
                raise HTTPException(401, "invalid credentials")
        raise HTTPException(401, "authentication required")

    # ---- embedded admin console (reference ui/ React app -> minimal
    # single-file console served at /) ----
    @app.get("/")
    def console():
        import os
        path = os.path.join(os.path.dirname(__file__), "static", "console.html")
        with open(path) as f:
            return Response(f.read(), media_type="text/html")

    # ---- health / status / metrics ----
    @app.get("/health")
    def health():
        return {"status": "ok"}

    @app.get("/status")
    def status():
        db = mgr.get()
        return {
            "version": version,
            "uptime_s": round(time.time() - START_TIME, 1),
            "databases": mgr.list(),
            "nodes": db.engine.node_count(),
            "relationships": db.engine.edge_count(),
        }

    @app.get("/metrics")
    def prom_metrics():
        db = mgr.get()
        metrics.gauge("nornicdb_nodes", db.engine.node_count())
        metrics.gauge("nornicdb_relationships", db.engine.edge_count())
        metrics.gauge("nornicdb_uptime_seconds", time.time() - START_TIME)
        return Response(metrics.render(), media_type="text/plain; version=0.0.4")

    # ---- auth ----
    @app.post("/auth/login")
    def login(req: LoginRequest):
        if auth is None:
            raise HTTPException(404, "auth disabled")
        try:
            token = auth.issue_token(req.username, req.password)
        except Exception as e:
            raise HTTPException(401, str(e))
        return {"token": token}

    @app.post("/auth/logout")
    def logout(_user=Depends(check_auth)):
        return {"status": "ok"}  # tokens are stateless JWTs

    # ---- OAuth2/OIDC (reference pkg/auth/oauth.go + cmd/oauth-provider) --
    oauth_provider = None
    oauth_client = None
    if auth is not None:
        import os as _os
        from ..auth.oauth import OAuthClientManager, OAuthProvider
        if _os.environ.get("NORNICDB_OAUTH_PROVIDER_ENABLED", "").lower() \
                in ("1", "true", "yes"):
            oauth_provider = OAuthProvider(
                _os.environ.get("NORNICDB_OAUTH_CLIENT_ID", "nornicdb"),
                _os.environ.get("NORNICDB_OAUTH_CLIENT_SECRET", "secret"),
                _os.environ.get("NORNICDB_OAUTH_ISSUER",
                                "http://127.0.0.1:7474"),
                authenticator=auth)
        if _os.environ.get("NORNICDB_AUTH_PROVIDER") == "oauth":
            oauth_client = OAuthClientManager(
                auth,
                _os.environ.get("NORNICDB_OAUTH_ISSUER", ""),
                _os.environ.get("NORNICDB_OAUTH_CLIENT_ID", ""),
                _os.environ.get("NORNICDB_OAUTH_CLIENT_SECRET", ""),
                _os.environ.get("NORNICDB_OAUTH_CALLBACK_URL", ""))

    @app.get("/.well-known/oauth-authorization-server")
    @app.get("/.well-known/openid-configuration")
    def oauth_discovery():
        if oauth_provider is None:
            raise HTTPException(404, "OAuth provider disabled")
        return oauth_provider.discovery()

    @app.get("/oauth2/v1/authorize")
    def oauth_authorize(request: Request):
        if oauth_provider is None:
            raise HTTPException(404, "OAuth provider disabled")
        status, body = oauth_provider.authorize(dict(request.query_params))
        if status != 200:
            raise HTTPException(status, body.get("error", "invalid_request"))
        return body

    async def _form(request: Request) -> dict:
        # urlencoded parse without the python-multipart dependency
        import urllib.parse as _up
        raw = (await request.body()).decode()
        return {k: v[0] for k, v in _up.parse_qs(raw).items()}

    @app.post("/oauth2/v1/authorize/consent")
    async def oauth_consent(request: Request):
        if oauth_provider is None:
            raise HTTPException(404, "OAuth provider disabled")
        form = await _form(request)
        status, body = oauth_provider.consent(
            form.get("username", ""), form.get("password", ""),
            form.get("redirect_uri", ""), form.get("state", ""),
            form.get("scope", "openid profile"))
        if status == 302:
            from fastapi.responses import RedirectResponse
            return RedirectResponse(body["location"], status_code=302)
        raise HTTPException(status, body.get("error_description",
                                             body.get("error", "denied")))

    @app.post("/oauth2/v1/token")
    async def oauth_token(request: Request):
        if oauth_provider is None:
            raise HTTPException(404, "OAuth provider disabled")
        form = await _form(request)
        status, body = oauth_provider.token(form)
        if status != 200:
            from fastapi.responses import JSONResponse
            return JSONResponse(body, status_code=status)
        return body

    @app.get("/oauth2/v1/userinfo")
    def oauth_userinfo(request: Request):
        if oauth_provider is None:
            raise HTTPException(404, "OAuth provider disabled")
        status, body = oauth_provider.userinfo(
            request.headers.get("authorization", ""))
        if status != 200:
            raise HTTPException(status, body.get("error", "invalid_token"))
        return body

    @app.get("/auth/oauth/login")
    def oauth_login_url():
        if oauth_client is None or not oauth_client.is_configured():
            raise HTTPException(404, "OAuth login not configured")
        url, state = oauth_client.generate_auth_url()
        return {"auth_url": url, "state": state}

    @app.get("/auth/oauth/callback")
    def oauth_callback(code: str, state: str):
        if oauth_client is None:
            raise HTTPException(404, "OAuth login not configured")
        try:
            return oauth_client.handle_callback(code, state)
        except AuthError as e:
            raise HTTPException(401, str(e))

    @app.get("/auth/config")
    def auth_config():
        return {"enabled": auth is not None,
                "oauth": oauth_client is not None or oauth_provider is not None,
                "methods": (["password", "token"]
                            + (["oauth"] if oauth_client else []))
                if auth else []}

    @app.get("/auth/me")
    def auth_me(user=Depends(check_auth)):
        if auth is None:
            raise HTTPException(404, "auth disabled")
        return {"username": user.get("sub"), "role": user.get("role")}

    @app.post("/auth/password")
    def auth_password(body: Dict[str, Any], user=Depends(check_auth)):
        if auth is None:
            raise HTTPException(404, "auth disabled")
        try:
            auth.login(user.get("sub"), body.get("current", ""))
        except Exception:
            raise HTTPException(403, "current password incorrect")
        auth.set_password(user.get("sub"), body.get("password", ""))
        return {"status": "ok"}

    @app.post("/auth/api-token")
    def auth_api_token(body: Dict[str, Any], user=Depends(check_auth)):
        if auth is None:
            raise HTTPException(404, "auth disabled")
        token = auth.issue_token(body.get("username", user.get("sub")),
                                 body.get("password", ""))
        return {"token": token}

    @app.get("/auth/users")
    def auth_users(user=Depends(check_auth)):
        if auth is None:
            raise HTTPException(404, "auth disabled")
        auth.require(user, "admin")
        return {"users": auth.list_users()}

    @app.post("/auth/users")
    def auth_create_user(body: Dict[str, Any], user=Depends(check_auth)):
        if auth is None:
            raise HTTPException(404, "auth disabled")
        auth.require(user, "admin")
        role = {"reader": "readonly", "editor": "readwrite",
                "publisher": "readwrite"}.get(body.get("role", "readonly"),
                                              body.get("role", "readonly"))
        auth.create_user(body["username"], body["password"], role=role)
        return {"status": "created"}

    # ---- admin (reference server_router.go /admin/*) ----
    @app.get("/admin/config")
    def admin_config(_user=Depends(check_auth)):
        from ..utils.config import Config
        cfg = Config()
        return {k: getattr(cfg, k) for k in sorted(vars(cfg))
                if not k.startswith("_")}

    @app.get("/admin/stats")
    def admin_stats(_user=Depends(check_auth)):
        db = mgr.get()
        labels = {}
        for n in db.engine.all_nodes():
            for lb in n.labels:
                labels[lb] = labels.get(lb, 0) + 1
        return {"nodes": db.engine.node_count(),
                "relationships": db.engine.edge_count(),
                "labels": labels, "databases": mgr.list()}

    # GPU admin (reference /admin/gpu/*; here: the MI355X torch backend)
    _gpu_enabled = [True]

    @app.get("/admin/gpu/status")
    def gpu_status():
        import torch
        avail = torch.cuda.is_available()
        info = {"available": avail, "enabled": _gpu_enabled[0],
                "backend": "rocm-hip" if avail else "cpu"}
        if avail:
            info["device"] = torch.cuda.get_device_name(0)
            info["devices"] = torch.cuda.device_count()
            free, total = torch.cuda.mem_get_info(0)
            info["memory"] = {"free": free, "total": total}
        return info

    @app.post("/admin/gpu/enable")
    def gpu_enable(_user=Depends(check_auth)):
        _gpu_enabled[0] = True
        return {"enabled": True}

    @app.post("/admin/gpu/disable")
    def gpu_disable(_user=Depends(check_auth)):
        _gpu_enabled[0] = False
        return {"enabled": False}

    @app.post("/admin/gpu/test")
    def gpu_test(_user=Depends(check_auth)):
        import torch
        if not torch.cuda.is_available():
            return {"ok": False, "reason": "no GPU visible"}
        t0 = time.time()
        a = torch.randn(512, 512, device="cuda")
        s = float((a @ a).sum())
        torch.cuda.synchronize()
        return {"ok": s == s, "elapsed_ms": round((time.time() - t0) * 1e3, 2)}

    # ---- embed queue + index admin (reference /nornicdb/embed/*) ----
    @app.get("/nornicdb/embed/stats")
    def embed_stats():
        db = mgr.get()
        pending = len(db.engine.pending_embeddings(10_000))
        return {"pending": pending, "indexed": len(db.search.emb)}

    @app.post("/nornicdb/embed/trigger")
    def embed_trigger(_user=Depends(check_auth)):
        db = mgr.get()
        done = db.embed_queue.drain(timeout=30.0) if db.embed_queue else True
        return {"drained": done}

    @app.post("/nornicdb/embed/clear")
    def embed_clear(_user=Depends(check_auth)):
        db = mgr.get()
        ids = db.engine.pending_embeddings(1_000_000)
        for nid in ids:
            db.engine.clear_pending_embedding(nid)
        return {"cleared": len(ids)}

    @app.post("/nornicdb/search/rebuild")
    def search_rebuild(_user=Depends(check_auth)):
        db = mgr.get()
        db.search.build_indexes()
        return {"status": "rebuilt", "indexed": len(db.search.emb)}

    @app.get("/nornicdb/decay")
    def decay_info():
        db = mgr.get()
        cfg = getattr(db, "decay_config", None)
        return {"enabled": cfg is not None,
                "halfLife": {"episodic": "7 days", "semantic": "69 days",
                             "procedural": "693 days"}}

    # ---- Neo4j HTTP transaction API ----
    # raw-body endpoint (no pydantic model): this is the hot path and
    # schema validation costs more than the query itself
    def _tx_commit_body(db_name: str, body: dict) -> dict:
        db = mgr.get(db_name)  # KeyError -> 404 at both call sites
        results, errors = [], []
        for stmt in body.get("statements", []):
            t0 = time.time()
            try:
                r = db.cypher(stmt.get("statement", ""),
                              stmt.get("parameters") or {})
                metrics.observe("nornicdb_query_seconds", time.time() - t0)
                results.append({
                    "columns": r.columns,
                    "data": [{"row": [_jsonable(v) for v in row],
                              "meta": []} for row in r.rows],
                    "stats": r.stats,
                })
            except (CypherSyntaxError,) as e:
                errors.append({"code": "Neo.ClientError.Statement.SyntaxError",
                               "message": str(e)})
                break
            except Exception as e:
                errors.append({"code": "Neo.ClientError.Statement.ExecutionFailed",
                               "message": str(e)})
                break
        return {"results": results, "errors": errors}

    @app.post("/db/{db_name}/tx/commit")
    async def tx_commit(db_name: str, request: Request,
                        _user=Depends(check_auth)):
        body = await request.json()
        try:
            return _tx_commit_body(db_name, body)
        except KeyError:
            raise HTTPException(404, f"database {db_name} not found")

    # ---- explicit transaction lifecycle (reference server_db.go:381-1226:
    # simplified semantics — statements execute eagerly, commit returns a
    # bookmark, rollback acknowledges) ----
    import itertools as _it
    _tx_counter = _it.count(1)
    _open_txs: Dict[str, tuple] = {}   # txid -> (db_name, executor, recorder)

    async def _run_statements(db_name: str, request: Request, executor=None):
        body = await request.json() if (await request.body()) else {}
        try:
            db = mgr.get(db_name)
        except KeyError:
            raise HTTPException(404, f"database {db_name} not found")
        results, errors = [], []
        runner = executor.execute if executor is not None else db.cypher
        for stmt in (body or {}).get("statements", []):
            try:
                r = runner(stmt.get("statement", ""),
                           stmt.get("parameters") or {})
                results.append({
                    "columns": r.columns,
                    "data": [{"row": [_jsonable(v) for v in row], "meta": []}
                             for row in r.rows],
                    "stats": r.stats})
            except Exception as e:
                code = ("Neo.ClientError.Statement.SyntaxError"
                        if isinstance(e, CypherSyntaxError)
                        else "Neo.ClientError.Statement.ExecutionFailed")
                errors.append({"code": code, "message": str(e)})
                break
        return results, errors

    @app.post("/db/{db_name}/tx")
    async def tx_open(db_name: str, request: Request, response: Response,
                      _user=Depends(check_auth)):
        txid = str(next(_tx_counter))
        try:
            ex, rec = mgr.get(db_name).begin_tx()
        except KeyError:
            raise HTTPException(404, f"database {db_name} not found")
        _open_txs[txid] = (db_name, ex, rec)
        results, errors = await _run_statements(db_name, request, executor=ex)
        response.headers["Location"] = f"/db/{db_name}/tx/{txid}"
        return {"results": results, "errors": errors,
                "commit": f"/db/{db_name}/tx/{txid}/commit",
                "transaction": {"expires": ""}}

    @app.post("/db/{db_name}/tx/{txid}")
    async def tx_execute(db_name: str, txid: str, request: Request,
                         _user=Depends(check_auth)):
        if txid not in _open_txs:
            raise HTTPException(
                404, f"transaction {txid} not found or already closed")
        _, ex, _rec = _open_txs[txid]
        results, errors = await _run_statements(db_name, request, executor=ex)
        return {"results": results, "errors": errors,
                "commit": f"/db/{db_name}/tx/{txid}/commit"}

    @app.post("/db/{db_name}/tx/{txid}/commit")
    async def tx_commit_open(db_name: str, txid: str, request: Request,
                             _user=Depends(check_auth)):
        ent = _open_txs.pop(txid, None)
        if ent is not None:
            _, ex, rec = ent
            results, errors = await _run_statements(db_name, request,
                                                    executor=ex)
            rec.commit()
        else:
            results, errors = await _run_statements(db_name, request)
        return {"results": results, "errors": errors,
                "lastBookmarks": [f"FB:bookmark-{txid}"]}

    @app.delete("/db/{db_name}/tx/{txid}")
    async def tx_rollback(db_name: str, txid: str, _user=Depends(check_auth)):
        ent = _open_txs.pop(txid, None)
        if ent is not None:
            ent[2].rollback()   # undo applied statements (reference
                                # BadgerTransaction.Rollback semantics)
        return {"results": [], "errors": []}

    # ---- multi-database management ----
    @app.post("/admin/databases/{name}")
    def create_db(name: str, _user=Depends(check_auth)):
        try:
            mgr.create(name)
        except ValueError as e:
            raise HTTPException(409, str(e))
        return {"created": name}

    @app.delete("/admin/databases/{name}")
    def drop_db(name: str, _user=Depends(check_auth)):
        try:
            mgr.drop(name)
        except (KeyError, ValueError) as e:
            raise HTTPException(400, str(e))
        return {"dropped": name}

    @app.get("/admin/databases")
    def list_dbs(_user=Depends(check_auth)):
        return {"databases": mgr.list()}

    @app.post("/admin/backup")
    def backup(body: Dict[str, Any] = None, _user=Depends(check_auth)):
        """Online backup. {"path": "/backups/x"} streams a consistent
        copy there (disk/LSM engine); without a path, snapshot-capable
        engines write their snapshot in place."""
        base = mgr._base
        path = (body or {}).get("path")
        if path and hasattr(base, "backup"):
            base.backup(path)
            return {"status": "backup written", "path": path}
        if hasattr(base, "snapshot"):
            base.snapshot()
            return {"status": "snapshot written"}
        if hasattr(base, "backup") and path is None:
            return {"status": "disk engine: POST {\"path\": ...} for an "
                              "online backup"}
        return {"status": "in-memory engine, nothing to snapshot"}

    # ---- NornicDB memory/search API ----
    @app.post("/nornicdb/store")
    def store(req: StoreRequest, db: str = None, _user=Depends(check_auth)):
        m = mgr.get(db).store(req.content, title=req.title,
                              memory_type=req.memory_type,
                              importance=req.importance, tags=req.tags,
                              metadata=req.metadata)
        return {"id": m.id}

    @app.post("/nornicdb/search")
    def search(req: SearchRequest, db: str = None, _user=Depends(check_auth)):
        d = mgr.get(db)
        qv = d.embedder.embed_query(req.query)
        res = d.search.search(query=req.query, query_vec=qv, k=req.limit,
                              labels=req.labels, mmr=req.mmr)
        return {"results": [{"id": r.id, "score": r.score,
                             "node": _jsonable(r.node)} for r in res]}

    @app.get("/nornicdb/similar/{node_id}")
    def similar(node_id: str, limit: int = 10, db: str = None,
                _user=Depends(check_auth)):
        d = mgr.get(db)
        try:
            node = d.engine.get_node(node_id)
        except Exception:
            raise HTTPException(404, "node not found")
        if node.embedding is None:
            raise HTTPException(400, "node has no embedding")
        res = d.search.vector_search(node.embedding, limit + 1)
        return {"results": [{"id": r.id, "score": r.score}
                            for r in res if r.id != node_id][:limit]}

    @app.post("/nornicdb/embed")
    def embed(body: Dict[str, Any], db: str = None, _user=Depends(check_auth)):
        texts = body.get("texts") or [body.get("text", "")]
        vecs = mgr.get(db).embedder.embed_batch(texts)
        return {"embeddings": [v.tolist() for v in vecs]}

    @app.post("/nornicdb/decay/run")
    def decay_run(db: str = None, _user=Depends(check_auth)):
        from ..cognitive.decay import DecayManager
        d = mgr.get(db)
        dm = getattr(d, "_decay", None) or DecayManager(d.engine)
        stats = dm.run_cycle()
        return stats

    # ---- GDPR (reference server_gdpr.go / pkg/retention) ----
    @app.get("/gdpr/export/{subject}")
    def gdpr_export(subject: str, db: str = None, _user=Depends(check_auth)):
        d = mgr.get(db)
        nodes = [
            _jsonable(n) for n in d.engine.all_nodes()
            if n.properties.get("subject") == subject
            or n.properties.get("user") == subject
        ]
        return {"subject": subject, "nodes": nodes}

    @app.delete("/gdpr/delete/{subject}")
    def gdpr_delete(subject: str, db: str = None, _user=Depends(check_auth)):
        d = mgr.get(db)
        deleted = 0
        for n in list(d.engine.all_nodes()):
            if (n.properties.get("subject") == subject
                    or n.properties.get("user") == subject):
                d.engine.detach_delete_node(n.id)
                deleted += 1
        return {"subject": subject, "deleted": deleted}

    # ---- MCP (JSON-RPC 2.0 + the reference's REST aliases) ----
    @app.post("/mcp")
    async def mcp_endpoint(request: Request):
        body = await request.json()
        return mcp.handle(body)

    @app.get("/mcp/health")
    def mcp_health():
        return {"status": "ok", "protocol": "mcp"}

    @app.post("/mcp/initialize")
    def mcp_initialize():
        return mcp.handle({"jsonrpc": "2.0", "id": 1,
                           "method": "initialize"})["result"]

    @app.get("/mcp/tools/list")
    def mcp_tools_list():
        return mcp.handle({"jsonrpc": "2.0", "id": 1,
                           "method": "tools/list"})["result"]

    @app.post("/mcp/tools/call")
    async def mcp_tools_call(request: Request):
        body = await request.json()
        out = mcp.handle({"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                          "params": body})
        return out.get("result", out)

    @app.get("/graphql/playground")
    def graphql_playground():
        return Response(
            "<!doctype html><title>GraphQL</title><body>"
            "<h3>NornicDB-AMD GraphQL</h3>"
            "<p>POST /graphql with {query, variables}; "
            "subscriptions stream from GET /graphql/stream (SSE).</p>"
            "<textarea id=q rows=8 cols=80>{ nodeCount }</textarea><br>"
            "<button onclick=\"fetch('/graphql',{method:'POST',"
            "headers:{'Content-Type':'application/json'},"
            "body:JSON.stringify({query:document.getElementById('q').value})})"
            ".then(r=>r.json()).then(d=>document.getElementById('o')"
            ".textContent=JSON.stringify(d,null,2))\">run</button>"
            "<pre id=o></pre></body>", media_type="text/html")

    # ---- Heimdall / Bifrost (reference pkg/heimdall/bifrost.go SSE) ----
    _heimdall = [None]

    def get_heimdall():
        if _heimdall[0] is None:
            from ..heimdall import HeimdallManager
            from ..models.heimdall import HeimdallConfig
            import torch
            cfg = None if torch.cuda.is_available() else HeimdallConfig.tiny()
            _heimdall[0] = HeimdallManager(mgr.get(), config=cfg)
        return _heimdall[0]

    @app.post("/bifrost/generate")
    def bifrost_generate(body: Dict[str, Any], _user=Depends(check_auth)):
        h = get_heimdall()
        text = h.generate(body.get("prompt", ""),
                          max_tokens=body.get("max_tokens"))
        return {"text": text, "stats": h.stats}

    @app.post("/bifrost/chat")
    def bifrost_chat(body: Dict[str, Any], _user=Depends(check_auth)):
        from ..heimdall import ChatMessage
        h = get_heimdall()
        msgs = [ChatMessage(m.get("role", "user"), m.get("content", ""))
                for m in body.get("messages", [])]
        return {"text": h.chat(msgs, max_tokens=body.get("max_tokens"))}

    @app.post("/bifrost/stream")
    def bifrost_stream(body: Dict[str, Any], _user=Depends(check_auth)):
        from starlette.responses import StreamingResponse
        h = get_heimdall()

        def sse():
            for tok in h.generate_stream(body.get("prompt", ""),
                                         max_tokens=body.get("max_tokens")):
                yield f"data: {tok}\n\n"
            yield "data: [DONE]\n\n"

        return StreamingResponse(sse(), media_type="text/event-stream")

    @app.get("/bifrost/metrics")
    def bifrost_metrics(_user=Depends(check_auth)):
        h = get_heimdall()
        return {"db": h.db_metrics(), "generation": h.stats,
                "plugins": h.plugin_health(),
                "tokens_per_second": h.tokens_per_second()}

    # reference route aliases (server_router.go /api/bifrost/*)
    @app.post("/api/bifrost/chat/completions")
    def bifrost_chat_completions(body: Dict[str, Any],
                                 _user=Depends(check_auth)):
        # OpenAI-shaped wrapper over chat
        out = bifrost_chat(body, _user)
        return {"object": "chat.completion",
                "choices": [{"index": 0, "finish_reason": "stop",
                             "message": {"role": "assistant",
                                         "content": out["text"]}}]}

    @app.get("/api/bifrost/status")
    def bifrost_status():
        try:
            h = get_heimdall()
            return {"available": True, "decoder": type(h.decoder).__name__
                    if getattr(h, "decoder", None) else "eager"}
        except Exception as e:
            return {"available": False, "reason": str(e)}

    @app.get("/api/bifrost/events")
    def bifrost_events(_user=Depends(check_auth)):
        from starlette.responses import StreamingResponse

        def sse():
            h = get_heimdall()
            import json as _json
            yield "data: " + _json.dumps(
                {"event": "status", "stats": h.stats}) + "\n\n"

        return StreamingResponse(sse(), media_type="text/event-stream")

    # ---- GraphQL (reference pkg/graphql) ----
    from .graphql import GraphQLExecutor

    _gql_cache = {}

    def _gql_for(db):
        d = mgr.get(db)
        if d.name not in _gql_cache:
            _gql_cache[d.name] = GraphQLExecutor(d)
        return _gql_cache[d.name]

    @app.post("/graphql")
    async def graphql_endpoint(request: Request, db: str = None):
        body = await request.json()
        return _gql_for(db).execute(body.get("query", ""),
                                    body.get("variables"))

    @app.get("/graphql/stream")
    async def graphql_stream(request: Request, db: str = None,
                             labels: str = None):
        """Subscription surface (reference resolvers/subscription_impl.go)
        as SSE: nodeCreated/nodeUpdated/nodeDeleted/relationship* events."""
        import asyncio
        import queue as _q

        broker = _gql_for(db).broker
        sub = broker.subscribe(labels.split(",") if labels else None)

        async def gen():
            try:
                while True:
                    if await request.is_disconnected():
                        break
                    try:
                        ev = sub.get_nowait()
                    except _q.Empty:
                        await asyncio.sleep(0.05)
                        continue
                    yield (f"event: {ev['event']}\n"
                           f"data: {json.dumps(ev['data'], default=str)}\n\n")
            finally:
                broker.unsubscribe(sub)

        return StreamingResponse(gen(), media_type="text/event-stream")

    # ---- Qdrant-compatible REST (reference pkg/qdrantgrpc) ----
    from .qdrant import QdrantRegistry, qdrant_router
    # engine-backed: collections/points persist as _QdrantCollection /
    # _QdrantPoint_* nodes and reload on boot (reference registry.go)
    try:
        _qeng = mgr.get().engine
    except Exception:
        _qeng = None
    app.state.qdrant = QdrantRegistry(engine=_qeng)
    app.include_router(qdrant_router(app.state.qdrant))

    app._tx_commit_body = _tx_commit_body  # reused by the ASGI fast path
    app._gql_for = _gql_for
    return _TxFastPath(app, auth)
