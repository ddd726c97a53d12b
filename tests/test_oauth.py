"""OAuth2/OIDC tests: provider endpoints + relying-party client flow.

Mirrors the reference (pkg/auth/oauth.go + cmd/oauth-provider): consent
-> code -> token exchange -> userinfo -> local user upsert with role
conversion, plus refresh and client_credentials grants, CSRF state
validation and discovery.
"""

import pytest

from nornicdb_amd.auth import AuthError, Authenticator
from nornicdb_amd.auth.oauth import (OAuthClientManager, OAuthProvider,
                                     convert_oauth_roles)
from nornicdb_amd.storage import MemoryEngine


@pytest.fixture
def auth():
    a = Authenticator(MemoryEngine())
    a.create_user("alice", "pw-alice", role="admin")
    return a


@pytest.fixture
def provider(auth):
    return OAuthProvider("cid", "csecret", "http://issuer", auth)


def _client(auth, provider):
    def fetch(method, url, data, headers):
        if url.endswith("/token"):
            return provider.token(data)
        if url.endswith("/userinfo"):
            return provider.userinfo(headers.get("Authorization", ""))
        raise AssertionError(url)
    return OAuthClientManager(auth, "http://issuer", "cid", "csecret",
                              "http://app/cb", fetch=fetch)


class TestProvider:
    def test_discovery(self, provider):
        d = provider.discovery()
        assert d["issuer"] == "http://issuer"
        assert d["token_endpoint"].endswith("/oauth2/v1/token")
        assert "authorization_code" in d["grant_types_supported"]

    def test_authorize_validations(self, provider):
        s, b = provider.authorize({"client_id": "wrong",
                                   "response_type": "code",
                                   "redirect_uri": "http://app/cb"})
        assert s == 400 and b["error"] == "invalid_client"
        s, b = provider.authorize({"client_id": "cid",
                                   "response_type": "token",
                                   "redirect_uri": "x"})
        assert s == 400 and b["error"] == "unsupported_response_type"
        s, b = provider.authorize({"client_id": "cid",
                                   "response_type": "code",
                                   "redirect_uri": "http://app/cb",
                                   "state": "xyz"})
        assert s == 200 and b["consent_required"]

    def test_consent_requires_valid_credentials(self, provider):
        s, b = provider.consent("alice", "WRONG", "http://app/cb", "st")
        assert s == 401
        s, b = provider.consent("alice", "pw-alice", "http://app/cb", "st")
        assert s == 302 and "code=" in b["location"] and "state=st" in b["location"]

    def test_code_single_use_and_redirect_match(self, provider):
        _, b = provider.consent("alice", "pw-alice", "http://app/cb", "st")
        code = b["code"]
        s, tok = provider.token({"grant_type": "authorization_code",
                                 "code": code, "client_id": "cid",
                                 "client_secret": "csecret",
                                 "redirect_uri": "http://app/cb"})
        assert s == 200 and tok["token_type"] == "Bearer"
        # second use fails
        s, b2 = provider.token({"grant_type": "authorization_code",
                                "code": code, "client_id": "cid",
                                "client_secret": "csecret"})
        assert s == 400 and b2["error"] == "invalid_grant"

    def test_refresh_and_client_credentials(self, provider):
        _, b = provider.consent("alice", "pw-alice", "http://app/cb", "st")
        _, tok = provider.token({"grant_type": "authorization_code",
                                 "code": b["code"], "client_id": "cid",
                                 "client_secret": "csecret"})
        s, tok2 = provider.token({"grant_type": "refresh_token",
                                  "refresh_token": tok["refresh_token"],
                                  "client_id": "cid",
                                  "client_secret": "csecret"})
        assert s == 200 and tok2["access_token"] != tok["access_token"]
        s, cc = provider.token({"grant_type": "client_credentials",
                                "client_id": "cid",
                                "client_secret": "csecret"})
        assert s == 200 and "refresh_token" not in cc

    def test_userinfo(self, provider):
        _, b = provider.consent("alice", "pw-alice", "http://app/cb", "st")
        _, tok = provider.token({"grant_type": "authorization_code",
                                 "code": b["code"], "client_id": "cid",
                                 "client_secret": "csecret"})
        s, info = provider.userinfo(f"Bearer {tok['access_token']}")
        assert s == 200 and info["preferred_username"] == "alice"
        assert info["roles"] == ["admin"]
        s, _ = provider.userinfo("Bearer nope")
        assert s == 401

    def test_bad_client_secret(self, provider):
        s, b = provider.token({"grant_type": "client_credentials",
                               "client_id": "cid", "client_secret": "nope"})
        assert s == 401


class TestRoleConversion:
    def test_mappings(self):
        assert convert_oauth_roles(["Administrator"]) == ["admin"]
        assert convert_oauth_roles(["editor", "viewer"]) == ["readwrite",
                                                             "readonly"]
        assert convert_oauth_roles(["unknown"]) == ["readonly"]
        assert convert_oauth_roles([]) == ["readonly"]


class TestClientFlow:
    def test_full_callback_flow(self, auth, provider):
        cm = _client(auth, provider)
        url, state = cm.generate_auth_url()
        assert "state=" in url and "client_id=cid" in url
        # user consents at the provider
        _, b = provider.consent("alice", "pw-alice", "http://app/cb", state)
        out = cm.handle_callback(b["code"], state)
        assert out["username"] == "alice"
        assert out["role"] == "admin"
        assert out["token"]  # local session JWT
        claims = auth.validate_token(out["token"])
        assert claims["sub"] == "alice" and claims["role"] == "admin"

    def test_state_csrf_rejected(self, auth, provider):
        cm = _client(auth, provider)
        _, state = cm.generate_auth_url()
        _, b = provider.consent("alice", "pw-alice", "http://app/cb", state)
        with pytest.raises(AuthError):
            cm.handle_callback(b["code"], "forged-state")
        # and states are single-use
        out = cm.handle_callback(b["code"], state)
        assert out["username"] == "alice"
        with pytest.raises(AuthError):
            cm.validate_state(state)

    def test_new_oauth_user_provisioned(self, auth, provider):
        auth.create_user("bob", "pw-bob", role="readonly")
        cm = _client(auth, provider)
        _, state = cm.generate_auth_url()
        _, b = provider.consent("bob", "pw-bob", "http://app/cb", state)
        out = cm.handle_callback(b["code"], state)
        assert out["username"] == "bob" and out["role"] == "readonly"


class TestHttpRoutes:
    def test_provider_over_http(self, monkeypatch, tmp_path):
        from fastapi.testclient import TestClient
        from nornicdb_amd.db import open_db
        from nornicdb_amd.server.http import create_app
        monkeypatch.setenv("NORNICDB_OAUTH_PROVIDER_ENABLED", "1")
        monkeypatch.setenv("NORNICDB_OAUTH_CLIENT_ID", "cid")
        monkeypatch.setenv("NORNICDB_OAUTH_CLIENT_SECRET", "cs")
        mgr = open_db()
        a = Authenticator(MemoryEngine())
        a.create_user("alice", "pw", role="readwrite")
        app = create_app(mgr, auth=a)
        c = TestClient(app)
        d = c.get("/.well-known/openid-configuration").json()
        assert d["grant_types_supported"]
        r = c.get("/oauth2/v1/authorize", params={
            "client_id": "cid", "response_type": "code",
            "redirect_uri": "http://app/cb", "state": "s1"})
        assert r.status_code == 200 and r.json()["consent_required"]
        r = c.post("/oauth2/v1/authorize/consent", data={
            "username": "alice", "password": "pw",
            "redirect_uri": "http://app/cb", "state": "s1"},
            follow_redirects=False)
        assert r.status_code == 302
        loc = r.headers["location"]
        code = loc.split("code=")[1].split("&")[0]
        r = c.post("/oauth2/v1/token", data={
            "grant_type": "authorization_code", "code": code,
            "client_id": "cid", "client_secret": "cs"})
        assert r.status_code == 200
        at = r.json()["access_token"]
        r = c.get("/oauth2/v1/userinfo",
                  headers={"Authorization": f"Bearer {at}"})
        assert r.json()["preferred_username"] == "alice"
        assert c.get("/auth/config").json()["oauth"] is True
        mgr.close()
