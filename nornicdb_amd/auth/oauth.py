"""OAuth2 / OIDC: authorization server + relying-party client.

Parity: reference pkg/auth/oauth.go (OAuthManager — auth-URL generation
with CSRF state, code exchange, userinfo fetch, local-user mapping with
role conversion, token refresh) and cmd/oauth-provider (a standalone
OAuth2 provider: /oauth2/v1/authorize with consent, /token with
authorization_code / refresh_token / client_credentials grants,
/userinfo, RFC 8414 discovery).

Both sides are transport-injectable so the full flow is testable
offline: the client's `fetch` hook can POST straight into the provider's
handlers (or an httpx client against a served app).
"""

from __future__ import annotations

import json
import secrets
import threading
import time
import urllib.parse
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

from . import AuthError, Authenticator

AUTH_CODE_TTL = 600.0
ACCESS_TOKEN_TTL = 3600.0
REFRESH_TOKEN_TTL = 30 * 86400.0
STATE_TTL = 600.0


@dataclass
class _Grant:
    username: str
    scope: str
    expires: float
    redirect_uri: str = ""
    roles: List[str] = field(default_factory=list)


class OAuthProvider:
    """Minimal OAuth2/OIDC authorization server
    (reference cmd/oauth-provider/main.go)."""

    def __init__(self, client_id: str, client_secret: str, issuer: str,
                 authenticator: Optional[Authenticator] = None,
                 now_fn=time.time):
        self.client_id = client_id
        self.client_secret = client_secret
        self.issuer = issuer.rstrip("/")
        self.auth = authenticator
        self.now = now_fn
        self._lock = threading.Lock()
        self._codes: Dict[str, _Grant] = {}
        self._access: Dict[str, _Grant] = {}
        self._refresh: Dict[str, _Grant] = {}

    # ---- endpoints (framework-agnostic: dict in, (status, dict|str) out) --
    def authorize(self, params: Dict[str, str]) -> Tuple[int, Dict]:
        """GET /oauth2/v1/authorize — validates the request; the caller
        renders consent and then calls `consent` (reference renders an
        HTML form; handler split is identical)."""
        if params.get("client_id") != self.client_id:
            return 400, {"error": "invalid_client"}
        if params.get("response_type") != "code":
            return 400, {"error": "unsupported_response_type"}
        if not params.get("redirect_uri"):
            return 400, {"error": "invalid_request",
                         "error_description": "redirect_uri required"}
        return 200, {
            "consent_required": True,
            "client_id": self.client_id,
            "redirect_uri": params["redirect_uri"],
            "scope": params.get("scope", "openid profile"),
            "state": params.get("state", ""),
        }

    def consent(self, username: str, password: str, redirect_uri: str,
                state: str, scope: str = "openid profile") -> Tuple[int, Dict]:
        """POST /oauth2/v1/authorize/consent — authenticates the resource
        owner and issues an authorization code."""
        roles = ["readwrite"]
        if self.auth is not None:
            try:
                claims = self.auth.login(username, password)
                roles = [claims.get("role", "readwrite")]
            except AuthError as e:
                return 401, {"error": "access_denied",
                             "error_description": str(e)}
        code = secrets.token_urlsafe(32)
        with self._lock:
            self._codes[code] = _Grant(username, scope,
                                       self.now() + AUTH_CODE_TTL,
                                       redirect_uri, roles)
        sep = "&" if "?" in redirect_uri else "?"
        return 302, {"location": f"{redirect_uri}{sep}code={code}"
                                 f"&state={urllib.parse.quote(state)}",
                     "code": code, "state": state}

    def token(self, form: Dict[str, str]) -> Tuple[int, Dict]:
        """POST /oauth2/v1/token — authorization_code, refresh_token and
        client_credentials grants."""
        if (form.get("client_id") != self.client_id
                or form.get("client_secret") != self.client_secret):
            return 401, {"error": "invalid_client"}
        grant_type = form.get("grant_type")
        now = self.now()
        if grant_type == "authorization_code":
            with self._lock:
                g = self._codes.pop(form.get("code", ""), None)
            if g is None or g.expires < now:
                return 400, {"error": "invalid_grant"}
            ru = form.get("redirect_uri", "")
            if ru and g.redirect_uri and ru != g.redirect_uri:
                return 400, {"error": "invalid_grant",
                             "error_description": "redirect_uri mismatch"}
            return 200, self._issue(g)
        if grant_type == "refresh_token":
            with self._lock:
                g = self._refresh.pop(form.get("refresh_token", ""), None)
            if g is None or g.expires < now:
                return 400, {"error": "invalid_grant"}
            return 200, self._issue(g)
        if grant_type == "client_credentials":
            g = _Grant("service:" + self.client_id, form.get("scope", ""),
                       0.0, roles=["readwrite"])
            return 200, self._issue(g, refresh=False)
        return 400, {"error": "unsupported_grant_type"}

    def _issue(self, g: _Grant, refresh: bool = True) -> Dict:
        at = secrets.token_urlsafe(32)
        now = self.now()
        with self._lock:
            self._access[at] = _Grant(g.username, g.scope,
                                      now + ACCESS_TOKEN_TTL, roles=g.roles)
        out = {"access_token": at, "token_type": "Bearer",
               "expires_in": int(ACCESS_TOKEN_TTL), "scope": g.scope}
        if refresh:
            rt = secrets.token_urlsafe(32)
            with self._lock:
                self._refresh[rt] = _Grant(g.username, g.scope,
                                           now + REFRESH_TOKEN_TTL,
                                           roles=g.roles)
            out["refresh_token"] = rt
        return out

    def userinfo(self, authorization: str) -> Tuple[int, Dict]:
        """GET /oauth2/v1/userinfo (Bearer token)."""
        if not authorization.startswith("Bearer "):
            return 401, {"error": "invalid_token"}
        with self._lock:
            g = self._access.get(authorization[7:])
        if g is None or g.expires < self.now():
            return 401, {"error": "invalid_token"}
        return 200, {"sub": g.username, "email": f"{g.username}@nornicdb",
                     "preferred_username": g.username, "roles": g.roles}

    def discovery(self) -> Dict:
        """RFC 8414 / OIDC discovery document."""
        base = self.issuer
        return {
            "issuer": base,
            "authorization_endpoint": f"{base}/oauth2/v1/authorize",
            "token_endpoint": f"{base}/oauth2/v1/token",
            "userinfo_endpoint": f"{base}/oauth2/v1/userinfo",
            "response_types_supported": ["code"],
            "grant_types_supported": ["authorization_code", "refresh_token",
                                      "client_credentials"],
            "scopes_supported": ["openid", "profile", "email"],
            "token_endpoint_auth_methods_supported": ["client_secret_post"],
        }

    def cleanup(self) -> int:
        now = self.now()
        removed = 0
        with self._lock:
            for store in (self._codes, self._access, self._refresh):
                for k in list(store):
                    if store[k].expires and store[k].expires < now:
                        del store[k]
                        removed += 1
        return removed


# role mapping (reference ConvertOAuthRoles, oauth.go:452)
_ROLE_MAP = {"admin": "admin", "administrator": "admin",
             "readwrite": "readwrite", "writer": "readwrite",
             "editor": "readwrite", "readonly": "readonly",
             "reader": "readonly", "viewer": "readonly"}


def convert_oauth_roles(roles: List[str]) -> List[str]:
    out = [_ROLE_MAP[r.lower()] for r in roles if r.lower() in _ROLE_MAP]
    return out or ["readonly"]


class OAuthClientManager:
    """Relying-party side (reference pkg/auth/oauth.go OAuthManager):
    builds the auth URL with CSRF state, exchanges codes, maps the
    provider's userinfo onto local users with converted roles.

    `fetch(method, url, data, headers) -> (status, dict)` is injectable:
    production uses an HTTP client; offline tests POST straight into an
    OAuthProvider instance.
    """

    def __init__(self, authenticator: Authenticator, issuer: str,
                 client_id: str, client_secret: str, callback_url: str,
                 fetch: Callable = None, now_fn=time.time):
        self.auth = authenticator
        self.issuer = issuer.rstrip("/")
        self.client_id = client_id
        self.client_secret = client_secret
        self.callback_url = callback_url
        self.fetch = fetch or _http_fetch
        self.now = now_fn
        self._states: Dict[str, float] = {}
        self._lock = threading.Lock()

    def is_configured(self) -> bool:
        return bool(self.issuer and self.client_id and self.client_secret
                    and self.callback_url)

    def generate_auth_url(self) -> Tuple[str, str]:
        state = secrets.token_urlsafe(24)
        with self._lock:
            self._states[state] = self.now() + STATE_TTL
            for s in list(self._states):
                if self._states[s] < self.now():
                    del self._states[s]
        q = urllib.parse.urlencode({
            "response_type": "code", "client_id": self.client_id,
            "redirect_uri": self.callback_url, "scope": "openid profile",
            "state": state})
        return f"{self.issuer}/oauth2/v1/authorize?{q}", state

    def validate_state(self, state: str) -> None:
        with self._lock:
            exp = self._states.pop(state, None)
        if exp is None:
            raise AuthError("invalid OAuth state (CSRF)")
        if exp < self.now():
            raise AuthError("expired OAuth state")

    def exchange_code(self, code: str) -> Dict:
        status, tok = self.fetch("POST", f"{self.issuer}/oauth2/v1/token", {
            "grant_type": "authorization_code", "code": code,
            "client_id": self.client_id, "client_secret": self.client_secret,
            "redirect_uri": self.callback_url}, {})
        if status != 200:
            raise AuthError(f"token exchange failed: {tok}")
        return tok

    def get_userinfo(self, access_token: str) -> Dict:
        status, info = self.fetch(
            "GET", f"{self.issuer}/oauth2/v1/userinfo", None,
            {"Authorization": f"Bearer {access_token}"})
        if status != 200:
            raise AuthError(f"userinfo failed: {info}")
        return info

    def handle_callback(self, code: str, state: str) -> Dict:
        """Full callback: state check -> code exchange -> userinfo ->
        local user upsert with converted roles -> local session token.
        Returns {username, role, token, oauth}."""
        self.validate_state(state)
        tok = self.exchange_code(code)
        info = self.get_userinfo(tok["access_token"])
        username = info.get("preferred_username") or info.get("sub")
        if not username:
            raise AuthError("userinfo missing subject")
        role = convert_oauth_roles(info.get("roles", []))[0]
        # upsert local user with an unguessable password (OAuth users log
        # in via the provider only — reference HandleCallback behavior)
        try:
            self.auth.get_user(username)  # type: ignore[attr-defined]
            self.auth.set_role(username, role)
        except (AttributeError, AuthError, KeyError):
            try:
                self.auth.create_user(username, secrets.token_urlsafe(24),
                                      role=role)
            except AuthError:
                self.auth.set_role(username, role)
        token = self.auth.issue_token_for(username)  \
            if hasattr(self.auth, "issue_token_for") else None
        return {"username": username, "role": role, "token": token,
                "oauth": tok}


def _http_fetch(method: str, url: str, data, headers) -> Tuple[int, Dict]:
    import urllib.request
    body = urllib.parse.urlencode(data).encode() if data else None
    req = urllib.request.Request(url, data=body, headers=headers or {},
                                 method=method)
    try:
        with urllib.request.urlopen(req, timeout=10) as resp:
            return resp.status, json.loads(resp.read() or b"{}")
    except urllib.error.HTTPError as e:  # pragma: no cover - network path
        try:
            return e.code, json.loads(e.read() or b"{}")
        except Exception:
            return e.code, {"error": str(e)}
