"""Authentication & RBAC.

Parity: reference pkg/auth/auth.go — users stored in the `system` database
(:678-707), salted password hashing, JWT-style token issue/validate
(:362,:582), roles admin/readwrite/readonly with granular permissions,
failed-login lockout (:842), audit hooks.

Token format: HMAC-SHA256 signed (header.payload.sig, JWT-compatible
layout) — no external jwt dependency.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import json
import os
import secrets
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..storage.types import Engine, Node, NotFoundError

ROLE_ADMIN = "admin"
ROLE_READWRITE = "readwrite"
ROLE_READONLY = "readonly"

PERMISSIONS = {
    ROLE_ADMIN: {"read", "write", "schema", "admin", "dbms"},
    ROLE_READWRITE: {"read", "write"},
    ROLE_READONLY: {"read"},
}


class AuthError(Exception):
    pass


def _b64(b: bytes) -> str:
    return base64.urlsafe_b64encode(b).rstrip(b"=").decode()


def _unb64(s: str) -> bytes:
    return base64.urlsafe_b64decode(s + "=" * (-len(s) % 4))


def hash_password(password: str, salt: bytes = None, iterations: int = 100_000):
    """PBKDF2-HMAC-SHA256 (reference uses bcrypt; same security class)."""
    salt = salt or secrets.token_bytes(16)
    dk = hashlib.pbkdf2_hmac("sha256", password.encode(), salt, iterations)
    return f"pbkdf2${iterations}${_b64(salt)}${_b64(dk)}"


def verify_password(password: str, stored: str) -> bool:
    try:
        _, iters, salt, dk = stored.split("$")
        calc = hashlib.pbkdf2_hmac("sha256", password.encode(),
                                   _unb64(salt), int(iters))
        return hmac.compare_digest(calc, _unb64(dk))
    except Exception:
        return False


class Authenticator:
    LOCKOUT_AFTER = 5
    LOCKOUT_SECONDS = 300.0

    def __init__(self, system_engine: Engine, secret: bytes = None,
                 token_ttl: float = 8 * 3600, now_fn=time.time):
        self.engine = system_engine
        self.secret = secret or secrets.token_bytes(32)
        self.ttl = token_ttl
        self.now = now_fn
        self._lock = threading.Lock()
        self._failures: Dict[str, List[float]] = {}
        self.audit_hook = None

    # ---- user management (users live in the system DB) ----
    def _user_id(self, username: str) -> str:
        return f"user:{username}"

    def create_user(self, username: str, password: str,
                    role: str = ROLE_READONLY) -> None:
        if role not in PERMISSIONS:
            raise AuthError(f"unknown role {role}")
        node = Node(id=self._user_id(username), labels=["User"],
                    properties={"username": username,
                                "password_hash": hash_password(password),
                                "role": role, "created_at": self.now(),
                                "suspended": False})
        self.engine.create_node(node)
        self._audit("user_created", username)

    def ensure_admin(self, username: str = "neo4j", password: str = None) -> Optional[str]:
        """Create the initial admin if missing; returns generated password."""
        try:
            self.engine.get_node(self._user_id(username))
            return None
        except NotFoundError:
            pw = password or secrets.token_urlsafe(12)
            self.create_user(username, pw, ROLE_ADMIN)
            return pw

    def set_password(self, username: str, new_password: str):
        node = self.engine.get_node(self._user_id(username))
        node.properties["password_hash"] = hash_password(new_password)
        self.engine.update_node(node)
        self._audit("password_changed", username)

    def set_role(self, username: str, role: str):
        if role not in PERMISSIONS:
            raise AuthError(f"unknown role {role}")
        node = self.engine.get_node(self._user_id(username))
        node.properties["role"] = role
        self.engine.update_node(node)

    def suspend_user(self, username: str, suspended: bool = True):
        node = self.engine.get_node(self._user_id(username))
        node.properties["suspended"] = suspended
        self.engine.update_node(node)

    def delete_user(self, username: str):
        self.engine.detach_delete_node(self._user_id(username))
        self._audit("user_deleted", username)

    def list_users(self) -> List[Dict]:
        return [{"username": n.properties["username"],
                 "role": n.properties["role"],
                 "suspended": n.properties.get("suspended", False)}
                for n in self.engine.get_nodes_by_label("User")]

    # ---- login / lockout ----
    def login(self, username: str, password: str) -> Dict:
        with self._lock:
            fails = [t for t in self._failures.get(username, [])
                     if self.now() - t < self.LOCKOUT_SECONDS]
            self._failures[username] = fails
            if len(fails) >= self.LOCKOUT_AFTER:
                self._audit("login_locked_out", username)
                raise AuthError("account locked; try again later")
        try:
            node = self.engine.get_node(self._user_id(username))
        except NotFoundError:
            self._fail(username)
            raise AuthError("invalid credentials")
        if node.properties.get("suspended"):
            raise AuthError("account suspended")
        if not verify_password(password, node.properties["password_hash"]):
            self._fail(username)
            raise AuthError("invalid credentials")
        with self._lock:
            self._failures.pop(username, None)
        self._audit("login_ok", username)
        return {"username": username, "role": node.properties["role"]}

    def _fail(self, username):
        with self._lock:
            self._failures.setdefault(username, []).append(self.now())
        self._audit("login_failed", username)

    # ---- tokens ----
    def issue_token(self, username: str, password: str) -> str:
        info = self.login(username, password)
        header = _b64(json.dumps({"alg": "HS256", "typ": "JWT"}).encode())
        payload = _b64(json.dumps({
            "sub": info["username"], "role": info["role"],
            "iat": int(self.now()), "exp": int(self.now() + self.ttl),
        }).encode())
        sig = _b64(hmac.new(self.secret, f"{header}.{payload}".encode(),
                            hashlib.sha256).digest())
        return f"{header}.{payload}.{sig}"

    def get_user(self, username: str) -> Dict:
        try:
            n = self.engine.get_node(self._user_id(username))
        except NotFoundError:
            raise AuthError(f"no such user {username!r}")
        return {"username": username, "role": n.properties["role"],
                "suspended": n.properties.get("suspended", False)}

    def issue_token_for(self, username: str) -> str:
        """Server-side token issuance WITHOUT a password — the OAuth
        callback path, where the identity was established by the
        provider (reference oauth.go HandleCallback issues the local
        JWT the same way)."""
        info = self.get_user(username)
        if info.get("suspended"):
            raise AuthError("account suspended")
        header = _b64(json.dumps({"alg": "HS256", "typ": "JWT"}).encode())
        payload = _b64(json.dumps({
            "sub": info["username"], "role": info["role"],
            "iat": int(self.now()), "exp": int(self.now() + self.ttl),
        }).encode())
        sig = _b64(hmac.new(self.secret, f"{header}.{payload}".encode(),
                            hashlib.sha256).digest())
        return f"{header}.{payload}.{sig}"

    def validate_token(self, token: str) -> Dict:
        try:
            header, payload, sig = token.split(".")
        except ValueError:
            raise AuthError("malformed token")
        expect = _b64(hmac.new(self.secret, f"{header}.{payload}".encode(),
                               hashlib.sha256).digest())
        if not hmac.compare_digest(sig, expect):
            raise AuthError("bad signature")
        claims = json.loads(_unb64(payload))
        if claims.get("exp", 0) < self.now():
            raise AuthError("token expired")
        return claims

    # ---- authorization ----
    def authorize(self, role: str, permission: str) -> bool:
        return permission in PERMISSIONS.get(role, set())

    def require(self, claims: Dict, permission: str):
        if not self.authorize(claims.get("role", ""), permission):
            raise AuthError(f"permission {permission} denied for role "
                            f"{claims.get('role')}")

    def _audit(self, event: str, username: str):
        if self.audit_hook:
            try:
                self.audit_hook(event, username)
            except Exception:
                pass
