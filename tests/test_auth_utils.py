"""Auth, encryption, audit/retention, config and cache tests."""

import time

import pytest

from nornicdb_amd.auth import (AuthError, Authenticator, ROLE_ADMIN,
                               ROLE_READONLY, hash_password, verify_password)
from nornicdb_amd.storage import MemoryEngine, Node
from nornicdb_amd.utils import (AuditLog, Config, EncryptionManager, LRUCache,
                                RetentionManager, RetentionPolicy, load_config)


class TestAuth:
    def test_password_hashing(self):
        h = hash_password("s3cret")
        assert verify_password("s3cret", h)
        assert not verify_password("wrong", h)

    def test_login_and_roles(self):
        a = Authenticator(MemoryEngine())
        a.create_user("admin", "pw", ROLE_ADMIN)
        a.create_user("bob", "pw2", ROLE_READONLY)
        info = a.login("admin", "pw")
        assert info["role"] == ROLE_ADMIN
        with pytest.raises(AuthError):
            a.login("bob", "nope")
        assert a.authorize(ROLE_ADMIN, "admin")
        assert not a.authorize(ROLE_READONLY, "write")

    def test_token_roundtrip_and_expiry(self):
        now = [1000.0]
        a = Authenticator(MemoryEngine(), token_ttl=10, now_fn=lambda: now[0])
        a.create_user("u", "p", ROLE_READONLY)
        tok = a.issue_token("u", "p")
        claims = a.validate_token(tok)
        assert claims["sub"] == "u" and claims["role"] == ROLE_READONLY
        now[0] += 11
        with pytest.raises(AuthError):
            a.validate_token(tok)
        with pytest.raises(AuthError):
            a.validate_token(tok[:-3] + "xyz")

    def test_lockout(self):
        now = [0.0]
        a = Authenticator(MemoryEngine(), now_fn=lambda: now[0])
        a.create_user("u", "p")
        for _ in range(5):
            with pytest.raises(AuthError):
                a.login("u", "bad")
        with pytest.raises(AuthError, match="locked"):
            a.login("u", "p")  # correct pw but locked
        now[0] += 301
        assert a.login("u", "p")["username"] == "u"

    def test_ensure_admin(self):
        a = Authenticator(MemoryEngine())
        pw = a.ensure_admin()
        assert pw is not None
        assert a.ensure_admin() is None
        a.login("neo4j", pw)


class TestEncryption:
    def test_roundtrip_and_tamper(self):
        em = EncryptionManager("passphrase")
        ct = em.encrypt(b"hello world", aad=b"ctx")
        assert em.decrypt(ct, aad=b"ctx") == b"hello world"
        bad = bytearray(ct)
        bad[-1] ^= 1
        with pytest.raises(ValueError):
            em.decrypt(bytes(bad), aad=b"ctx")
        with pytest.raises(ValueError):
            em.decrypt(ct, aad=b"other")

    def test_key_rotation(self):
        em = EncryptionManager("old")
        ct_old = em.encrypt(b"data")
        em.rotate("new")
        ct_new = em.encrypt(b"data2")
        assert em.decrypt(ct_old) == b"data"
        assert em.decrypt(ct_new) == b"data2"


class TestAuditRetention:
    def test_audit_file(self, tmp_path):
        log = AuditLog(str(tmp_path / "audit.jsonl"))
        log.record("login", actor="u1")
        log.record("delete", actor="u2", target="n1")
        entries = list(log.entries())
        assert len(entries) == 2 and entries[0]["action"] == "login"
        log.close()

    def test_retention_and_legal_hold(self):
        eng = MemoryEngine()
        now = [100.0 * 86400]
        old = Node("old", ["Event"], {"created_at": 10 * 86400.0})
        held = Node("held", ["Event"], {"created_at": 10 * 86400.0})
        fresh = Node("fresh", ["Event"], {"created_at": now[0] - 100})
        for n in (old, held, fresh):
            eng.create_node(n)
        rm = RetentionManager(eng, now_fn=lambda: now[0])
        rm.add_policy(RetentionPolicy("Event", max_age_days=30))
        rm.legal_hold("held")
        stats = rm.enforce()
        assert stats["deleted"] == 1 and stats["held"] == 1
        assert not eng.has_node("old") and eng.has_node("held")

    def test_gdpr_erasure(self):
        eng = MemoryEngine()
        eng.create_node(Node("a", [], {"subject": "alice"}))
        eng.create_node(Node("b", [], {"subject": "bob"}))
        rm = RetentionManager(eng)
        assert rm.erase_subject("alice") == 1
        assert eng.has_node("b")


class TestConfig:
    def test_yaml_env_precedence(self, tmp_path):
        p = tmp_path / "nornicdb.yaml"
        p.write_text("bolt_port: 7777\nembedding_dims: 256\n"
                     "flags:\n  inference: true\n")
        cfg = load_config(str(p), env={"NORNICDB_BOLT_PORT": "8888",
                                       "NORNICDB_FLAG_EDGE_DECAY": "true"})
        assert cfg.bolt_port == 8888        # env beats yaml
        assert cfg.embedding_dims == 256    # yaml beats default
        assert cfg.flag("inference") and cfg.flag("edge_decay")
        cfg2 = load_config(str(p), env={}, overrides={"bolt_port": 9999})
        assert cfg2.bolt_port == 9999       # overrides beat all

    def test_defaults(self):
        cfg = load_config(env={})
        assert cfg.bolt_port == 7687
        assert cfg.flag("auto_embed")


class TestCache:
    def test_lru_ttl(self):
        now = [0.0]
        c = LRUCache(capacity=2, ttl=10, now_fn=lambda: now[0])
        c.put("a", 1)
        c.put("b", 2)
        assert c.get("a") == 1
        c.put("c", 3)  # evicts b (a was touched)
        assert c.get("b") is None
        now[0] += 11
        assert c.get("a") is None  # expired
