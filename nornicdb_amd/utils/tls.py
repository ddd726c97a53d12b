"""TLS helpers (reference pkg/security/middleware.go — bolt+s / https).

make_ssl_context(cert, key) builds the server context used by BOTH the
Bolt listener (asyncio ssl) and uvicorn (ssl_certfile/ssl_keyfile);
ensure_self_signed() generates a development certificate with the
system openssl binary when none is configured.
"""

from __future__ import annotations

import os
import ssl
import subprocess


def make_ssl_context(certfile: str, keyfile: str) -> ssl.SSLContext:
    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
    ctx.minimum_version = ssl.TLSVersion.TLSv1_2
    ctx.load_cert_chain(certfile, keyfile)
    return ctx


def ensure_self_signed(dir_: str, cn: str = "localhost"):
    """Generate (once) a self-signed cert+key under dir_; returns paths."""
    os.makedirs(dir_, exist_ok=True)
    cert = os.path.join(dir_, "tls-cert.pem")
    key = os.path.join(dir_, "tls-key.pem")
    if not (os.path.exists(cert) and os.path.exists(key)):
        subprocess.run(
            ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
             "-keyout", key, "-out", cert, "-days", "825",
             "-subj", f"/CN={cn}",
             "-addext", f"subjectAltName=DNS:{cn},IP:127.0.0.1"],
            check=True, capture_output=True)
    return cert, key
