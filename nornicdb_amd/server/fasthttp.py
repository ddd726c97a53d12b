"""Minimal asyncio HTTP/1.1 server for the ASGI app — the serving path.

uvicorn on this stack runs pure-Python h11 (no httptools/uvloop wheels),
costing ~200 us/request of parse/state-machine overhead on the hot
tx-commit / GraphQL routes. This is a hand-rolled asyncio.Protocol that
speaks exactly the HTTP/1.1 subset our clients use (the same approach as
bolt/server.py for Bolt) and drives the SAME ASGI app, so every route —
REST, GraphQL, SSE streams, the console SPA — is served unchanged.

Replaces the reference's Go net/http serving layer (cmd/nornicdb serve,
pkg/server/server_router.go) with an asyncio-native one. Supports: keep-alive,
Content-Length bodies, chunked request decoding, streamed (chunked or
close-delimited) responses for SSE, TLS via loop.start_server(ssl=...).
uvicorn remains available behind NORNICDB_HTTP_SERVER=uvicorn.
"""

from __future__ import annotations

import asyncio
from typing import Optional
from urllib.parse import unquote_to_bytes

_KEEPALIVE_MAX = 10_000


class _Disconnect(Exception):
    pass


class AsgiHttpProtocol(asyncio.Protocol):
    __slots__ = ("app", "transport", "buf", "task", "peer", "is_tls",
                 "_closed", "_wake", "_can_write")

    def __init__(self, app, is_tls: bool = False):
        self.app = app
        self.transport = None
        self.buf = bytearray()
        self.task = None
        self.peer = ("", 0)
        self.is_tls = is_tls
        self._closed = False

    # ---- transport plumbing ----
    def connection_made(self, transport):
        self.transport = transport
        try:
            sock = transport.get_extra_info("socket")
            if sock is not None:
                import socket as _s
                sock.setsockopt(_s.IPPROTO_TCP, _s.TCP_NODELAY, 1)
        except OSError:
            pass
        self.peer = transport.get_extra_info("peername") or ("", 0)
        self._wake = asyncio.Event()
        self._can_write = asyncio.Event()
        self._can_write.set()
        self.task = asyncio.get_event_loop().create_task(self._serve())

    def data_received(self, data):
        self.buf += data
        self._wake.set()

    # asyncio flow control: the loop calls these when the socket send
    # buffer fills — streaming responses to slow clients block in
    # _drain instead of growing the transport buffer unboundedly
    def pause_writing(self):
        self._can_write.clear()

    def resume_writing(self):
        self._can_write.set()

    def eof_received(self):
        self._closed = True
        self._wake.set()
        return False

    def connection_lost(self, exc):
        self._closed = True
        self._wake.set()
        if self.task is not None:
            self.task.cancel()

    async def _read_more(self):
        if self._closed:
            raise _Disconnect
        self._wake.clear()
        await self._wake.wait()
        if self._closed and not self.buf:
            raise _Disconnect

    # ---- request loop ----
    async def _serve(self):
        try:
            for _ in range(_KEEPALIVE_MAX):
                keep = await self._one_request()
                if not keep:
                    break
        except (_Disconnect, asyncio.CancelledError, ConnectionError):
            pass
        except Exception:
            try:
                self.transport.write(
                    b"HTTP/1.1 500 Internal Server Error\r\n"
                    b"content-length: 0\r\nconnection: close\r\n\r\n")
            except Exception:
                pass
        finally:
            try:
                self.transport.close()
            except Exception:
                pass

    async def _one_request(self) -> bool:
        # headers
        while True:
            end = self.buf.find(b"\r\n\r\n")
            if end >= 0:
                break
            if len(self.buf) > 65536:
                self.transport.write(
                    b"HTTP/1.1 431 Request Header Fields Too Large\r\n"
                    b"content-length: 0\r\nconnection: close\r\n\r\n")
                return False
            await self._read_more()
        head = bytes(self.buf[:end])
        del self.buf[:end + 4]
        lines = head.split(b"\r\n")
        try:
            method, target, version = lines[0].split(b" ", 2)
        except ValueError:
            self.transport.write(b"HTTP/1.1 400 Bad Request\r\n"
                                 b"content-length: 0\r\n\r\n")
            return False
        headers = []
        clen = 0
        chunked = False
        connection = b""
        for ln in lines[1:]:
            i = ln.find(b":")
            if i <= 0:
                continue
            k = ln[:i].lower()
            v = ln[i + 1:].strip()
            headers.append((k, v))
            if k == b"content-length":
                try:
                    clen = int(v)
                except ValueError:
                    clen = 0
            elif k == b"transfer-encoding" and b"chunked" in v.lower():
                chunked = True
            elif k == b"connection":
                connection = v.lower()
            elif k == b"expect" and v.lower() == b"100-continue":
                self.transport.write(b"HTTP/1.1 100 Continue\r\n\r\n")
        if b"?" in target:
            raw_path, _, qs = target.partition(b"?")
        else:
            raw_path, qs = target, b""
        body = await (self._read_chunked() if chunked
                      else self._read_body(clen))
        keep = not (connection == b"close" or version == b"HTTP/1.0")

        scope = {
            "type": "http",
            "asgi": {"version": "3.0", "spec_version": "2.3"},
            "http_version": "1.1",
            "method": method.decode("latin1"),
            "scheme": "https" if self.is_tls else "http",
            "path": unquote_to_bytes(raw_path).decode("latin1"),
            "raw_path": raw_path,
            "query_string": qs,
            "root_path": "",
            "headers": headers,
            "client": (self.peer[0], self.peer[1] or 0),
            "server": ("0.0.0.0", 0),
        }
        sent = {"v": False}

        async def receive():
            if not sent["v"]:
                sent["v"] = True
                return {"type": "http.request", "body": body,
                        "more_body": False}
            # starlette's StreamingResponse awaits a second receive() as a
            # disconnect listener — block until the client actually goes
            # away (returning http.disconnect immediately aborts streams)
            while not self._closed:
                self._wake.clear()
                await self._wake.wait()
            return {"type": "http.disconnect"}

        state = {"started": False, "clen": None, "chunked_out": False,
                 "keep": keep}
        w = self.transport.write

        async def send(msg):
            t = msg["type"]
            if t == "http.response.start":
                status = msg["status"]
                hdrs = list(msg.get("headers", []))
                has_len = any(k.lower() == b"content-length"
                              for k, _ in hdrs)
                parts = [b"HTTP/1.1 " + str(status).encode() + b" " +
                         _REASON.get(status, b"OK") + b"\r\n"]
                for k, v in hdrs:
                    parts.append(k + b": " + v + b"\r\n")
                if not has_len:
                    # length unknown (streaming): chunked keeps the
                    # connection reusable
                    state["chunked_out"] = True
                    parts.append(b"transfer-encoding: chunked\r\n")
                parts.append(b"connection: keep-alive\r\n" if state["keep"]
                             else b"connection: close\r\n")
                parts.append(b"\r\n")
                w(b"".join(parts))
                state["started"] = True
            elif t == "http.response.body":
                chunk = msg.get("body", b"") or b""
                if state["chunked_out"]:
                    if chunk:
                        w(b"%x\r\n" % len(chunk) + chunk + b"\r\n")
                    if not msg.get("more_body"):
                        w(b"0\r\n\r\n")
                else:
                    if chunk:
                        w(chunk)
                if self.transport.is_closing():
                    raise _Disconnect
                # flow control: block while the socket send buffer is
                # over the high-water mark (pause_writing fired)
                if not self._can_write.is_set():
                    await self._can_write.wait()

        try:
            await self.app(scope, receive, send)
        except _Disconnect:
            return False
        except Exception:
            if not state["started"]:
                w(b"HTTP/1.1 500 Internal Server Error\r\n"
                  b"content-length: 0\r\nconnection: close\r\n\r\n")
            return False
        return keep and not self.transport.is_closing()

    async def _read_body(self, clen: int) -> bytes:
        while len(self.buf) < clen:
            await self._read_more()
        body = bytes(self.buf[:clen])
        del self.buf[:clen]
        return body

    async def _read_chunked(self) -> bytes:
        out = bytearray()
        while True:
            while True:
                i = self.buf.find(b"\r\n")
                if i >= 0:
                    break
                await self._read_more()
            size_line = bytes(self.buf[:i]).split(b";")[0]
            del self.buf[:i + 2]
            n = int(size_line, 16)
            while len(self.buf) < n + 2:
                await self._read_more()
            out += self.buf[:n]
            del self.buf[:n + 2]
            if n == 0:
                break
        return bytes(out)


_REASON = {
    200: b"OK", 201: b"Created", 204: b"No Content", 301: b"Moved Permanently",
    302: b"Found", 304: b"Not Modified", 307: b"Temporary Redirect",
    400: b"Bad Request", 401: b"Unauthorized", 403: b"Forbidden",
    404: b"Not Found", 405: b"Method Not Allowed", 409: b"Conflict",
    422: b"Unprocessable Entity", 429: b"Too Many Requests",
    500: b"Internal Server Error", 501: b"Not Implemented",
    503: b"Service Unavailable",
}


async def start_http_server(app, host: str, port: int,
                            ssl_context=None) -> asyncio.AbstractServer:
    """Bind and serve the ASGI app; returns the asyncio server (use
    server.sockets[0].getsockname()[1] for the bound port)."""
    loop = asyncio.get_event_loop()
    return await loop.create_server(
        lambda: AsgiHttpProtocol(app, is_tls=ssl_context is not None),
        host, port, ssl=ssl_context, backlog=512, reuse_address=True)
