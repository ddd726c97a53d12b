"""TCP cluster transport: real network replication between processes.

Parity: reference pkg/replication/transport.go:1-28 — the cluster runs
its own TCP protocol on a dedicated port (:7688, separate from Bolt
:7687). This transport plugs into the same Transport interface as
InProcTransport/ChaosTransport, so RaftNode / HAPrimary / HAStandby /
MultiRegion work unchanged across real processes and machines.

Wire format per message: magic u16 "NT" | len u32 | msgpack payload.
Connections are persistent and lazily (re-)established per destination;
a failed send drops the cached connection and retries once, then gives
up silently (Raft/HA tolerate lost messages by design — retry is the
consensus layer's job, exactly as in the reference).
"""

from __future__ import annotations

import socket
import struct
import threading
from typing import Any, Callable, Dict, Optional, Tuple

import msgpack

from .transport import Transport

_HDR = struct.Struct("<HI")
_MAGIC = 0x544E  # "NT"
MAX_MSG = 64 << 20


class TcpTransport(Transport):
    """One instance per process/node. `peers` maps node id -> (host, port);
    the local node's entry defines the bind address."""

    def __init__(self, node_id: str, peers: Dict[str, Tuple[str, int]],
                 bind: Optional[Tuple[str, int]] = None):
        self.node_id = node_id
        self.peers = dict(peers)
        self._handler: Optional[Callable[[Dict[str, Any]], None]] = None
        self._out: Dict[str, socket.socket] = {}
        self._out_lock = threading.Lock()
        self._accepted: list = []
        self._stop = threading.Event()

        host, port = bind if bind is not None else peers[node_id]
        self._srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._srv.bind((host, port))
        self._srv.listen(32)
        self.bound_port = self._srv.getsockname()[1]
        self._acceptor = threading.Thread(target=self._accept_loop, daemon=True)
        self._acceptor.start()

    # ---- Transport interface ----
    def register(self, node_id, handler):
        # single-handler per process (this node); id kept for parity
        self._handler = handler

    def send(self, dst: str, msg: Dict[str, Any]) -> None:
        if self._stop.is_set():
            return
        if dst == self.node_id:
            h = self._handler
            if h:
                try:
                    h(msg)
                except Exception:
                    pass
            return
        payload = msgpack.packb(msg, use_bin_type=True)
        frame = _HDR.pack(_MAGIC, len(payload)) + payload
        for attempt in (0, 1):
            s = self._conn(dst, fresh=attempt > 0)
            if s is None:
                return
            try:
                s.sendall(frame)
                return
            except OSError:
                with self._out_lock:
                    if self._out.get(dst) is s:
                        self._out.pop(dst, None)
                try:
                    s.close()
                except OSError:
                    pass

    # ---- connections ----
    def _conn(self, dst: str, fresh: bool = False) -> Optional[socket.socket]:
        with self._out_lock:
            s = self._out.get(dst)
            if s is not None and not fresh:
                # liveness peek: a peer that closed/reset shows up as a
                # readable-EOF; plain sendall would "succeed" into the
                # kernel buffer and silently lose the frame.
                try:
                    if s.recv(1, socket.MSG_DONTWAIT | socket.MSG_PEEK) == b"":
                        s.close()
                        s = None
                        self._out.pop(dst, None)
                except BlockingIOError:
                    pass  # alive, nothing to read
                except OSError:
                    s = None
                    self._out.pop(dst, None)
                if s is not None:
                    return s
            addr = self.peers.get(dst)
            if addr is None:
                return None
            try:
                s = socket.create_connection(tuple(addr), timeout=2.0)
                s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
                s.settimeout(None)
            except OSError:
                return None
            self._out[dst] = s
            return s

    def _accept_loop(self):
        while not self._stop.is_set():
            try:
                conn, _ = self._srv.accept()
            except OSError:
                return
            conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            # REUSEADDR on accepted conns too: Linux refuses a new listener
            # bind while an old accepted socket lingers in FIN_WAIT unless
            # BOTH sockets carry the flag (restart-after-crash path).
            conn.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            self._accepted.append(conn)
            t = threading.Thread(target=self._read_loop, args=(conn,),
                                 daemon=True)
            t.start()

    def _read_loop(self, conn: socket.socket):
        buf = b""
        try:
            while not self._stop.is_set():
                while len(buf) < _HDR.size:
                    chunk = conn.recv(65536)
                    if not chunk:
                        return
                    buf += chunk
                magic, ln = _HDR.unpack_from(buf)
                if magic != _MAGIC or ln > MAX_MSG:
                    return  # protocol error: drop connection
                while len(buf) < _HDR.size + ln:
                    chunk = conn.recv(65536)
                    if not chunk:
                        return
                    buf += chunk
                payload = buf[_HDR.size:_HDR.size + ln]
                buf = buf[_HDR.size + ln:]
                msg = msgpack.unpackb(payload, raw=False)
                h = self._handler
                if h is not None:
                    try:
                        h(msg)
                    except Exception:
                        pass
        except OSError:
            pass
        finally:
            try:
                conn.close()
            except OSError:
                pass

    def close(self):
        self._stop.set()
        try:
            # shutdown() wakes the acceptor thread blocked in accept();
            # close() alone leaves the listener alive inside the syscall
            # (the port stays in LISTEN until accept returns).
            self._srv.shutdown(socket.SHUT_RDWR)
        except OSError:
            pass
        try:
            self._srv.close()
        except OSError:
            pass
        self._acceptor.join(timeout=1)
        for c in self._accepted:
            try:
                # linger=0 -> RST + immediate kernel teardown: no
                # FIN_WAIT/TIME_WAIT orphan holding the port against a
                # restarted listener (crash-restart parity).
                c.setsockopt(socket.SOL_SOCKET, socket.SO_LINGER,
                             struct.pack("ii", 1, 0))
                c.close()
            except OSError:
                pass
        with self._out_lock:
            for s in self._out.values():
                try:
                    s.close()
                except OSError:
                    pass
            self._out.clear()


class MultiTcpTransport(Transport):
    """Adapter for hosting SEVERAL logical nodes in one process over real
    TCP (used by tests and by single-process multi-node dev clusters):
    each registered node gets its own TcpTransport; send() routes from
    the caller's implicit source via any member (messages carry 'from').
    """

    def __init__(self, peers: Dict[str, Tuple[str, int]]):
        self.peers = dict(peers)
        self._members: Dict[str, TcpTransport] = {}

    def register(self, node_id, handler):
        t = TcpTransport(node_id, self.peers)
        t.register(node_id, handler)
        self._members[node_id] = t

    def send(self, dst, msg):
        # any member can originate; prefer the sender's own transport
        src = msg.get("from", "")
        t = self._members.get(src) or next(iter(self._members.values()), None)
        if t is not None:
            t.send(dst, msg)

    def close(self):
        for t in self._members.values():
            t.close()
