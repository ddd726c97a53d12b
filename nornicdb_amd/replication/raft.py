"""Raft consensus for replicated writes.

Parity: reference pkg/replication/raft.go (election/heartbeat/log
replication, 3-5 nodes, leader writes + follower reads). Log entries are
storage commands (the same wire ops as the WAL); commit applies them to
the local engine through the storage adapter.
"""

from __future__ import annotations

import random
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

FOLLOWER = "follower"
CANDIDATE = "candidate"
LEADER = "leader"


@dataclass
class LogEntry:
    term: int
    command: Dict[str, Any]


class RaftNode:
    """One Raft participant. Drive time with tick() (deterministic tests)
    or start()'s background ticker."""

    HEARTBEAT = 0.05
    ELECTION_MIN = 0.15
    ELECTION_MAX = 0.30

    def __init__(self, node_id: str, peers: List[str], transport,
                 apply_fn: Callable[[Dict[str, Any]], None] = None,
                 now_fn=time.monotonic, seed: int = None):
        self.id = node_id
        self.peers = [p for p in peers if p != node_id]
        self.transport = transport
        self.apply_fn = apply_fn or (lambda cmd: None)
        self.now = now_fn
        self._rng = random.Random(seed if seed is not None else hash(node_id) & 0xFFFF)

        self.state = FOLLOWER
        self.term = 0
        self.voted_for: Optional[str] = None
        self.log: List[LogEntry] = []
        self.commit_index = -1
        self.last_applied = -1
        self.leader_id: Optional[str] = None

        # leader state
        self.next_index: Dict[str, int] = {}
        self.match_index: Dict[str, int] = {}
        self._votes: set = set()

        self._lock = threading.RLock()
        self._last_heard = self.now()
        self._last_heartbeat = 0.0
        self._election_timeout = self._rand_timeout()
        self._stop = threading.Event()
        self._ticker: Optional[threading.Thread] = None

        transport.register(node_id, self.on_message)

    # ---------------------------------------------------------------- time
    def _rand_timeout(self):
        return self._rng.uniform(self.ELECTION_MIN, self.ELECTION_MAX)

    def start(self):
        self._ticker = threading.Thread(target=self._tick_loop, daemon=True)
        self._ticker.start()
        return self

    def stop(self):
        self._stop.set()
        if self._ticker:
            self._ticker.join(timeout=1)

    def _tick_loop(self):
        while not self._stop.wait(0.01):
            self.tick()

    def tick(self):
        with self._lock:
            now = self.now()
            if self.state == LEADER:
                if now - self._last_heartbeat >= self.HEARTBEAT:
                    self._broadcast_append()
            elif now - self._last_heard >= self._election_timeout:
                self._start_election()
            self._apply_committed()

    # ------------------------------------------------------------ election
    def _start_election(self):
        self.state = CANDIDATE
        self.term += 1
        self.voted_for = self.id
        self._votes = {self.id}
        self._last_heard = self.now()
        self._election_timeout = self._rand_timeout()
        last_term = self.log[-1].term if self.log else 0
        for p in self.peers:
            self.transport.send(p, {
                "type": "request_vote", "from": self.id, "term": self.term,
                "last_log_index": len(self.log) - 1, "last_log_term": last_term})
        if not self.peers:
            self._become_leader()

    def _become_leader(self):
        self.state = LEADER
        self.leader_id = self.id
        for p in self.peers:
            self.next_index[p] = len(self.log)
            self.match_index[p] = -1
        self._broadcast_append()

    # ------------------------------------------------------------ messages
    def on_message(self, msg: Dict[str, Any]):
        with self._lock:
            t = msg.get("type")
            if msg.get("term", 0) > self.term:
                self.term = msg["term"]
                self.state = FOLLOWER
                self.voted_for = None
            if t == "request_vote":
                self._on_request_vote(msg)
            elif t == "vote":
                self._on_vote(msg)
            elif t == "append_entries":
                self._on_append_entries(msg)
            elif t == "append_reply":
                self._on_append_reply(msg)
            elif t == "client_command":
                self.propose(msg["command"])
            self._apply_committed()

    def _on_request_vote(self, msg):
        grant = False
        if msg["term"] >= self.term and self.voted_for in (None, msg["from"]):
            my_last_term = self.log[-1].term if self.log else 0
            up_to_date = (msg["last_log_term"], msg["last_log_index"]) >= \
                         (my_last_term, len(self.log) - 1)
            if up_to_date:
                grant = True
                self.voted_for = msg["from"]
                self._last_heard = self.now()
        self.transport.send(msg["from"], {
            "type": "vote", "from": self.id, "term": self.term, "granted": grant})

    def _on_vote(self, msg):
        if self.state != CANDIDATE or msg["term"] != self.term:
            return
        if msg.get("granted"):
            self._votes.add(msg["from"])
            if len(self._votes) > (len(self.peers) + 1) // 2:
                self._become_leader()

    def _broadcast_append(self):
        self._last_heartbeat = self.now()
        for p in self.peers:
            ni = self.next_index.get(p, len(self.log))
            prev_idx = ni - 1
            prev_term = self.log[prev_idx].term if 0 <= prev_idx < len(self.log) else 0
            entries = [(e.term, e.command) for e in self.log[ni:ni + 64]]
            self.transport.send(p, {
                "type": "append_entries", "from": self.id, "term": self.term,
                "prev_index": prev_idx, "prev_term": prev_term,
                "entries": entries, "leader_commit": self.commit_index})

    def _on_append_entries(self, msg):
        if msg["term"] < self.term:
            self.transport.send(msg["from"], {
                "type": "append_reply", "from": self.id, "term": self.term,
                "success": False, "match_index": -1})
            return
        self.state = FOLLOWER
        self.leader_id = msg["from"]
        self._last_heard = self.now()
        self._election_timeout = self._rand_timeout()
        prev_idx = msg["prev_index"]
        if prev_idx >= 0 and (prev_idx >= len(self.log)
                              or self.log[prev_idx].term != msg["prev_term"]):
            self.transport.send(msg["from"], {
                "type": "append_reply", "from": self.id, "term": self.term,
                "success": False, "match_index": -1})
            return
        idx = prev_idx + 1
        for term, cmd in msg["entries"]:
            if idx < len(self.log):
                if self.log[idx].term != term:
                    del self.log[idx:]
                    self.log.append(LogEntry(term, cmd))
            else:
                self.log.append(LogEntry(term, cmd))
            idx += 1
        if msg["leader_commit"] > self.commit_index:
            self.commit_index = min(msg["leader_commit"], len(self.log) - 1)
        self.transport.send(msg["from"], {
            "type": "append_reply", "from": self.id, "term": self.term,
            "success": True, "match_index": idx - 1})

    def _on_append_reply(self, msg):
        if self.state != LEADER or msg["term"] != self.term:
            return
        p = msg["from"]
        if msg["success"]:
            self.match_index[p] = max(self.match_index.get(p, -1),
                                      msg["match_index"])
            self.next_index[p] = self.match_index[p] + 1
            self._advance_commit()
        else:
            self.next_index[p] = max(0, self.next_index.get(p, 1) - 1)

    def _advance_commit(self):
        for n in range(len(self.log) - 1, self.commit_index, -1):
            if self.log[n].term != self.term:
                continue
            count = 1 + sum(1 for p in self.peers
                            if self.match_index.get(p, -1) >= n)
            if count > (len(self.peers) + 1) // 2:
                self.commit_index = n
                break

    def _apply_committed(self):
        while self.last_applied < self.commit_index:
            self.last_applied += 1
            try:
                self.apply_fn(self.log[self.last_applied].command)
            except Exception:
                pass

    # ------------------------------------------------------------- client
    def propose(self, command: Dict[str, Any]) -> bool:
        """Leader: append + replicate. Follower: forward to leader
        (reference: write forwarding via Bolt, transport.go:1-13)."""
        with self._lock:
            if self.state == LEADER:
                self.log.append(LogEntry(self.term, command))
                self._broadcast_append()
                return True
            if self.leader_id:
                self.transport.send(self.leader_id, {
                    "type": "client_command", "from": self.id,
                    "command": command})
                return True
            return False

    @property
    def is_leader(self):
        return self.state == LEADER

    def health(self) -> Dict[str, Any]:
        with self._lock:
            return {"id": self.id, "state": self.state, "term": self.term,
                    "leader": self.leader_id, "log_len": len(self.log),
                    "commit_index": self.commit_index}
