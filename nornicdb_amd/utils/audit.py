"""Append-only JSON audit trail + retention/GDPR policies.

Parity: reference pkg/audit/audit.go (append-only JSON log) and
pkg/retention/retention.go (retention policies, legal hold, GDPR Art.17
erasure).
"""

from __future__ import annotations

import json
import os
import threading
import time
from dataclasses import dataclass
from typing import Dict, Iterator, List, Optional

from ..storage.types import Engine


class AuditLog:
    def __init__(self, path: Optional[str] = None, now_fn=time.time):
        self.path = path
        self.now = now_fn
        self._lock = threading.Lock()
        self._mem: List[dict] = []
        self._f = open(path, "a") if path else None

    def record(self, action: str, actor: str = "", target: str = "",
               detail: Dict = None):
        entry = {"ts": self.now(), "action": action, "actor": actor,
                 "target": target, "detail": detail or {}}
        with self._lock:
            if self._f:
                self._f.write(json.dumps(entry) + "\n")
                self._f.flush()
            else:
                self._mem.append(entry)

    def entries(self) -> Iterator[dict]:
        if self.path and os.path.exists(self.path):
            with open(self.path) as f:
                for line in f:
                    if line.strip():
                        yield json.loads(line)
        else:
            with self._lock:
                yield from list(self._mem)

    def close(self):
        if self._f:
            self._f.close()


@dataclass
class RetentionPolicy:
    label: str
    max_age_days: float
    action: str = "delete"   # "delete" | "archive"


class RetentionManager:
    """Retention + legal hold + right-to-erasure (GDPR Art. 17)."""

    def __init__(self, engine: Engine, audit: AuditLog = None, now_fn=time.time):
        self.engine = engine
        self.audit = audit
        self.now = now_fn
        self.policies: List[RetentionPolicy] = []
        self._holds: set = set()

    def add_policy(self, policy: RetentionPolicy):
        self.policies.append(policy)

    def legal_hold(self, node_id: str, hold: bool = True):
        if hold:
            self._holds.add(node_id)
        else:
            self._holds.discard(node_id)

    def enforce(self) -> Dict[str, int]:
        stats = {"deleted": 0, "archived": 0, "held": 0}
        now = self.now()
        for pol in self.policies:
            cutoff = now - pol.max_age_days * 86400
            for node in self.engine.get_nodes_by_label(pol.label):
                created = node.properties.get("created_at", now)
                if created >= cutoff:
                    continue
                if node.id in self._holds:
                    stats["held"] += 1
                    continue
                if pol.action == "archive":
                    if "Archived" not in node.labels:
                        node.labels.append("Archived")
                        self.engine.update_node(node)
                        stats["archived"] += 1
                else:
                    self.engine.detach_delete_node(node.id)
                    stats["deleted"] += 1
                if self.audit:
                    self.audit.record(f"retention_{pol.action}", "system", node.id)
        return stats

    def erase_subject(self, subject: str) -> int:
        """GDPR Art.17: erase all nodes tagged with the data subject."""
        deleted = 0
        for node in list(self.engine.all_nodes()):
            props = node.properties
            if props.get("subject") == subject or props.get("user") == subject:
                if node.id in self._holds:
                    continue
                self.engine.detach_delete_node(node.id)
                deleted += 1
        if self.audit:
            self.audit.record("gdpr_erasure", "system", subject,
                              {"deleted": deleted})
        return deleted
