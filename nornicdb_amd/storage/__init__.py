"""Storage engines.

Stack (mirrors reference assembly pkg/nornicdb/db.go:762-915):
    DiskEngine (LSM on-disk store, replaces BadgerEngine — the default
                persistent engine; datasets may exceed RAM)
    PersistentEngine (RAM state + WAL + snapshots — the fast in-RAM
                durable engine, kept for workloads that fit memory)
      -> AsyncEngine (write-behind, optional)
        -> NamespacedEngine (multi-tenant prefixing)
"""

from .types import (ConstraintViolation, Edge, Engine, EventType, Node,
                    NotFoundError, StorageError, new_id)
from .memory import MemoryEngine
from .persistent import PersistentEngine, Transaction
from .async_engine import AsyncEngine
from .namespaced import NamespacedEngine
from .wal import WAL, WALCorruption, WALDegraded
from .schema import Constraint, SchemaManager, VectorIndexMeta
from .composite import CompositeEngine
from .disk import DiskEngine, DiskTransaction
from .lsm import LSMStore
from .node_config import (LabelConfig, NodeConfig, NodeConfigStore,
                          TRUST_DEFAULT, TRUST_HIGH, TRUST_LOW,
                          TRUST_VERIFIED)

__all__ = [
    "Node", "Edge", "Engine", "EventType", "StorageError", "NotFoundError",
    "ConstraintViolation", "new_id", "MemoryEngine", "PersistentEngine",
    "Transaction", "AsyncEngine", "NamespacedEngine", "WAL", "WALCorruption",
    "SchemaManager", "Constraint", "VectorIndexMeta", "CompositeEngine",
    "WALDegraded", "DiskEngine", "DiskTransaction", "LSMStore", "NodeConfig", "NodeConfigStore", "LabelConfig",
]
