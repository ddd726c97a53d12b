"""Infra utilities: config, encryption, audit/retention, caching."""

from .config import Config, find_config_file, load_config
from .encryption import EncryptionManager, derive_key
from .audit import AuditLog, RetentionManager, RetentionPolicy
from .cache import LRUCache, QueryCache

__all__ = ["Config", "load_config", "find_config_file", "EncryptionManager",
           "derive_key", "AuditLog", "RetentionManager", "RetentionPolicy",
           "LRUCache", "QueryCache"]
