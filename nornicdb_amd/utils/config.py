"""Configuration: YAML file + environment + explicit overrides, with
feature flags.

Parity: reference pkg/config/config.go (precedence CLI flags > YAML > env
> defaults, :82,:766,:1941,:2420) and feature_flags.go (:25-116, ~20
NORNICDB_* flags).
"""

from __future__ import annotations

import os
from dataclasses import asdict, dataclass, field
from typing import Any, Dict, Optional

import yaml

ENV_PREFIX = "NORNICDB_"


@dataclass
class Config:
    # paths / server
    data_dir: str = ""                    # "" = in-memory
    bolt_host: str = "0.0.0.0"
    bolt_port: int = 7687
    http_host: str = "0.0.0.0"
    http_port: int = 7474
    # embedding
    embedder: str = "local"               # local | mock | ollama | openai
    embedding_dims: int = 1024
    embed_workers: int = 2
    chunk_tokens: int = 512
    chunk_overlap: int = 50
    # storage
    wal_sync_on_write: bool = False
    snapshot_interval_s: float = 300.0
    async_writes: bool = True
    encryption_passphrase: str = ""
    # "disk" (LSM on-disk engine, default) or "wal" (RAM + WAL snapshot)
    storage_engine: str = "disk"
    # search
    hnsw_m: int = 16
    hnsw_ef_construction: int = 200
    hnsw_ef_search: int = 100
    brute_force_max: int = 5000
    kmeans_min: int = 100000
    # "" | "int8" | "fp8": 1 B/element GPU corpus (2x capacity; int8
    # keeps recall@10 >= 0.95 worst-case). Also NORNICDB_SEARCH_QUANT.
    search_quant: str = ""
    # observability
    log_queries: bool = False   # reference --log-queries / bolt LogQueries
    # auth
    auth_enabled: bool = False
    initial_admin_password: str = ""
    # feature flags (reference feature_flags.go)
    flags: Dict[str, bool] = field(default_factory=lambda: {
        "auto_embed": True,
        "auto_tlp": False,          # automatic link prediction on store
        "kalman_decay": False,
        "kmeans_clustering": True,
        "edge_decay": False,
        "wal": True,
        "gpu": True,
        "query_cache": True,
        "parallel_match": True,
        "inference": False,
        "temporal_tracking": True,
        "heimdall": False,
        # reference feature_flags.go parity set
        "evidence_buffering": True,   # NORNICDB_EVIDENCE_BUFFERING_ENABLED
        "cooldown": True,             # NORNICDB_COOLDOWN_ENABLED
        "edge_provenance": False,
        "per_node_config": False,
        "auto_tlp_llm_qc": False,     # HeimdallQC gate for auto links
        "gpu_clustering_auto": True,
    })

    def flag(self, name: str, default: bool = False) -> bool:
        return self.flags.get(name, default)


def _coerce(value: str, target_type) -> Any:
    if target_type is bool:
        return value.lower() in ("1", "true", "yes", "on")
    if target_type is int:
        return int(value)
    if target_type is float:
        return float(value)
    return value


def load_config(path: Optional[str] = None, env: Dict[str, str] = None,
                overrides: Dict[str, Any] = None) -> Config:
    """Precedence: overrides > env (NORNICDB_*) > YAML > defaults."""
    cfg = Config()
    # YAML
    if path and os.path.exists(path):
        with open(path) as f:
            data = yaml.safe_load(f) or {}
        for k, v in data.items():
            k = k.replace("-", "_")
            if k == "flags" and isinstance(v, dict):
                cfg.flags.update({fk.replace("-", "_"): bool(fv)
                                  for fk, fv in v.items()})
            elif hasattr(cfg, k):
                setattr(cfg, k, v)
    # env
    env = env if env is not None else os.environ
    for key, value in env.items():
        if not key.startswith(ENV_PREFIX):
            continue
        name = key[len(ENV_PREFIX):].lower()
        if name.startswith("flag_"):
            cfg.flags[name[5:]] = _coerce(value, bool)
        elif name.endswith("_enabled"):
            # reference-style flags: NORNICDB_EVIDENCE_BUFFERING_ENABLED=1
            cfg.flags[name[:-8]] = _coerce(value, bool)
        elif hasattr(cfg, name):
            cur = getattr(cfg, name)
            setattr(cfg, name, _coerce(value, type(cur)))
    # explicit overrides
    for k, v in (overrides or {}).items():
        if k == "flags":
            cfg.flags.update(v)
        elif hasattr(cfg, k):
            setattr(cfg, k, v)
    return cfg


def find_config_file(start_dir: str = ".") -> Optional[str]:
    """Search for nornicdb.yaml upward (reference FindConfigFile)."""
    cur = os.path.abspath(start_dir)
    for _ in range(10):
        for name in ("nornicdb.yaml", "nornicdb.yml", ".nornicdb.yaml"):
            p = os.path.join(cur, name)
            if os.path.exists(p):
                return p
        parent = os.path.dirname(cur)
        if parent == cur:
            break
        cur = parent
    return None
