"""Bolt protocol server + PackStream serialization."""

from .packstream import IdMap, Structure, pack, unpack
from .server import BoltServer, BoltSession

__all__ = ["BoltServer", "BoltSession", "Structure", "pack", "unpack", "IdMap"]
