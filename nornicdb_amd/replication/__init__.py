"""Replication & clustering: Raft consensus, HA standby streaming, chaos
transports. Modes (reference docs/architecture/replication.md):
standalone / HA-standby / Raft / multi-region."""

from .transport import ChaosConfig, ChaosTransport, InProcTransport, Transport
from .raft import CANDIDATE, FOLLOWER, LEADER, LogEntry, RaftNode
from .ha import HAPrimary, HAStandby
from .adapter import StorageAdapter, command_for
from .multi_region import Region, RegionReceiver
from .tcp import MultiTcpTransport, TcpTransport
from .cluster import ClusterNode, NotLeader, ReplicatedEngine

__all__ = ["Transport", "InProcTransport", "ChaosTransport", "ChaosConfig",
           "RaftNode", "LogEntry", "LEADER", "FOLLOWER", "CANDIDATE",
           "HAPrimary", "HAStandby", "StorageAdapter", "command_for",
           "Region", "RegionReceiver", "TcpTransport", "MultiTcpTransport", "ClusterNode", "ReplicatedEngine", "NotLeader"]
