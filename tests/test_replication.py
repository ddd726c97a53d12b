"""Replication tests: Raft election/replication/failover under chaos,
HA standby streaming + promotion — all in-process with simulated
transports (reference pkg/replication/chaos_test.go + scenario_test.go).
"""

import time

import pytest

from nornicdb_amd.replication import (ChaosConfig, ChaosTransport, HAPrimary,
                                      HAStandby, InProcTransport, LEADER,
                                      RaftNode, StorageAdapter, command_for)
from nornicdb_amd.storage import MemoryEngine, Node
from nornicdb_amd.storage import wal as W


def node_cmd(i):
    return command_for(W.OP_CREATE_NODE, {
        "id": f"n{i}", "labels": ["R"], "props": {"i": i},
        "emb": None, "ca": 0, "ua": 0})


def make_cluster(n, transport=None, applies=None):
    bus = transport or InProcTransport()
    ids = [f"node{i}" for i in range(n)]
    nodes = []
    for i, nid in enumerate(ids):
        apply_fn = applies[i] if applies else None
        nodes.append(RaftNode(nid, ids, bus, apply_fn=apply_fn, seed=i))
    return bus, nodes


def pump(nodes, seconds, step=0.01):
    t0 = time.monotonic()
    while time.monotonic() - t0 < seconds:
        for nd in nodes:
            nd.tick()
        time.sleep(step)


def wait_leader(nodes, timeout=5.0):
    t0 = time.monotonic()
    while time.monotonic() - t0 < timeout:
        for nd in nodes:
            nd.tick()
        leaders = [n for n in nodes if n.is_leader]
        if len(leaders) == 1:
            # all other nodes acknowledge the same leader
            return leaders[0]
        time.sleep(0.01)
    raise AssertionError("no single leader elected")


class TestRaft:
    def test_election_single_leader(self):
        bus, nodes = make_cluster(3)
        leader = wait_leader(nodes)
        pump(nodes, 0.2)
        assert sum(1 for n in nodes if n.is_leader) == 1
        for n in nodes:
            if n is not leader:
                assert n.leader_id == leader.id
        bus.close()

    def test_log_replication_and_apply(self):
        engines = [MemoryEngine() for _ in range(3)]
        adapters = [StorageAdapter(e).apply for e in engines]
        bus, nodes = make_cluster(3, applies=adapters)
        leader = wait_leader(nodes)
        for i in range(5):
            assert leader.propose(node_cmd(i))
        pump(nodes, 0.5)
        for e in engines:
            assert e.node_count() == 5, e.node_count()

    def test_follower_forwarding(self):
        engines = [MemoryEngine() for _ in range(3)]
        adapters = [StorageAdapter(e).apply for e in engines]
        bus, nodes = make_cluster(3, applies=adapters)
        leader = wait_leader(nodes)
        follower = next(n for n in nodes if not n.is_leader)
        follower.propose(node_cmd(0))
        pump(nodes, 0.5)
        assert all(e.node_count() == 1 for e in engines)

    def test_leader_failover(self):
        bus, nodes = make_cluster(3)
        leader = wait_leader(nodes)
        survivors = [n for n in nodes if n is not leader]
        bus.unregister(leader.id)  # "kill" the leader
        new_leader = wait_leader(survivors, timeout=5)
        assert new_leader.id != leader.id
        assert new_leader.term > leader.term

    def test_replication_under_chaos(self):
        engines = [MemoryEngine() for _ in range(3)]
        adapters = [StorageAdapter(e).apply for e in engines]
        bus = InProcTransport()
        chaos = ChaosTransport(bus, ChaosConfig(drop_rate=0.10,
                                                duplicate_rate=0.10, seed=7))
        ids = [f"node{i}" for i in range(3)]
        nodes = [RaftNode(nid, ids, chaos, apply_fn=adapters[i], seed=i)
                 for i, nid in enumerate(ids)]
        leader = wait_leader(nodes, timeout=10)
        for i in range(10):
            leader.tick()
            leader.propose(node_cmd(i))
            pump(nodes, 0.05)
        pump(nodes, 1.5)
        counts = [e.node_count() for e in engines]
        # quorum must have everything; retries cover the dropped 10%
        assert max(counts) == 10
        assert sorted(counts)[1] == 10  # at least 2 of 3 fully caught up
        bus.close()

    def test_partition_heals(self):
        bus = InProcTransport()
        chaos = ChaosTransport(bus)
        ids = [f"node{i}" for i in range(3)]
        nodes = [RaftNode(nid, ids, chaos, seed=i) for i, nid in enumerate(ids)]
        leader = wait_leader(nodes)
        # partition the leader away
        chaos.set_partition({leader.id})
        others = [n for n in nodes if n is not leader]
        new_leader = wait_leader(others, timeout=5)
        # heal: old leader steps down on higher term
        chaos.set_partition(set())
        pump(nodes, 0.5)
        assert sum(1 for n in nodes if n.is_leader) == 1
        assert leader.state != LEADER or leader.term >= new_leader.term
        bus.close()


class TestHA:
    def test_stream_and_ack(self):
        bus = InProcTransport()
        standby_engine = MemoryEngine()
        primary = HAPrimary("p", ["s"], bus)
        standby = HAStandby("s", "p", bus, StorageAdapter(standby_engine).apply)
        for i in range(5):
            primary.replicate(node_cmd(i))
        for _ in range(100):          # poll: async apply on slow machines
            if standby_engine.node_count() == 5:
                break
            time.sleep(0.05)
        assert standby_engine.node_count() == 5
        assert primary.lag("s") == 0
        assert not standby.check_failover.__self__.promoted
        bus.close()

    def test_out_of_order_applies_in_order(self):
        bus = InProcTransport()
        eng = MemoryEngine()
        standby = HAStandby("s", "p", bus, StorageAdapter(eng).apply)
        # deliver 1 then 0
        standby._on_message({"type": "wal_entry", "seq": 1, "command": node_cmd(1)})
        assert eng.node_count() == 0  # buffered
        standby._on_message({"type": "wal_entry", "seq": 0, "command": node_cmd(0)})
        assert eng.node_count() == 2
        bus.close()

    def test_failover_promotion(self):
        now = [0.0]
        bus = InProcTransport()
        standby = HAStandby("s", "p", bus, lambda c: None, now_fn=lambda: now[0])
        assert not standby.check_failover()
        now[0] += 1.0  # primary silent past PROMOTE_AFTER
        assert standby.check_failover()
        assert standby.health()["role"] == "primary"
        bus.close()


class TestMultiRegion:
    def test_cross_region_streaming(self):
        from nornicdb_amd.replication import (InProcTransport, Region,
                                              RegionReceiver, StorageAdapter)
        from nornicdb_amd.storage import MemoryEngine
        import time as _time

        bus = InProcTransport()
        eng_us = MemoryEngine()
        eng_eu = MemoryEngine()
        recv_eu = RegionReceiver("eu", bus, StorageAdapter(eng_eu).apply)
        region_us = Region("us", ["us0", "us1", "us2"], bus,
                           StorageAdapter(eng_us).apply,
                           remote_regions=["eu"])
        # elect + replicate
        t0 = _time.monotonic()
        while region_us.leader() is None and _time.monotonic() - t0 < 5:
            region_us.tick_all()
            _time.sleep(0.01)
        assert region_us.leader() is not None
        for i in range(3):
            assert region_us.propose(node_cmd(i))
        t0 = _time.monotonic()
        while eng_eu.node_count() < 3 and _time.monotonic() - t0 < 5:
            region_us.tick_all()
            _time.sleep(0.01)
        assert eng_us.node_count() == 3
        assert eng_eu.node_count() == 3
        bus.close()

    def test_out_of_order_xregion(self):
        from nornicdb_amd.replication import InProcTransport, RegionReceiver, StorageAdapter
        from nornicdb_amd.storage import MemoryEngine
        bus = InProcTransport()
        eng = MemoryEngine()
        r = RegionReceiver("x", bus, StorageAdapter(eng).apply)
        r._on_message({"type": "xregion_entry", "region": "y", "seq": 1,
                       "command": node_cmd(1)})
        assert eng.node_count() == 0
        r._on_message({"type": "xregion_entry", "region": "y", "seq": 0,
                       "command": node_cmd(0)})
        assert eng.node_count() == 2
        bus.close()
