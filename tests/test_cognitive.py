"""Cognitive layer tests: decay scoring/archival, Kalman, temporal
tracking, link prediction, inference auto-linking.

Models reference pkg/decay, pkg/filter, pkg/temporal, pkg/linkpredict,
pkg/inference tests."""

import numpy as np
import pytest

from nornicdb_amd.cognitive import (AccessTracker, DecayConfig, DecayManager,
                                    HALF_LIVES, InferenceConfig,
                                    InferenceEngine, Kalman1D,
                                    QueryLoadTracker, adamic_adar,
                                    common_neighbors, jaccard, predict_links)
from nornicdb_amd.search import SearchService
from nornicdb_amd.storage import Edge, MemoryEngine, Node

DAY = 86400.0


def mem(id, created, accessed=None, tier="episodic", importance=0.5, count=0):
    return Node(id, ["Memory"], {
        "memory_type": tier, "importance": importance,
        "created_at": created, "last_accessed": accessed or created,
        "access_count": count, "content": f"content {id}"})


class TestDecay:
    def test_halflife_recency(self):
        eng = MemoryEngine()
        now = [1000.0 * DAY]
        dm = DecayManager(eng, now_fn=lambda: now[0])
        fresh = mem("fresh", now[0])
        week_old = mem("old", now[0] - 7 * DAY)
        eng.create_node(fresh); eng.create_node(week_old)
        s_fresh = dm.score(eng.get_node("fresh"))
        s_old = dm.score(eng.get_node("old"))
        assert s_fresh > s_old
        # one episodic half-life halves the recency component
        assert abs((s_fresh - s_old) - 0.5 * 0.5) < 0.02

    def test_tier_halflives(self):
        eng = MemoryEngine()
        now = [1000.0 * DAY]
        dm = DecayManager(eng, now_fn=lambda: now[0])
        for tier in ("episodic", "semantic", "procedural"):
            eng.create_node(mem(tier, now[0] - 30 * DAY, tier=tier))
        se = dm.score(eng.get_node("episodic"))
        ss = dm.score(eng.get_node("semantic"))
        sp = dm.score(eng.get_node("procedural"))
        assert se < ss < sp

    def test_cycle_archives_and_deletes(self):
        eng = MemoryEngine()
        now = [1000.0 * DAY]
        dm = DecayManager(eng, DecayConfig(archive_threshold=0.3,
                                           delete_threshold=0.05),
                          now_fn=lambda: now[0])
        eng.create_node(mem("dead", now[0] - 400 * DAY, importance=0.0))
        eng.create_node(mem("fading", now[0] - 40 * DAY, importance=0.4))
        eng.create_node(mem("alive", now[0], importance=0.9, count=10))
        stats = dm.run_cycle()
        assert stats["deleted"] == 1
        assert not eng.has_node("dead")
        assert "Archived" in eng.get_node("fading").labels
        assert "Archived" not in eng.get_node("alive").labels

    def test_reinforce(self):
        eng = MemoryEngine()
        eng.create_node(mem("m", 0))
        dm = DecayManager(eng)
        imp = dm.reinforce("m")
        assert imp == 0.6
        assert eng.get_node("m").properties["access_count"] == 1


class TestKalman:
    def test_converges_to_constant(self):
        kf = Kalman1D(q=0.001, r=0.5, initial=0.0)
        for _ in range(100):
            kf.update(5.0)
        assert abs(kf.x - 5.0) < 0.05

    def test_smooths_noise(self):
        rng = np.random.default_rng(0)
        kf = Kalman1D(q=0.01, r=1.0)
        xs = [kf.update(3.0 + rng.normal(0, 0.5)) for _ in range(200)]
        assert abs(np.mean(xs[-50:]) - 3.0) < 0.2
        assert np.std(xs[-50:]) < 0.2


class TestTemporal:
    def test_next_access_prediction(self):
        now = [0.0]
        t = AccessTracker(now_fn=lambda: now[0])
        for i in range(10):
            now[0] = i * 100.0
            t.record("n")
        pred = t.predict_next_access("n")
        assert pred is not None
        assert abs(pred - 1000.0) < 30

    def test_sessions(self):
        now = [0.0]
        t = AccessTracker(now_fn=lambda: now[0])
        t.record("a"); now[0] = 60; t.record("b")
        now[0] = 10000; t.record("c")
        assert len(t.sessions()) == 2

    def test_co_access(self):
        now = [0.0]
        t = AccessTracker(now_fn=lambda: now[0])
        t.record("x"); now[0] = 5; t.record("y")
        now[0] = 5000; t.record("z")
        pairs = t.co_accessed(window=60)
        assert pairs[0][:2] == ("x", "y")

    def test_period_detection(self):
        now = [0.0]
        t = AccessTracker(now_fn=lambda: now[0])
        for i in range(8):
            now[0] = i * 3600.0
            t.record("daily")
        p = t.detect_period("daily")
        assert p is not None and abs(p - 3600) < 1

    def test_query_load(self):
        now = [0.0]
        q = QueryLoadTracker(window=10, now_fn=lambda: now[0])
        for _ in range(50):
            q.record_query()
        assert q.qps() == 5.0
        assert q.decay_interval(100) > 100


class TestLinkPredict:
    def _triangle_plus(self):
        eng = MemoryEngine()
        for n in "abcd":
            eng.create_node(Node(n, [], {}))
        eng.create_edge(Edge("e1", "R", "a", "b"))
        eng.create_edge(Edge("e2", "R", "b", "c"))
        eng.create_edge(Edge("e3", "R", "a", "d"))
        eng.create_edge(Edge("e4", "R", "d", "c"))
        return eng

    def test_scores(self):
        eng = self._triangle_plus()
        assert common_neighbors(eng, "a", "c") == 2.0
        assert 0 < jaccard(eng, "a", "c") <= 1
        assert adamic_adar(eng, "a", "c") > 0

    def test_predict_links_ranks_closure(self):
        eng = self._triangle_plus()
        preds = predict_links(eng, "a", method="common_neighbors")
        assert preds and preds[0][0] == "c"


class TestInference:
    def test_similarity_autolink(self):
        eng = MemoryEngine()
        svc = SearchService(eng, dims=8, device="cpu", use_hnsw=False)
        v = np.zeros(8); v[0] = 1.0
        a = Node("a", ["Memory"], {"created_at": 0.0}, embedding=list(v))
        v2 = v + 0.01
        b = Node("b", ["Memory"], {"created_at": 1e9}, embedding=list(v2 / np.linalg.norm(v2)))
        eng.create_node(a)
        eng.create_node(b)
        inf = InferenceEngine(eng, svc, config=InferenceConfig(
            similarity_threshold=0.9, temporal_window_s=1))
        created = inf.on_store(eng.get_node("b"))
        assert created and created[0].type == "RELATES_TO"
        assert created[0].properties["reason"] == "similarity"
        assert "a" in eng.neighbors("b")

    def test_cooldown_prevents_duplicates(self):
        eng = MemoryEngine()
        svc = SearchService(eng, dims=8, device="cpu", use_hnsw=False)
        v = [1.0] + [0.0] * 7
        eng.create_node(Node("a", ["Memory"], {"created_at": 0.0}, embedding=v))
        eng.create_node(Node("b", ["Memory"], {"created_at": 1e9}, embedding=v))
        inf = InferenceEngine(eng, svc, config=InferenceConfig(
            similarity_threshold=0.9, temporal_window_s=1))
        first = inf.on_store(eng.get_node("b"))
        second = inf.on_store(eng.get_node("b"))
        assert len(first) == 1 and len(second) == 0

    def test_edge_decay_prunes(self):
        eng = MemoryEngine()
        now = [0.0]
        inf = InferenceEngine(eng, None, now_fn=lambda: now[0],
                              config=InferenceConfig(edge_decay_per_day=0.1,
                                                     prune_below=0.2))
        eng.create_node(Node("a", [], {}))
        eng.create_node(Node("b", [], {}))
        eng.create_edge(Edge("e", "RELATES_TO", "a", "b",
                             {"inferred": True, "confidence": 0.5,
                              "created_at": 0.0}))
        now[0] = 10 * 86400.0  # conf -> 0.5 - 1.0 < prune
        stats = inf.decay_inferred_edges()
        assert stats["pruned"] == 1
        assert eng.edge_count() == 0


class TestKalmanDecayAB:
    def test_kalman_variant_smooths_decay_scores(self):
        """A/B comparison (reference kalman_adapter_ab_test.go): under a
        bursty access pattern the Kalman-filtered decay score must vary
        less step-to-step than the raw score, while tracking its level."""
        eng_raw, eng_kal = MemoryEngine(), MemoryEngine()
        now = [0.0]
        raw = DecayManager(eng_raw, DecayConfig(use_kalman=False),
                           now_fn=lambda: now[0])
        kal = DecayManager(eng_kal, DecayConfig(use_kalman=True),
                           now_fn=lambda: now[0])
        for e in (eng_raw, eng_kal):
            e.create_node(mem("m", 0.0))
        raw_scores, kal_scores = [], []
        import random
        rng = random.Random(0)
        for step in range(40):
            now[0] += 86400.0 * rng.choice([0.1, 3.0])  # bursty gaps
            if rng.random() < 0.4:  # sporadic reinforcement
                for e in (eng_raw, eng_kal):
                    n = e.get_node("m")
                    n.properties["last_accessed"] = now[0]
                    n.properties["access_count"] += 1
                    e.update_node(n)
            raw_scores.append(raw.score(eng_raw.get_node("m")))
            kal_scores.append(kal.score(eng_kal.get_node("m")))

        def roughness(xs):
            return sum(abs(b - a) for a, b in zip(xs, xs[1:])) / (len(xs) - 1)

        assert roughness(kal_scores) < roughness(raw_scores)
        # still tracks the same level
        assert abs(sum(kal_scores) / 40 - sum(raw_scores) / 40) < 0.15


class TestInferenceQCAndClusters:
    """HeimdallQC veto + ClusterIntegration (reference
    pkg/inference/heimdall_qc.go + cluster_integration.go)."""

    def test_qc_accepts_without_manager(self):
        from nornicdb_amd.cognitive.inference import HeimdallQC
        from nornicdb_amd.storage.types import Node
        qc = HeimdallQC(manager=None)
        a = Node(id="a", labels=[], properties={"name": "x"})
        b = Node(id="b", labels=[], properties={"name": "y"})
        assert qc.check(a, b, "similar embeddings") is True
        assert qc.stats["checked"] == 1

    def test_qc_veto_with_stub_manager(self):
        from nornicdb_amd.cognitive.inference import HeimdallQC
        from nornicdb_amd.storage.types import Node

        class StubMgr:
            def generate(self, prompt, max_tokens=4, temperature=0.0):
                return "no , unrelated"
        qc = HeimdallQC(manager=StubMgr())
        a = Node(id="a", labels=[], properties={})
        b = Node(id="b", labels=[], properties={})
        assert qc.check(a, b, "co-access") is False
        assert qc.stats["vetoed"] == 1

    def test_cluster_integration(self):
        import numpy as np

        from nornicdb_amd.cognitive.inference import ClusterIntegration
        from nornicdb_amd.search.embedding_index import EmbeddingIndex

        class S:
            pass
        s = S()
        s.emb = EmbeddingIndex(dims=4, device="cpu")
        rng = np.random.default_rng(0)
        for i in range(10):
            s.emb.add(f"a{i}", [1, 0, 0, 0] + rng.normal(0, 0.01, 4))
        for i in range(10):
            s.emb.add(f"b{i}", [0, 1, 0, 0] + rng.normal(0, 0.01, 4))
        ci = ClusterIntegration(s)
        n = ci.recluster(k=2)
        assert n == 2
        assert ci.same_cluster("a0", "a5")
        assert not ci.same_cluster("a0", "b0")
        assert ci.boost_for("a0", "a1") > 0
        assert ci.boost_for("a0", "b1") == 0.0


class TestRelationshipEvolution:
    """Co-access reinforcement with half-life decay (reference
    pkg/temporal/relationship_evolution.go)."""

    def test_reinforce_decay_persist(self):
        from nornicdb_amd.cognitive.temporal import RelationshipEvolution
        from nornicdb_amd.storage.memory import MemoryEngine
        from nornicdb_amd.storage.types import Edge, Node
        eng = MemoryEngine()
        eng.create_node(Node(id="a", labels=[], properties={}))
        eng.create_node(Node(id="b", labels=[], properties={}))
        eng.create_edge(Edge(id="e1", type="K", start_node="a",
                             end_node="b", properties={}))
        t = [0.0]
        ev = RelationshipEvolution(eng, now_fn=lambda: t[0])
        for _ in range(10):
            ev.record_coaccess("e1")
        assert ev.strength("e1") > 0.8
        assert ev.evolution_class("e1") == "strengthening"
        t[0] = 30 * 86400  # a month idle -> fades past 4 half-lives
        assert ev.strength("e1") < 0.1
        assert ev.evolution_class("e1") == "fading"
        assert ev.persist() == 1
        assert "_strength" in eng.get_edge("e1").properties


# ---------------------------------------------------------------------------
# reference-behavior tests for the deepened cognitive layer
# ---------------------------------------------------------------------------

class FakeClock:
    def __init__(self, t=1_700_000_000.0):
        self.t = t

    def __call__(self):
        return self.t

    def advance(self, dt):
        self.t += dt


class TestEvidenceBuffer:
    def test_three_signal_progression(self):
        """The reference's doc example: two signals accumulate, the third
        (from a distinct session) crosses the relates_to threshold."""
        from nornicdb_amd.cognitive.evidence import EvidenceBuffer
        clock = FakeClock()
        eb = EvidenceBuffer(now_fn=clock)
        assert eb.add_evidence("A", "B", "relates_to", 0.8, "coaccess", "s1") is False
        ok, reason = eb.check_threshold("A", "B", "relates_to")
        assert not ok and "more signal" in reason
        assert eb.add_evidence("A", "B", "relates_to", 0.7, "coaccess", "s2") is False
        assert eb.add_evidence("A", "B", "relates_to", 0.9, "similarity", "s3") is True
        # materialized entries leave the buffer
        assert eb.get_evidence("A", "B", "relates_to") is None
        assert eb.stats()["materialized"] == 1

    def test_session_requirement_blocks_single_session(self):
        from nornicdb_amd.cognitive.evidence import EvidenceBuffer
        eb = EvidenceBuffer(now_fn=FakeClock())
        for _ in range(5):
            assert eb.add_evidence("A", "B", "relates_to", 0.9, "x", "same") is False
        ok, reason = eb.check_threshold("A", "B", "relates_to")
        assert not ok and "session" in reason

    def test_low_scores_block(self):
        from nornicdb_amd.cognitive.evidence import EvidenceBuffer
        eb = EvidenceBuffer(now_fn=FakeClock())
        for i in range(4):
            assert eb.add_evidence("A", "B", "relates_to", 0.1, "x", f"s{i}") is False
        ok, reason = eb.check_threshold("A", "B", "relates_to")
        assert not ok and "score" in reason

    def test_evidence_expiry(self):
        from nornicdb_amd.cognitive.evidence import EvidenceBuffer
        clock = FakeClock()
        eb = EvidenceBuffer(now_fn=clock)
        eb.add_evidence("A", "B", "relates_to", 0.9, "x", "s1")
        eb.add_evidence("A", "B", "relates_to", 0.9, "x", "s2")
        clock.advance(25 * 3600)  # past relates_to MaxAge (24h)
        # stale evidence resets; this is signal 1 of a fresh window
        assert eb.add_evidence("A", "B", "relates_to", 0.9, "x", "s3") is False
        assert eb.get_evidence("A", "B", "relates_to").count == 1
        assert eb.stats()["expired"] == 1

    def test_per_label_thresholds(self):
        from nornicdb_amd.cognitive.evidence import EvidenceBuffer
        eb = EvidenceBuffer(now_fn=FakeClock())
        # similar_to: 2 signals, 1 session, score 0.7
        assert eb.add_evidence("A", "B", "similar_to", 0.9, "sim", "s1") is False
        assert eb.add_evidence("A", "B", "similar_to", 0.8, "sim", "s1") is True
        # coaccess: 5 signals, 3 sessions
        for i in range(4):
            assert eb.add_evidence("A", "B", "coaccess", 0.9, "co", f"s{i}") is False
        assert eb.add_evidence("A", "B", "coaccess", 0.9, "co", "s9") is True


class TestCooldownTable:
    def test_window_blocks_then_allows(self):
        from nornicdb_amd.cognitive.evidence import CooldownTable
        clock = FakeClock()
        ct = CooldownTable(now_fn=clock)
        assert ct.can_materialize("A", "B", "coaccess") is True
        ct.record_materialization("A", "B", "coaccess")
        assert ct.can_materialize("A", "B", "coaccess") is False
        assert 0 < ct.time_until_allowed("A", "B", "coaccess") <= 5 * 60
        clock.advance(5 * 60 + 1)  # coaccess cooldown is 5 min
        assert ct.can_materialize("A", "B", "coaccess") is True

    def test_per_label_windows_and_cleanup(self):
        from nornicdb_amd.cognitive.evidence import CooldownTable
        clock = FakeClock()
        ct = CooldownTable(now_fn=clock)
        ct.record_materialization("A", "B", "coaccess")    # 5 min
        ct.record_materialization("A", "B", "similar_to")  # 30 min
        clock.advance(6 * 60)
        assert ct.can_materialize("A", "B", "coaccess")
        assert not ct.can_materialize("A", "B", "similar_to")
        assert ct.cleanup() == 1
        assert len(ct) == 1


class TestInferenceDedupAndCooldown:
    def _engine_pair(self):
        from nornicdb_amd.storage import MemoryEngine, Node
        eng = MemoryEngine()
        eng.create_node(Node("a", ["Memory"], {"created_at": 0.0}))
        eng.create_node(Node("b", ["Memory"], {"created_at": 1.0}))
        return eng

    def test_no_duplicate_edges_on_repeat_store(self):
        from nornicdb_amd.cognitive import InferenceConfig, InferenceEngine
        clock = FakeClock()
        eng = self._engine_pair()
        inf = InferenceEngine(eng, config=InferenceConfig(
            evidence_required=1, temporal_window_s=10), now_fn=clock)
        n = eng.get_node("b")
        created1 = inf.on_store(n, "s1")
        assert len(created1) == 1  # temporal proximity link a<->b
        # repeat within cooldown: suggestion recurs, edge does not
        created2 = inf.on_store(n, "s1")
        assert created2 == []
        assert eng.edge_count() == 1

    def test_evidence_required_two_sessions(self):
        from nornicdb_amd.cognitive import InferenceConfig, InferenceEngine
        clock = FakeClock()
        eng = self._engine_pair()
        inf = InferenceEngine(eng, config=InferenceConfig(
            evidence_required=2, temporal_window_s=10, cooldown_s=0.0),
            now_fn=clock)
        n = eng.get_node("b")
        assert inf.on_store(n, "s1") == []      # 1 signal: buffered
        clock.advance(1)
        assert len(inf.on_store(n, "s2")) == 1  # 2nd signal, 2nd session
        assert eng.edge_count() == 1


class TestPatternDetector:
    def test_daily_pattern(self):
        import time as _time
        from nornicdb_amd.cognitive.patterns import DAILY, PatternDetector
        clock = FakeClock()
        pd = PatternDetector(now_fn=clock)
        # accesses concentrated at one wall-clock hour over 2 weeks
        base = clock.t - (clock.t % 86400)
        for day in range(14):
            pd.record_access("n", base + day * 86400 + 9 * 3600 + 100)
        pats = pd.detect_patterns("n")
        daily = [p for p in pats if p.type == DAILY]
        assert daily and daily[0].confidence > 0.5
        hour, _, conf = pd.peak_access_time("n")
        expect_hour = _time.localtime(base + 9 * 3600 + 100).tm_hour
        assert hour == expect_hour

    def test_burst_pattern(self):
        from nornicdb_amd.cognitive.patterns import BURST, PatternDetector
        clock = FakeClock()
        pd = PatternDetector(now_fn=clock)
        for i in range(12):
            pd.record_access("n", clock.t - 30 + i * 2)
        assert pd.has_pattern("n", BURST)

    def test_trend_patterns_from_velocity(self):
        from nornicdb_amd.cognitive.patterns import (DECAYING, GROWING,
                                                     PatternDetector)
        pd = PatternDetector(now_fn=FakeClock())
        assert pd.has_pattern("unseen", GROWING, velocity=0.2)
        assert pd.has_pattern("unseen", DECAYING, velocity=-0.2)
        assert not pd.has_pattern("unseen", GROWING, velocity=0.0)


class TestQueryLoadAdaptiveDecay:
    def test_trend_and_adaptive_interval(self):
        from nornicdb_amd.cognitive.patterns import QueryLoadPredictor
        clock = FakeClock()
        qlp = QueryLoadPredictor(now_fn=clock)
        # idle: base decay interval
        idle_interval = qlp.decay_interval(base=300.0)
        assert abs(idle_interval - 300.0) < 30.0
        # ramp load: 100 QPS for 30 seconds
        for s in range(30):
            qlp.record_queries(100)
            clock.advance(1.0)
        p = qlp.prediction()
        assert p.current_qps > 20
        assert p.total_queries == 3000
        # A/B: decay sweep stretches under load (reference query-load
        # adaptive decay behavior)
        loaded_interval = qlp.decay_interval(base=300.0)
        assert loaded_interval > idle_interval * 3

    def test_spike_anomaly(self):
        from nornicdb_amd.cognitive.patterns import QueryLoadPredictor
        clock = FakeClock()
        qlp = QueryLoadPredictor(now_fn=clock, spike_threshold=2.0)
        for s in range(10):
            qlp.record_queries(1)
            clock.advance(1.0)
        for s in range(5):
            qlp.record_queries(200)
            clock.advance(1.0)
        p = qlp.prediction()
        assert p.trend == "increasing"
        assert p.is_anomaly and p.anomaly_type == "spike"


class TestSessionBoundaries:
    def test_session_id_changes_after_gap(self):
        from nornicdb_amd.cognitive.temporal import AccessTracker, SESSION_GAP
        clock = FakeClock()
        tr = AccessTracker(now_fn=clock)
        assert tr.is_session_boundary()
        tr.record("n1")
        s1 = tr.session_id
        clock.advance(10)
        tr.record("n2")
        assert tr.session_id == s1            # same session
        assert not tr.is_session_boundary()
        clock.advance(SESSION_GAP + 1)
        assert tr.is_session_boundary()
        tr.record("n3")
        assert tr.session_id != s1            # new session
        assert len(tr.sessions()) == 2


class TestNodeConfig:
    """Per-node overrides gating auto-links (reference node_config.go)."""

    def _inf(self):
        from nornicdb_amd.cognitive import InferenceConfig, InferenceEngine
        from nornicdb_amd.storage import MemoryEngine, Node
        clock = FakeClock()
        eng = MemoryEngine()
        eng.create_node(Node("a", ["Memory"], {"created_at": 0.0}))
        eng.create_node(Node("b", ["Memory"], {"created_at": 1.0}))
        inf = InferenceEngine(eng, config=InferenceConfig(
            evidence_required=1, temporal_window_s=10), now_fn=clock)
        return eng, inf

    def test_denied_pair_never_links(self):
        eng, inf = self._inf()
        inf.node_configs.get_or_create("a").add_deny("b")
        assert inf.on_store(eng.get_node("b"), "s1") == []
        assert eng.edge_count() == 0

    def test_low_trust_raises_confidence_bar(self):
        eng, inf = self._inf()
        from nornicdb_amd.storage import TRUST_LOW
        inf.node_configs.get_or_create("b").trust_level = TRUST_LOW
        # temporal suggestion confidence ~0.5-0.8 < 0.5+0.2 bar -> blocked
        created = inf.on_store(eng.get_node("b"), "s1")
        assert created == [] or all(
            e.properties["confidence"] >= 0.7 for e in created)

    def test_pinned_always_allowed(self):
        eng, inf = self._inf()
        from nornicdb_amd.storage import TRUST_LOW
        c = inf.node_configs.get_or_create("b")
        c.trust_level = TRUST_LOW
        c.add_pin("a")
        assert len(inf.on_store(eng.get_node("b"), "s1")) == 1

    def test_label_cap(self):
        from nornicdb_amd.storage import (LabelConfig, MemoryEngine, Node,
                                          NodeConfigStore)
        from nornicdb_amd.storage.types import Edge
        eng = MemoryEngine()
        for nid in ("x", "y", "z"):
            eng.create_node(Node(nid, [], {}))
        eng.create_edge(Edge("e1", "REL", "x", "y", {}))
        store = NodeConfigStore(eng)
        c = store.get_or_create("x")
        c.label_configs["REL"] = LabelConfig(max_edges=1)
        ok, why = store.is_edge_allowed("x", "z", "REL", 1.0, 0.0)
        assert not ok and "max capacity" in why
        ok, _ = store.is_edge_allowed("x", "z", "OTHER", 1.0, 0.0)
        assert ok


def test_decay_background_ticker():
    """DecayManager.start/stop (reference pkg/decay Manager.Start): the
    ticker runs cycles at the interval and stop() joins cleanly."""
    import time

    from nornicdb_amd.cognitive import DecayManager
    from nornicdb_amd.storage.memory import MemoryEngine
    from nornicdb_amd.storage.types import Node

    eng = MemoryEngine()
    eng.create_node(Node(id="m1", labels=["Memory"],
                         properties={"importance": 0.9,
                                     "created_at": time.time(),
                                     "last_accessed": time.time()}))
    dm = DecayManager(eng)
    calls = []
    orig = dm.run_cycle
    dm.run_cycle = lambda: calls.append(orig())
    dm.start(interval_s=0.05)
    time.sleep(0.3)
    dm.stop()
    assert len(calls) >= 2
    n = len(calls)
    time.sleep(0.15)
    assert len(calls) == n  # stopped for real
