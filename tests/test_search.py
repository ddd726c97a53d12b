"""Search layer tests: HNSW recall, BM25 ranking, RRF/MMR, k-means
routing, embedding index, hybrid service with storage sync.

Models reference pkg/search tests (hnsw_recall_test.go recall floor,
kmeans_benchmark_test.go, search service tests).
"""

import numpy as np
import pytest
import torch

from nornicdb_amd.search import (ClusterIndex, EmbeddingIndex, FulltextIndex,
                                 HNSWIndex, SearchService, kmeans,
                                 mmr_diversify, optimal_k, rrf_fuse)
from nornicdb_amd.storage import MemoryEngine, Node


def unit_rows(n, d, seed=0):
    rng = np.random.default_rng(seed)
    x = rng.standard_normal((n, d)).astype(np.float32)
    return x / np.linalg.norm(x, axis=1, keepdims=True)


class TestHNSW:
    def test_recall_floor(self):
        """HNSW recall@10 vs exact must clear 0.9 on 2000x64."""
        x = unit_rows(2000, 64, seed=1)
        idx = HNSWIndex(64)
        for i, v in enumerate(x):
            idx.add(f"v{i}", v)
        rng = np.random.default_rng(2)
        hits = total = 0
        for _ in range(20):
            q = x[rng.integers(0, 2000)] + 0.05 * rng.standard_normal(64).astype(np.float32)
            q /= np.linalg.norm(q)
            exact = np.argsort(-(x @ q))[:10]
            exact_ids = {f"v{i}" for i in exact}
            got = {i for i, _ in idx.search(q, 10)}
            hits += len(got & exact_ids)
            total += 10
        assert hits / total >= 0.9, f"recall {hits/total}"

    def test_delete_tombstone(self):
        x = unit_rows(100, 16)
        idx = HNSWIndex(16)
        for i, v in enumerate(x):
            idx.add(f"v{i}", v)
        assert idx.remove("v0")
        assert not idx.remove("v0")
        res = idx.search(x[0], 5)
        assert "v0" not in [i for i, _ in res]
        assert len(idx) == 99

    def test_rebuild_after_many_deletes(self):
        x = unit_rows(200, 16)
        idx = HNSWIndex(16)
        for i, v in enumerate(x):
            idx.add(f"v{i}", v)
        for i in range(100):
            idx.remove(f"v{i}")
        assert len(idx) == 100
        res = idx.search(x[150], 5)
        assert res[0][0] == "v150"


class TestBM25:
    def test_ranking(self):
        ft = FulltextIndex()
        ft.index("d1", "the quick brown fox jumps over the lazy dog")
        ft.index("d2", "quick quick quick brown")
        ft.index("d3", "completely unrelated text about databases")
        res = ft.search("quick brown", 3)
        assert res[0][0] == "d2"
        assert {r[0] for r in res[:2]} == {"d1", "d2"}

    def test_remove(self):
        ft = FulltextIndex()
        ft.index("d1", "hello world")
        ft.remove("d1")
        assert ft.search("hello") == []

    def test_idf_favors_rare_terms(self):
        ft = FulltextIndex()
        for i in range(10):
            ft.index(f"c{i}", "common words everywhere common")
        ft.index("rare", "common words plus zebra")
        assert ft.search("zebra")[0][0] == "rare"


class TestFusion:
    def test_rrf(self):
        a = [("x", 1.0), ("y", 0.9), ("z", 0.8)]
        b = [("y", 5.0), ("x", 4.0)]
        fused = rrf_fuse([a, b])
        assert {fused[0][0], fused[1][0]} == {"x", "y"}
        assert fused[2][0] == "z"

    def test_mmr_diversifies(self):
        vecs = {
            "a": np.array([1.0, 0.0], np.float32),
            "a2": np.array([0.999, 0.04], np.float32),
            "b": np.array([0.0, 1.0], np.float32),
        }
        for k in vecs:
            vecs[k] = vecs[k] / np.linalg.norm(vecs[k])
        cands = [("a", 1.0), ("a2", 0.99), ("b", 0.8)]
        out = mmr_diversify(cands, vecs, 2, lambda_=0.5)
        assert [o[0] for o in out] == ["a", "b"]


class TestKMeans:
    def test_separated_clusters(self):
        rng = np.random.default_rng(3)
        c1 = rng.standard_normal((200, 8)).astype(np.float32) * 0.05 + 5
        c2 = rng.standard_normal((200, 8)).astype(np.float32) * 0.05 - 5
        x = torch.from_numpy(np.vstack([c1, c2]))
        cents, assign = kmeans(x, 2, seed=1)
        a = assign[:200]
        b = assign[200:]
        assert len(set(a.tolist())) == 1
        assert len(set(b.tolist())) == 1
        assert a[0] != b[0]

    def test_optimal_k(self):
        assert optimal_k(100_000) == int((100_000 / 2) ** 0.5)

    def test_cluster_index_routing(self):
        x = unit_rows(500, 16, seed=5)
        ids = [f"v{i}" for i in range(500)]
        ci = ClusterIndex(nprobe=2)
        ci.cluster(ids, torch.from_numpy(x), k=8)
        cands = ci.candidates(x[7])
        assert "v7" in cands
        assert len(cands) < 500

    def test_incremental_add_and_drift(self):
        x = unit_rows(100, 8, seed=6)
        ids = [f"v{i}" for i in range(100)]
        ci = ClusterIndex()
        ci.cluster(ids, torch.from_numpy(x), k=4)
        ci.add("new", x[0])
        assert "new" in ci.candidates(x[0], nprobe=1)


class TestEmbeddingIndex:
    def test_add_search_remove(self):
        x = unit_rows(300, 32, seed=7)
        ei = EmbeddingIndex(32, device="cpu")
        ei.add_batch([f"v{i}" for i in range(300)], x)
        res = ei.search(x[5], 3)
        assert res[0][0] == "v5"
        assert res[0][1] > 0.99
        ei.remove("v5")
        res = ei.search(x[5], 3)
        assert res[0][0] != "v5"
        assert len(ei) == 299

    def test_score_subset(self):
        x = unit_rows(50, 16, seed=8)
        ei = EmbeddingIndex(16, device="cpu")
        ei.add_batch([f"v{i}" for i in range(50)], x)
        res = ei.score_subset(x[3], ["v1", "v3", "v9"])
        assert res[0][0] == "v3"

    def test_update_existing(self):
        ei = EmbeddingIndex(4, device="cpu")
        ei.add("a", [1, 0, 0, 0])
        ei.add("a", [0, 1, 0, 0])
        assert len(ei) == 1
        assert ei.search([0, 1, 0, 0], 1)[0][1] > 0.99


class TestSearchService:
    def _svc(self, dims=16):
        eng = MemoryEngine()
        svc = SearchService(eng, dims=dims, device="cpu", use_hnsw=True)
        return eng, svc

    def test_event_sync_and_hybrid(self):
        eng, svc = self._svc()
        x = unit_rows(20, 16, seed=9)
        for i in range(20):
            eng.create_node(Node(f"n{i}", ["Doc"],
                                 {"title": f"doc {i} about topic{i % 3}"},
                                 embedding=list(map(float, x[i]))))
        # text only
        res = svc.text_search("topic1", k=5)
        assert res and all("topic1" in r.node.properties["title"] for r in res)
        # vector only
        res = svc.vector_search(x[4], k=3)
        assert res[0].id == "n4"
        # hybrid
        res = svc.search(query="doc topic2", query_vec=x[2], k=5)
        assert any(r.id == "n2" for r in res)

    def test_delete_removes_from_indexes(self):
        eng, svc = self._svc()
        x = unit_rows(5, 16)
        for i in range(5):
            eng.create_node(Node(f"n{i}", ["D"], {"title": "hello"},
                                 embedding=list(map(float, x[i]))))
        eng.delete_node("n0")
        assert all(r.id != "n0" for r in svc.vector_search(x[0], k=5))
        assert len(svc.fulltext) == 4

    def test_label_filter(self):
        eng, svc = self._svc()
        x = unit_rows(4, 16)
        eng.create_node(Node("a", ["A"], {"title": "same text"},
                             embedding=list(map(float, x[0]))))
        eng.create_node(Node("b", ["B"], {"title": "same text"},
                             embedding=list(map(float, x[1]))))
        res = svc.text_search("same text", k=5, labels=["B"])
        assert [r.id for r in res] == ["b"]

    def test_build_indexes_scan(self):
        eng = MemoryEngine()
        x = unit_rows(3, 16)
        for i in range(3):
            eng.create_node(Node(f"n{i}", [], {"title": "prebuilt"},
                                 embedding=list(map(float, x[i]))))
        svc = SearchService(eng, dims=16, device="cpu")
        assert len(svc.fulltext) == 0
        svc.build_indexes()
        assert len(svc.fulltext) == 3


class TestVectorSpaceRegistry:
    """pkg/vectorspace parity: named vector spaces per (db, entity, name)."""

    def test_register_get_list_drop(self):
        from nornicdb_amd.search.vectorspace import Registry, VectorSpace
        reg = Registry()
        reg.register(VectorSpace("neo4j", "Doc", "emb", 1024, "cosine"))
        reg.register(VectorSpace("neo4j", "Chunk", "emb2", 512, "euclidean"))
        assert reg.get("neo4j", "Doc", "emb").dims == 1024
        assert len(reg.list("neo4j")) == 2
        assert reg.drop("neo4j", "Doc", "emb") is True
        assert reg.get("neo4j", "Doc", "emb") is None


class TestHTTPEmbedderOffline:
    """embed/http_providers: transport injection (no network here)."""

    def test_injected_transport(self):
        import numpy as np

        from nornicdb_amd.embed.http_providers import HTTPEmbedder
        calls = []

        def fake_transport(url, payload, headers):
            calls.append(url)
            texts = payload.get("input") or payload.get("prompt") or []
            if isinstance(texts, str):
                texts = [texts]
            return {"data": [{"embedding": [0.1] * 8} for _ in texts]}

        e = HTTPEmbedder("openai", base_url="http://fake", dims=8,
                         transport=fake_transport)
        out = e.embed_batch(["a", "b"])
        assert out.shape == (2, 8)
        assert calls


# ---------------------------------------------------------------------------
# HIP k-means kernel suite (csrc/kmeans.hip) vs torch oracle
# ---------------------------------------------------------------------------

def test_cluster_index_incremental_updates_cpu():
    import torch
    from nornicdb_amd.search.kmeans import ClusterIndex
    torch.manual_seed(0)
    ci = ClusterIndex()
    mat = torch.cat([torch.randn(20, 8) + 4, torch.randn(20, 8) - 4])
    ids = [f"v{i}" for i in range(40)]
    ci.cluster(ids, mat, k=2, seed=1)
    c_before = ci.centroids.clone()
    cnt_before = ci.counts.clone()
    # adding a point near cluster of v0 must move that centroid toward it
    v = mat[0] + 0.5
    ci.add("new", v)
    cl = ci._id2cluster["new"]
    assert int(ci.counts[cl]) == int(cnt_before[cl]) + 1
    moved = (ci.centroids[cl] - c_before[cl]).abs().sum()
    assert float(moved) > 0
    # removing it restores the centroid (same arithmetic inverted)
    ci.remove("new", v)
    assert torch.allclose(ci.centroids[cl], c_before[cl], atol=1e-4)
    assert int(ci.counts[cl]) == int(cnt_before[cl])


@pytest.mark.gpu
def test_kmeans_kernels_match_torch_gpu():
    import torch
    from nornicdb_amd.ops import require_native
    nat = require_native()
    torch.manual_seed(3)
    n, d, k = 5000, 512, 32
    x = torch.randn(n, d, device="cuda")
    xb = x.to(torch.bfloat16)
    cent = torch.randn(k, d, device="cuda")
    cb = cent.to(torch.bfloat16)
    a, d2 = nat.kmeans_assign(xb, cb, (cent * cent).sum(-1))
    # oracle in the SAME bf16 precision
    xf, cf = xb.float(), cb.float()
    ref_d2 = ((xf * xf).sum(-1, keepdim=True) + (cf * cf).sum(-1)[None]
              - 2 * xf @ cf.T)
    ref_m, ref_a = ref_d2.min(dim=1)
    agree = (a.long() == ref_a).float().mean().item()
    assert agree > 0.99, agree  # ties may flip at bf16
    match = a.long() == ref_a
    assert (d2[match] - ref_m[match].clamp_min(0)).abs().max().item() < 0.5
    # accumulate + finalize
    sums, counts = nat.kmeans_accum(xb, a, k)
    ref_counts = torch.bincount(a.long(), minlength=k).int()
    assert (counts == ref_counts).all()
    ref_sums = torch.zeros(k, d, device="cuda")
    ref_sums.index_add_(0, a.long(), xf)
    assert (sums - ref_sums).abs().max().item() < 0.05
    newc, drift2 = nat.kmeans_finalize(sums, counts, cent)
    ref_new = ref_sums / ref_counts.clamp_min(1)[:, None].float()
    ref_new[ref_counts == 0] = cent[ref_counts == 0]
    assert (newc - ref_new).abs().max().item() < 0.05
    ref_drift = ((ref_new - cent) ** 2).sum(-1)
    assert (drift2 - ref_drift).abs().max().item() < 0.5
    # k-means++ min-distance update
    d2b = torch.full((n,), 3.4e38, device="cuda")
    nat.kmeanspp_update(xb, cb[0].contiguous(), float((cf[0] ** 2).sum()), d2b)
    ref_pp = ((xf - cf[0]) ** 2).sum(-1)
    assert (d2b - ref_pp).abs().max().item() < 0.5
    # single-point incremental update
    cents = cent.clone()
    cnts = ref_counts.clone()
    v = xb[0].contiguous()
    nat.kmeans_point_update(cents, cnts, v, 3, 1)
    cnt3 = int(ref_counts[3])
    ref_upd = (cent[3] * cnt3 + xf[0]) / (cnt3 + 1)
    assert (cents[3] - ref_upd).abs().max().item() < 1e-2
    assert int(cnts[3]) == cnt3 + 1


@pytest.mark.gpu
def test_kmeans_gpu_end_to_end_quality():
    """Full kmeans() on the HIP path converges to the same inertia
    ballpark as the torch oracle path."""
    import torch
    from nornicdb_amd.search.kmeans import kmeans
    torch.manual_seed(5)
    centers = torch.randn(8, 512, device="cuda") * 5
    x = (centers.repeat_interleave(500, 0)
         + 0.3 * torch.randn(4000, 512, device="cuda"))
    c, a = kmeans(x, 8, iters=20, seed=2)
    assert c.shape == (8, 512)
    # every true center has one recovered centroid nearby
    dists = ((centers[:, None, :] - c[None]) ** 2).sum(-1).min(dim=1).values
    assert float(dists.max()) < 30.0, float(dists.max())
    # assignment is consistent with nearest centroid
    d2 = ((x[:100, None, :] - c[None]) ** 2).sum(-1)
    assert (d2.argmin(1) == a[:100]).float().mean().item() > 0.97
