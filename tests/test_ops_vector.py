"""Numerics tests for tier-1 vector kernels.

GPU tests compare every HIP kernel against a plain PyTorch fp32 reference
(same contract as the reference repo's GPU-vs-CPU parity tests,
e.g. pkg/gpu/cuda/cuda_test.go).
"""

import pytest
import torch

from nornicdb_amd.ops import knn_search, knn_search_exact, l2_normalize_, fill_random_unit_


def test_l2_normalize_cpu():
    x = torch.randn(64, 128)
    l2_normalize_(x)
    n = torch.linalg.vector_norm(x, dim=-1)
    assert torch.allclose(n, torch.ones_like(n), atol=1e-4)


def test_knn_cpu_matches_exact():
    torch.manual_seed(0)
    db = torch.randn(500, 64)
    q = torch.randn(7, 64)
    s, i = knn_search(db, q, 5)
    s2, i2 = knn_search_exact(db, q, 5)
    assert torch.equal(i, i2)
    assert torch.allclose(s, s2)


def test_fill_random_unit_cpu():
    x = torch.empty(16, 64, dtype=torch.bfloat16)
    fill_random_unit_(x, row_base=0, seed=7)
    n = torch.linalg.vector_norm(x.float(), dim=-1)
    assert torch.allclose(n, torch.ones_like(n), atol=2e-2)


@pytest.mark.gpu
def test_l2_normalize_gpu_bf16():
    torch.manual_seed(1)
    x = torch.randn(4096, 1024, device="cuda").to(torch.bfloat16)
    ref = x.float()
    ref = ref / torch.linalg.vector_norm(ref, dim=-1, keepdim=True)
    l2_normalize_(x)
    torch.cuda.synchronize()
    err = (x.float() - ref).abs().max().item()
    assert err < 2e-2  # bf16 storage error
    n = torch.linalg.vector_norm(x.float(), dim=-1)
    assert (n - 1).abs().max().item() < 1e-2


@pytest.mark.gpu
def test_l2_normalize_gpu_f32():
    torch.manual_seed(2)
    x = torch.randn(1000, 512, device="cuda")
    ref = x / torch.linalg.vector_norm(x, dim=-1, keepdim=True)
    l2_normalize_(x)
    torch.cuda.synchronize()
    assert torch.allclose(x, ref, atol=1e-5)


@pytest.mark.gpu
def test_fill_random_unit_gpu():
    x = torch.empty(10000, 1024, device="cuda", dtype=torch.bfloat16)
    fill_random_unit_(x, row_base=1234, seed=42)
    torch.cuda.synchronize()
    n = torch.linalg.vector_norm(x.float(), dim=-1)
    assert (n - 1).abs().max().item() < 2e-2
    # determinism + row_base consistency: regenerating a slice with shifted
    # base must reproduce the same rows
    y = torch.empty(100, 1024, device="cuda", dtype=torch.bfloat16)
    fill_random_unit_(y, row_base=1234 + 500, seed=42)
    torch.cuda.synchronize()
    assert torch.equal(x[500:600], y)
    # rows are distinct
    assert not torch.equal(x[0], x[1])


@pytest.mark.gpu
@pytest.mark.parametrize("q_count", [1, 3, 16])
def test_knn_gemv_gpu_matches_fp32(q_count):
    torch.manual_seed(3)
    db = torch.randn(50_000, 256, device="cuda")
    db = db / torch.linalg.vector_norm(db, dim=-1, keepdim=True)
    q = db[:q_count] + 0.1 * torch.randn(q_count, 256, device="cuda")
    q = q / torch.linalg.vector_norm(q, dim=-1, keepdim=True)
    dbb = db.to(torch.bfloat16)
    qb = q.to(torch.bfloat16)

    s, i = knn_search(dbb, qb, 10, row_base=777)
    torch.cuda.synchronize()
    s_ref, i_ref = knn_search_exact(dbb.float(), qb.float(), 10, row_base=777)

    # exact index match demanded at fp32-identical inputs (bf16 both paths)
    match = (i == i_ref).float().mean().item()
    assert match > 0.95, f"top-k overlap too low: {match}"
    assert torch.allclose(s, s_ref, atol=1e-2)
    # scores must be sorted descending
    assert (s[:, :-1] >= s[:, 1:] - 1e-6).all()


@pytest.mark.gpu
def test_knn_large_q_gemm_path():
    torch.manual_seed(4)
    db = torch.randn(100_000, 128, device="cuda").to(torch.bfloat16)
    q = torch.randn(64, 128, device="cuda").to(torch.bfloat16)
    s, i = knn_search(db, q, 10)
    torch.cuda.synchronize()
    s_ref, i_ref = knn_search_exact(db.float(), q.float(), 10)
    overlap = 0.0
    for r in range(64):
        overlap += len(set(i[r].tolist()) & set(i_ref[r].tolist())) / 10
    assert overlap / 64 > 0.95


@pytest.mark.gpu
@pytest.mark.parametrize("n,q_count", [(50_000, 256), (12_345, 100), (128, 256), (100_001, 256)])
def test_knn_mfma_gpu_matches_fp32(n, q_count):
    """Random asymmetric inputs: catches any MFMA fragment-layout transpose
    (guide G9: symmetric inputs cannot detect row/col swaps)."""
    torch.manual_seed(n + q_count)
    db = torch.randn(n, 1024, device="cuda")
    db = db / torch.linalg.vector_norm(db, dim=-1, keepdim=True)
    q = torch.randn(q_count, 1024, device="cuda")
    q = q / torch.linalg.vector_norm(q, dim=-1, keepdim=True)
    dbb = db.to(torch.bfloat16)
    qb = q.to(torch.bfloat16)

    s, i = knn_search(dbb, qb, 10, row_base=1000)
    torch.cuda.synchronize()
    s_ref, i_ref = knn_search_exact(dbb.float(), qb.float(), 10, row_base=1000)

    match = (i == i_ref).float().mean().item()
    assert match > 0.93, f"top-k index overlap too low: {match}"
    assert torch.allclose(s, s_ref, atol=2e-2), (s - s_ref).abs().max()
    assert (s[:, :-1] >= s[:, 1:] - 1e-6).all()


@pytest.mark.gpu
def test_knn_mfma_large_query_batch():
    """Multi-GPU shape: 2048 gathered queries chunk through the fused kernel."""
    torch.manual_seed(77)
    db = torch.randn(200_000, 1024, device="cuda")
    db = db / torch.linalg.vector_norm(db, dim=-1, keepdim=True)
    q = db[:2048] + 0.05 * torch.randn(2048, 1024, device="cuda")
    q = (q / torch.linalg.vector_norm(q, dim=-1, keepdim=True)).to(torch.bfloat16)
    dbb = db.to(torch.bfloat16)
    s, i = knn_search(dbb, q, 10)
    torch.cuda.synchronize()
    assert s.shape == (2048, 10) and i.shape == (2048, 10)
    # spot-check a sample against exact
    s_ref, i_ref = knn_search_exact(dbb[:, :].float(), q[:64].float(), 10)
    match = (i[:64] == i_ref).float().mean().item()
    assert match > 0.9, match


def test_fp8_quantize_cpu_roundtrip():
    """e4m3fn quantization error on normalized vectors is small and the
    CPU (dequantized-exact) search path handles fp8 corpora."""
    from nornicdb_amd.ops.knn import knn_search, quantize_fp8
    torch.manual_seed(0)
    db = torch.nn.functional.normalize(torch.randn(200, 128), dim=-1)
    db8 = quantize_fp8(db)
    err = (db8.float() - db).abs().max()
    assert float(err) < 0.05, err
    q = db[:4]
    s, i = knn_search(db8, q, 5)
    assert (i[:, 0] == torch.arange(4)).all()


@pytest.mark.gpu
class TestKnnFp8GPU:
    def test_scores_match_dequantized_oracle(self):
        """Kernel fp8 x fp8 fp32-accum scores == torch fp32 matmul over
        the SAME e4m3 values (both sides see identical inputs)."""
        from nornicdb_amd.ops import require_native
        from nornicdb_amd.ops.knn import quantize_fp8
        nat = require_native()
        torch.manual_seed(1)
        db = torch.nn.functional.normalize(
            torch.randn(96 * 13, 1024, device="cuda"), dim=-1)
        db8 = quantize_fp8(db)
        q = db[:17]
        q8 = q.to(torch.float8_e4m3fn)
        qpad = torch.cat([q8, torch.zeros(256 - 17, 1024, device="cuda",
                                          dtype=torch.float8_e4m3fn)])
        s, i = nat.knn_fp8(db8.view(torch.uint8),
                           qpad.contiguous().view(torch.uint8), 0, 10)
        ref = (q8.float() @ db8.float().T)
        rs, ri = torch.topk(ref, 10, dim=-1)
        assert torch.allclose(s[:17], rs, atol=1e-3), (s[:17] - rs).abs().max()
        # ties can permute equal-score indices; top-1 must agree
        assert (i[:17, 0] == ri[:, 0]).all()

    def test_recall_vs_fp32_exact(self):
        """fp8 e4m3 recall@10 vs fp32 exact on WORST-CASE gaussian
        vectors (near-uniform score gaps): measured ~0.91. Real embedding
        corpora (clustered) fare much better; int8 mode (below) holds
        >= 0.95 even here."""
        from nornicdb_amd.ops.knn import knn_search, knn_search_exact, quantize_fp8
        torch.manual_seed(2)
        db = torch.nn.functional.normalize(
            torch.randn(96 * 500, 1024, device="cuda"), dim=-1)
        q = torch.nn.functional.normalize(
            torch.randn(64, 1024, device="cuda"), dim=-1)
        es, ei = knn_search_exact(db, q, 10)
        s, i = knn_search(quantize_fp8(db), q, 10)
        hit = 0
        for r in range(64):
            hit += len(set(i[r].tolist()) & set(ei[r].tolist()))
        recall = hit / (64 * 10)
        assert recall >= 0.85, recall
        # self-match: query = db row must return itself top-1
        s2, i2 = knn_search(quantize_fp8(db), db[:32], 10)
        assert (i2[:, 0] == torch.arange(32, device="cuda")).float().mean() > 0.95

    def test_embedding_index_fp8_gpu(self):
        import numpy as np

        from nornicdb_amd.search.embedding_index import EmbeddingIndex
        idx = EmbeddingIndex(128, device="cuda", quant="fp8")
        rng = np.random.default_rng(3)
        vecs = rng.normal(size=(500, 128)).astype("float32")
        vecs /= np.linalg.norm(vecs, axis=1, keepdims=True)
        idx.add_batch([f"n{i}" for i in range(500)], vecs)
        hits = idx.search(vecs[42], k=5)
        assert hits[0][0] == "n42" and hits[0][1] > 0.97


@pytest.mark.gpu
class TestKnnInt8GPU:
    def test_scores_match_dequantized_oracle(self):
        """i8 kernel (i32 dot * sa * sq) == fp32 matmul over the same
        dequantized values — i32 dots are exact, so agreement is to
        fp32 rounding."""
        from nornicdb_amd.ops import require_native
        from nornicdb_amd.ops.knn import knn_search_int8, quantize_int8
        torch.manual_seed(4)
        db = torch.nn.functional.normalize(
            torch.randn(96 * 13, 1024, device="cuda"), dim=-1)
        db8, sa = quantize_int8(db)
        q = db[:17]
        s, i = knn_search_int8(db8, sa, q, 10)
        qi, sq = quantize_int8(q)
        ref = (qi.float() * sq[:, None]) @ (db8.float() * sa[:, None]).T
        rs, ri = torch.topk(ref, 10, dim=-1)
        assert torch.allclose(s, rs, atol=1e-4), (s - rs).abs().max()
        assert (i[:, 0] == ri[:, 0]).all()

    def test_recall_vs_fp32_exact(self):
        """int8 per-row-scale recall@10 >= 0.95 even on worst-case
        gaussian corpora — the quantized-mode contract."""
        from nornicdb_amd.ops.knn import (knn_search_exact, knn_search_int8,
                                          quantize_int8)
        torch.manual_seed(5)
        db = torch.nn.functional.normalize(
            torch.randn(96 * 500, 1024, device="cuda"), dim=-1)
        q = torch.nn.functional.normalize(
            torch.randn(64, 1024, device="cuda"), dim=-1)
        es, ei = knn_search_exact(db, q, 10)
        db8, sa = quantize_int8(db)
        s, i = knn_search_int8(db8, sa, q, 10)
        hit = 0
        for r in range(64):
            hit += len(set(i[r].tolist()) & set(ei[r].tolist()))
        recall = hit / (64 * 10)
        assert recall >= 0.95, recall
        s2, i2 = knn_search_int8(db8, sa, db[:32], 10)
        assert (i2[:, 0] == torch.arange(32, device="cuda")).all()

    def test_embedding_index_int8_gpu(self):
        import numpy as np

        from nornicdb_amd.search.embedding_index import EmbeddingIndex
        idx = EmbeddingIndex(128, device="cuda", quant="int8")
        rng = np.random.default_rng(6)
        vecs = rng.normal(size=(500, 128)).astype("float32")
        vecs /= np.linalg.norm(vecs, axis=1, keepdims=True)
        idx.add_batch([f"n{i}" for i in range(500)], vecs)
        hits = idx.search(vecs[42], k=5)
        assert hits[0][0] == "n42" and hits[0][1] > 0.99
