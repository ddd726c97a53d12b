"""Numerics tests for fused encoder kernels vs fp32 torch references."""

import pytest
import torch
import torch.nn.functional as F

from nornicdb_amd.ops.encoder import add_layernorm, bias_gelu, mean_pool_l2norm
from nornicdb_amd.ops.attention import flash_attention_nc


def test_cpu_fallbacks():
    a = torch.randn(4, 8, 64)
    b = torch.randn(4, 8, 64)
    g = torch.ones(64); be = torch.zeros(64)
    y = add_layernorm(a, b, g, be)
    ref = F.layer_norm(a + b, (64,), g, be)
    assert torch.allclose(y, ref, atol=1e-5)
    x = torch.randn(4, 64)
    assert torch.allclose(bias_gelu(x, torch.zeros(64)), F.gelu(x), atol=1e-5)
    p = mean_pool_l2norm(torch.randn(2, 5, 64))
    assert torch.allclose(torch.linalg.vector_norm(p, dim=-1), torch.ones(2), atol=1e-5)


@pytest.mark.gpu
def test_add_layernorm_gpu():
    torch.manual_seed(0)
    a = torch.randn(333, 1024, device="cuda").to(torch.bfloat16)
    b = torch.randn(333, 1024, device="cuda").to(torch.bfloat16)
    g = torch.randn(1024, device="cuda")
    be = torch.randn(1024, device="cuda")
    y = add_layernorm(a, b, g, be)
    ref = F.layer_norm((a.float() + b.float()), (1024,), g, be)
    torch.cuda.synchronize()
    assert (y.float() - ref).abs().max().item() < 0.1


@pytest.mark.gpu
def test_bias_gelu_gpu():
    torch.manual_seed(1)
    x = torch.randn(1000, 4096, device="cuda").to(torch.bfloat16)
    b = torch.randn(4096, device="cuda")
    y = bias_gelu(x, b)
    ref = F.gelu(x.float() + b.float())
    torch.cuda.synchronize()
    assert (y.float() - ref).abs().max().item() < 0.1


@pytest.mark.gpu
def test_mean_pool_l2norm_gpu():
    torch.manual_seed(2)
    x = torch.randn(8, 128, 1024, device="cuda").to(torch.bfloat16)
    mask = torch.ones(8, 128, device="cuda", dtype=torch.long)
    mask[:, 100:] = 0
    y = mean_pool_l2norm(x, mask)
    m = mask[..., None].float()
    ref = (x.float() * m).sum(1) / m.sum(1)
    ref = ref / torch.linalg.vector_norm(ref, dim=-1, keepdim=True)
    torch.cuda.synchronize()
    assert (y - ref).abs().max().item() < 0.01


@pytest.mark.gpu
@pytest.mark.parametrize("b,h,s", [(2, 4, 128), (1, 16, 256), (3, 2, 64)])
def test_flash_attn_nc_gpu(b, h, s):
    torch.manual_seed(3)
    q = torch.randn(b, h, s, 64, device="cuda").to(torch.bfloat16)
    k = torch.randn(b, h, s, 64, device="cuda").to(torch.bfloat16)
    v = torch.randn(b, h, s, 64, device="cuda").to(torch.bfloat16)
    with torch.no_grad():
        y = flash_attention_nc(q, k, v)
    ref = F.scaled_dot_product_attention(q.float(), k.float(), v.float())
    torch.cuda.synchronize()
    err = (y.float() - ref).abs().max().item()
    assert err < 0.05, f"max err {err}"


@pytest.mark.gpu
def test_encoder_fused_vs_eager_gpu():
    """Whole-model check: fused inference path matches eager fp32-ish."""
    from nornicdb_amd.models import BgeM3Config, BgeM3Encoder
    torch.manual_seed(4)
    cfg = BgeM3Config(vocab_size=5000, hidden_size=256, num_layers=2,
                      num_heads=4, intermediate_size=512, max_position=512)
    m = BgeM3Encoder(cfg).init_small().cuda().to(torch.bfloat16).eval()
    tok = torch.randint(0, 5000, (4, 128), device="cuda")
    with torch.no_grad():
        fused = m(tok)
    ref = BgeM3Encoder(cfg).cuda().float().eval()
    ref.load_state_dict({k: v.float() for k, v in m.state_dict().items()})
    with torch.enable_grad():  # disables fused path
        eager = ref(tok)
    cos = F.cosine_similarity(fused, eager).min().item()
    assert cos > 0.98, f"fused/eager cosine {cos}"


# ---------------------------------------------------------------------------
# hand-written MFMA GEMM (csrc/gemm.hip)
# ---------------------------------------------------------------------------

def test_linear_act_cpu_fallback():
    from nornicdb_amd.ops.gemm import ACT_GELU, linear_act
    x = torch.randn(5, 7, 96)
    w = torch.randn(33, 96)
    b = torch.randn(33)
    y = linear_act(x, w, b, ACT_GELU)
    assert torch.allclose(y, F.gelu(F.linear(x, w, b)), atol=1e-5)


@pytest.mark.gpu
@pytest.mark.parametrize("m,n,k,act", [
    (256, 256, 128, 0),      # single tile
    (512, 768, 256, 0),      # multi-tile, XCD remap with nwg%8 != 0
    (1000, 512, 128, 1),     # M padding path + GELU
    (2048, 1024, 1024, 0),   # encoder attn-out shape (scaled M)
    (2048, 4096, 1024, 1),   # encoder FFN-up shape (scaled M)
    (2048, 1024, 4096, 0),   # encoder FFN-down shape (scaled M)
    (2048, 3072, 1024, 0),   # encoder QKV shape (scaled M)
])
def test_gemm_nt_gpu(m, n, k, act):
    # gemm_nt is ALWAYS the hand-written MFMA kernel (dispatch policy in
    # linear_act does not apply) — this is the kernel numerics oracle.
    from nornicdb_amd.ops.gemm import gemm_nt
    torch.manual_seed(m * 31 + n + k + act)
    x = (torch.randn(m, k, device="cuda") / k ** 0.25).to(torch.bfloat16)
    w = (torch.randn(n, k, device="cuda") / k ** 0.25).to(torch.bfloat16)
    b = torch.randn(n, device="cuda").to(torch.bfloat16)
    y = gemm_nt(x, w, b, act)
    ref = F.linear(x.float(), w.float(), b.float())
    if act == 1:
        ref = F.gelu(ref)
    torch.cuda.synchronize()
    err = (y.float() - ref).abs()
    rel = err.max().item() / max(ref.abs().max().item(), 1e-6)
    assert rel < 0.02, f"max rel err {rel}"
    # transpose detector (guide G9): also check a random row/col slice
    i, j = m // 3, n // 3
    assert abs(y[i, j].float().item() - ref[i, j].item()) < 0.05 * max(1.0, abs(ref[i, j].item()))


@pytest.mark.gpu
def test_gemm_nt_no_bias_gpu():
    from nornicdb_amd.ops.gemm import gemm_nt
    torch.manual_seed(7)
    x = (torch.randn(256, 128, device="cuda") / 3).to(torch.bfloat16)
    w = (torch.randn(256, 128, device="cuda") / 3).to(torch.bfloat16)
    y = gemm_nt(x, w, None, 0)
    ref = x.float() @ w.float().T
    torch.cuda.synchronize()
    assert (y.float() - ref).abs().max().item() < 0.1


@pytest.mark.gpu
def test_linear_act_autograd_gpu(monkeypatch):
    from nornicdb_amd.ops import gemm as G
    monkeypatch.setattr(G, "_FORCE_LIB", False)
    monkeypatch.setattr(G, "_FORCE_HAND", True)
    ACT_GELU, linear_act = G.ACT_GELU, G.linear_act
    torch.manual_seed(11)
    x = (torch.randn(256, 128, device="cuda") / 3).to(torch.bfloat16).requires_grad_()
    w = (torch.randn(256, 128, device="cuda") / 3).to(torch.bfloat16).requires_grad_()
    b = torch.zeros(256, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y = linear_act(x, w, b, ACT_GELU)
    y.sum().backward()
    xr = x.detach().clone().float().requires_grad_()
    wr = w.detach().clone().float().requires_grad_()
    br = b.detach().clone().float().requires_grad_()
    F.gelu(F.linear(xr, wr, br)).sum().backward()
    for g, gr in ((x.grad, xr.grad), (w.grad, wr.grad), (b.grad, br.grad)):
        d = (g.float() - gr).abs().max().item()
        scale = max(gr.abs().max().item(), 1e-3)
        assert d / scale < 0.05


@pytest.mark.gpu
@pytest.mark.parametrize("m,n,k,act", [
    (960, 512, 512, 0),      # variant B path: M % 96 == 0
    (1920, 1024, 1024, 1),   # variant B + GELU
    (96, 256, 128, 0),       # single variant-B tile
])
def test_gemm_nt96_variant_gpu(m, n, k, act):
    """The 96x256 3-WG/CU kernel (selected when M % 96 == 0) — the
    earlier suite's shapes all silently fell back to variant A."""
    from nornicdb_amd.ops.gemm import gemm_nt
    torch.manual_seed(m + n + k)
    x = (torch.randn(m, k, device="cuda") / k ** 0.25).to(torch.bfloat16)
    w = (torch.randn(n, k, device="cuda") / k ** 0.25).to(torch.bfloat16)
    b = torch.randn(n, device="cuda").to(torch.bfloat16)
    y = gemm_nt(x, w, b, act)
    ref = F.linear(x.float(), w.float(), b.float())
    if act == 1:
        ref = F.gelu(ref)
    torch.cuda.synchronize()
    rel = (y.float() - ref).abs().max().item() / max(ref.abs().max().item(), 1e-6)
    assert rel < 0.02, rel
    i, j = m // 2, n // 3
    assert abs(y[i, j].float().item() - ref[i, j].item()) \
        < 0.05 * max(1.0, abs(ref[i, j].item()))
