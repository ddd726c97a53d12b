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
