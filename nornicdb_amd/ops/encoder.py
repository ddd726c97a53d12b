"""Fused encoder ops with torch CPU/eager fallbacks (numerics oracles)."""

import torch
import torch.nn.functional as F

from . import native_or_none


def _use_native(x):
    nat = native_or_none()
    return nat if (nat is not None and x.is_cuda and x.dtype == torch.bfloat16) else None


def add_layernorm(a, b, gamma, beta, eps=1e-5):
    """LN(a + b) * gamma + beta (b optional)."""
    nat = _use_native(a)
    if nat is not None and a.shape[-1] % 8 == 0 and a.shape[-1] <= 8192:
        return nat.add_layernorm(a, b, gamma, beta, eps)
    x = a if b is None else a + b
    return F.layer_norm(x.float(), (x.shape[-1],), gamma.float(), beta.float(),
                        eps).to(a.dtype)


def bias_gelu(x, bias):
    nat = _use_native(x)
    if nat is not None and x.shape[-1] % 8 == 0:
        return nat.bias_gelu(x, bias)
    return F.gelu((x.float() + bias.float())).to(x.dtype)


def mean_pool_l2norm(x, mask=None):
    """x [B,S,D] -> [B,D] fp32 unit vectors (masked mean)."""
    nat = _use_native(x)
    if nat is not None and x.dim() == 3 and x.shape[-1] % 8 == 0:
        return nat.mean_pool_l2norm(x, mask)
    xf = x.float()
    if mask is None:
        pooled = xf.mean(dim=1)
    else:
        m = mask[..., None].float()
        pooled = (xf * m).sum(1) / m.sum(1).clamp_min(1)
    return pooled / torch.linalg.vector_norm(pooled, dim=-1, keepdim=True).clamp_min(1e-12)
