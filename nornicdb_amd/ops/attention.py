"""Non-causal (encoder) attention.

Two entry points:
- flash_attention_bshd(q, k, v): the production path. Takes [B, S, H, D]
  VIEWS straight out of the qkv projection (zero transposes; the HIP
  kernel is stride-aware) and returns contiguous [B, S, H, D].
- flash_attention_nc(q, k, v, attn_bias): [B, H, S, D] API used by tests
  and the masked/padded fallback (torch sdpa).
"""

import torch
import torch.nn.functional as F

from . import native_or_none


def _native_ok(q, need_s_mult=True):
    nat = native_or_none()
    return (
        nat
        if (
            nat is not None
            and q.is_cuda
            and q.dtype == torch.bfloat16
            and hasattr(nat, "flash_attn_nc")
            and q.shape[-1] == 64
            and not (q.requires_grad and torch.is_grad_enabled())
        )
        else None
    )


def flash_attention_bshd(q, k, v):
    """q,k,v: [B, S, H, D] views (head stride D). Returns [B, S, H, D]."""
    nat = _native_ok(q)
    if (nat is not None and q.shape[1] % 64 == 0
            and q.stride(3) == 1 and q.stride(2) == q.shape[3]):
        return nat.flash_attn_nc(q, k, v)
    # eager fallback
    qt = q.permute(0, 2, 1, 3)
    kt = k.permute(0, 2, 1, 3)
    vt = v.permute(0, 2, 1, 3)
    o = F.scaled_dot_product_attention(qt, kt, vt)
    return o.permute(0, 2, 1, 3).contiguous()


def flash_attention_nc(q, k, v, attn_bias=None):
    """q,k,v: [B, H, S, D]. Returns [B, H, S, D]. Non-causal."""
    nat = _native_ok(q)
    if nat is not None and attn_bias is None and q.shape[-2] % 64 == 0:
        qb = q.transpose(1, 2).contiguous()
        kb = k.transpose(1, 2).contiguous()
        vb = v.transpose(1, 2).contiguous()
        return nat.flash_attn_nc(qb, kb, vb).transpose(1, 2)
    return F.scaled_dot_product_attention(q, k, v, attn_mask=attn_bias)
