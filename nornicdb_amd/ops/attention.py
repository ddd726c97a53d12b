"""Non-causal (encoder) attention.

The HIP flash-style kernel (csrc/attention.hip) will own this on GPU;
until it lands we dispatch to torch.scaled_dot_product_attention, which on
ROCm lowers to hipBLASLt GEMMs + softmax (still all-AMD-native, but not our
fused kernel). The CPU path is the numerics oracle.
"""

import torch
import torch.nn.functional as F

from . import native_or_none

_WARNED = False


def flash_attention_nc(q, k, v, attn_bias=None):
    """q,k,v: [B, H, S, D]. Returns [B, H, S, D]. Non-causal."""
    nat = native_or_none()
    if (
        q.is_cuda
        and nat is not None
        and hasattr(nat, "flash_attn_nc")
        and attn_bias is None
        and q.dtype == torch.bfloat16
        and q.shape[-1] == 64
        and q.shape[-2] % 64 == 0
        and not (q.requires_grad and torch.is_grad_enabled())
    ):
        return nat.flash_attn_nc(q.contiguous(), k.contiguous(), v.contiguous())
    return F.scaled_dot_product_attention(q, k, v, attn_mask=attn_bias)
