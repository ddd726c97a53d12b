"""Vector-math ops: L2 normalize, synthetic corpus generation.

Replaces reference pkg/math/vector + pkg/simd (CPU SIMD) and the
normalize/norm CUDA kernels (reference pkg/gpu/cuda/cuda_kernels.cu:185-225)
with a single fused wave64 HIP kernel on GPU and torch on CPU.
"""

import torch

from . import require_native


def l2_normalize_(x: torch.Tensor) -> torch.Tensor:
    """In-place row-wise L2 normalization of a 2D tensor."""
    if x.is_cuda:
        require_native().l2_normalize_(x)
        return x
    # CPU reference path
    n = torch.linalg.vector_norm(x.float(), dim=-1, keepdim=True).clamp_min(1e-6)
    x.copy_((x.float() / n).to(x.dtype))
    return x


def l2_normalize(x: torch.Tensor) -> torch.Tensor:
    return l2_normalize_(x.clone())


def fill_random_unit_(x: torch.Tensor, row_base: int = 0, seed: int = 0x6E6F726E):
    """Fill a 2D bf16 tensor with deterministic unit-norm pseudo-gaussian rows.

    Row identity = row_base + i, so GPU shards of one logical corpus are
    globally consistent. The GPU kernel writes at HBM speed; the CPU path is
    a torch.Generator-based stand-in (NOT bit-identical to the GPU kernel —
    it exists so CPU test runs have something of the right shape/norm).
    """
    if x.is_cuda:
        require_native().fill_random_unit_(x, row_base, seed)
        return x
    g = torch.Generator().manual_seed(seed ^ row_base)
    t = torch.randn(x.shape, generator=g, dtype=torch.float32)
    t /= torch.linalg.vector_norm(t, dim=-1, keepdim=True).clamp_min(1e-12)
    x.copy_(t.to(x.dtype))
    return x
