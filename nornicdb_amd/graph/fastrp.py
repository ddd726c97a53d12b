"""FastRP graph embeddings (GDS compatibility).

Parity: reference pkg/cypher/fastrp.go:84-459 (graph projection + FastRP
node embeddings exposed through gds.fastRP.* procedures).

FastRP: sparse random projection R [n, d], then iterated neighbor
averaging with per-iteration weights; output = sum_i w_i * normalize(A^i R).
Runs as dense torch ops (GPU when available) over the CSR adjacency.
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import numpy as np
import torch

from .csr import CSRGraph


def fastrp_embeddings(g: CSRGraph, dims: int = 128,
                      iteration_weights: Sequence[float] = (0.0, 1.0, 1.0),
                      normalization_strength: float = 0.0,
                      seed: int = 42, device: str = None) -> np.ndarray:
    """Returns [n, dims] float32 embeddings."""
    n = g.n
    if n == 0:
        return np.zeros((0, dims), np.float32)
    device = device or ("cuda" if torch.cuda.is_available() and g.m > 50_000
                        else "cpu")
    gen = torch.Generator().manual_seed(seed)

    # sparse random projection: entries {+s, 0, -s} with p {1/2s, 1-1/s, 1/2s}
    s = 3.0
    probs = torch.rand((n, dims), generator=gen)
    R = torch.zeros(n, dims)
    R[probs < 1 / (2 * s)] = np.sqrt(s)
    R[probs > 1 - 1 / (2 * s)] = -np.sqrt(s)
    R = R.to(device)

    # degree normalization weights: deg^normalization_strength
    deg = torch.as_tensor(g.out_degrees(), dtype=torch.float32,
                          device=device).clamp_min(1)
    dnorm = deg ** normalization_strength

    rows = torch.repeat_interleave(
        torch.arange(n, device=device),
        torch.as_tensor(np.diff(g.row_ptr), device=device))
    cols = torch.as_tensor(g.col_idx, dtype=torch.long, device=device)

    def propagate(x):
        out = torch.zeros_like(x)
        out.index_add_(0, rows, x[cols])
        return out / deg[:, None]

    def l2n(x):
        return x / torch.linalg.vector_norm(x, dim=1, keepdim=True).clamp_min(1e-12)

    emb = torch.zeros(n, dims, device=device)
    cur = R * dnorm[:, None]
    for w in iteration_weights:
        if w != 0.0:
            emb = emb + w * l2n(cur)
        cur = propagate(cur)
    emb = l2n(emb)
    return emb.cpu().numpy()
