"""Grid-barrier cost: cg::grid.sync vs two-level custom barrier (GPU).

Decides whether the decode kernel's sync wall at large grids is arrival
contention (custom barrier fixes it) or fence cost (it doesn't).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from nornicdb_amd.ops import require_native

nat = require_native()
scratch = torch.zeros(512, dtype=torch.int32, device="cuda")
ITERS = 2000
print(f"{'grid':>5} {'cg us/sync':>11} {'custom us/sync':>15}")
for grid in (32, 64, 96, 128, 192, 256):
    row = [grid]
    for which in (0, 1):
        scratch.zero_()
        nat.sync_bench(50, which, grid, scratch)   # warm
        scratch.zero_()
        ms = nat.sync_bench(ITERS, which, grid, scratch)
        row.append(ms * 1e3 / ITERS)
    print(f"{row[0]:>5} {row[1]:>11.2f} {row[2]:>15.2f}")
