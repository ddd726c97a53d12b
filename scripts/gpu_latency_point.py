"""BASELINE parity point: single-query latency on 1M x 1024 (A100: 1 ms),
plus 100M single-query serving latency."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from nornicdb_amd import ops

for n in (1 << 20, 100_000_000):
    db = torch.empty(n, 1024, device="cuda", dtype=torch.bfloat16)
    ops.fill_random_unit_(db)
    q = db[:1].clone()
    for _ in range(5):
        ops.knn_search(db, q, 10)
    torch.cuda.synchronize()
    t0 = time.time()
    iters = 50 if n < 2_000_000 else 10
    for _ in range(iters):
        ops.knn_search(db, q, 10)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / iters
    print(f"N={n:>11,}  Q=1 k=10: {dt*1e3:7.3f} ms   "
          f"({n*1024*2/dt/1e12:.2f} TB/s)")
    del db
    torch.cuda.empty_cache()
