"""Quantized kNN kernels (int8 per-row-scale, fp8 e4m3) vs bf16: timing + recall."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from nornicdb_amd.ops import require_native
from nornicdb_amd.ops.knn import (knn_search, knn_search_exact,
                                  knn_search_int8, quantize_fp8,
                                  quantize_int8)

nat = require_native()
n, d = 8_000_000, 1024
db = torch.empty(n, d, device="cuda", dtype=torch.bfloat16)
nat.fill_random_unit_(db)
q = db[:256].clone()
db8 = quantize_fp8(db)
dbi, sa = quantize_int8(db)

def search_i8(corpus, qq, k):
    return knn_search_int8(dbi, sa, qq.to(torch.float32), k)

for name, corpus in (("bf16", db), ("fp8 ", db8), ("int8", None)):
    fn = (lambda: search_i8(None, q, 10)) if corpus is None else \
         (lambda: knn_search(corpus, q, 10))
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 10
    for _ in range(iters):
        s, i = fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    bpe = 2 if name == "bf16" else 1
    tb = n * d * bpe / dt / 1e12
    print(f"{name} 8Mx1024 Q=256: {dt*1e3:6.2f} ms  ({tb:.2f} TB/s corpus read)")

# recall vs the bf16 kernel results (and self-match)
sb, ib = knn_search(db, q, 10)
for name, res in (("fp8 ", knn_search(db8, q, 10)),
                  ("int8", search_i8(None, q, 10))):
    s8, i8 = res
    hit = sum(len(set(i8[r].tolist()) & set(ib[r].tolist())) for r in range(256))
    self1 = (i8[:, 0] == torch.arange(256, device='cuda')).float().mean()
    print(f"{name} recall@10 vs bf16: {hit / 2560:.4f}  self-match: {float(self1):.4f}")
