"""kNN kernel BM-tile probe: timed fused search at the bench operating
point shape (chunked 4M corpus to keep it quick; grid identical)."""
import os, sys, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from nornicdb_amd.ops import require_native
from nornicdb_amd.ops.knn import knn_search, _knn_bm
nat = require_native()
print("compiled BM =", _knn_bm())
n, d = 8_000_000, 1024
db = torch.empty(n, d, device="cuda", dtype=torch.bfloat16)
nat.fill_random_unit_(db)
q = db[:256].clone()
# warm
for _ in range(3):
    knn_search(db, q, 10)
torch.cuda.synchronize()
t0 = time.perf_counter()
iters = 10
for _ in range(iters):
    s, i = knn_search(db, q, 10)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / iters
tb = n * d * 2 / dt / 1e12
tf = 2.0 * n * d * 256 / dt / 1e12
print(f"fused knn 8Mx1024 Q=256: {dt*1e3:.2f} ms  ({tb:.2f} TB/s shard read, {tf:.0f} TF)")
# sanity: top1 self-match
assert (i[:, 0] == torch.arange(256, device="cuda")).float().mean() > 0.99
print("self-match OK")
