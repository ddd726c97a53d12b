import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from nornicdb_amd import ops
n = 1 << 20
db = torch.empty(n, 1024, device="cuda", dtype=torch.bfloat16)
ops.fill_random_unit_(db)
q = db[:1].clone()
for cap in ("1024", "512", "256", "128", "64", "2048"):
    os.environ["NORNICDB_GEMV_BLOCKS"] = cap
    for _ in range(5): ops.knn_search(db, q, 10)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(100): ops.knn_search(db, q, 10)
    torch.cuda.synchronize()
    print(f"blocks={cap}: {(time.time()-t0)/100*1e3:.3f} ms", flush=True)
