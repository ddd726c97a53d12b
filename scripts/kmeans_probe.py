"""Warm k-means timing at the reference config: 100K x 1024, k=223.
Reference baselines: CUDA 80 ms, Metal 120 ms (gpu-acceleration.md:124-131).
VERDICT r1 target: < 50 ms warm with our own HIP kernels."""
import os, sys, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from nornicdb_amd.search.kmeans import kmeans, optimal_k

n, d = 100_000, 1024
k = optimal_k(n)
x = torch.randn(n, d, device="cuda").to(torch.bfloat16)
# warm (hipBLASLt autotune etc. kept out of the timed run)
kmeans(x, k, iters=25, seed=0)
torch.cuda.synchronize()
for trial in range(3):
    t0 = time.perf_counter()
    c, a = kmeans(x, k, iters=25, seed=trial)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) * 1e3
    print(f"kmeans 100Kx1024 k={k} iters<=25: {dt:.1f} ms (trial {trial})")
