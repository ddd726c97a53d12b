"""Ablation probe: where do the GEMM kernel's cycles go?
abl bits: 1=no staging, 2=no ds_reads, 4=no MFMA, 8=no C store."""
import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from nornicdb_amd.ops import require_native
nat = require_native()
M, N, K = 65536, 3072, 1024
x = (torch.randn(M, K, device="cuda") / 5).to(torch.bfloat16)
w = (torch.randn(N, K, device="cuda") / 5).to(torch.bfloat16)
def bench(abl, iters=30):
    act = -(1 + abl) if abl >= 0 else 0
    for _ in range(5): nat.gemm_nt(x, w, None, act)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): nat.gemm_nt(x, w, None, act)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3
full = bench(-1)
flops = 2.0 * M * N * K
print(f"FULL                 {full:7.3f} ms  {flops/full/1e9:7.1f} TF")
for abl, name in [(0, "abl0 (=full,nobias)"), (8, "no C store"),
                  (1, "no staging"), (2, "no ds_reads"),
                  (3, "no stage+reads"), (4, "no MFMA"),
                  (6, "no reads+MFMA"), (12, "no MFMA+store")]:
    t = bench(abl)
    print(f"{name:20s} {t:7.3f} ms  delta vs abl0: {t - full:+7.3f}")
