"""Quick GPU micro-benchmarks for tier-1 kernels. Prints GB/s and ms."""
import sys
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch

from nornicdb_amd import ops


def t(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


def main():
    dev = "cuda:0"
    n, d = 4_000_000, 1024
    x = torch.empty(n, d, device=dev, dtype=torch.bfloat16)

    dt = t(lambda: ops.fill_random_unit_(x))
    gb = n * d * 2 / 1e9
    print(f"fill_random_unit: {dt*1e3:.2f} ms  {gb/dt:.0f} GB/s (write-only {gb:.1f} GB)")

    dt = t(lambda: ops.l2_normalize_(x))
    print(f"l2_normalize bf16: {dt*1e3:.2f} ms  {3*gb/dt:.0f} GB/s effective (r+r+w)")

    q8 = x[:8].clone()
    dt = t(lambda: ops.knn_search(x, q8, 10))
    print(f"knn_gemv Q=8 k=10 over {n}x{d}: {dt*1e3:.2f} ms  {gb/dt:.0f} GB/s  {8/dt:.0f} qps")

    q1 = x[:1].clone()
    dt = t(lambda: ops.knn_search(x, q1, 10))
    print(f"knn_gemv Q=1: {dt*1e3:.2f} ms  {gb/dt:.0f} GB/s  {1/dt:.0f} qps")

    q16 = x[:16].clone()
    dt = t(lambda: ops.knn_search(x, q16, 10))
    print(f"knn_gemv Q=16: {dt*1e3:.2f} ms  {gb/dt:.0f} GB/s  {16/dt:.0f} qps")

    q256 = x[:256].clone()
    dt = t(lambda: ops.knn_search(x, q256, 10))
    print(f"knn fused-MFMA Q=256: {dt*1e3:.2f} ms  {gb/dt:.0f} GB/s(db)  {256/dt:.0f} qps")

    gb4 = n * d * 2 / 1e9
    print(f"  (mfma fused path active for Q=256: reads {gb4:.1f} GB once)")

    # raw matmul reference for the same shape
    w = x[:1 << 20]
    qq = q256.clone()
    dt = t(lambda: (qq @ w.T).float())
    fl = 2 * (1 << 20) * d * 256
    print(f"hipBLASLt bf16 GEMM 256x{1<<20}x{d}: {dt*1e3:.3f} ms  {fl/dt/1e12:.1f} TFLOP/s")

    print("OK")


if __name__ == "__main__":
    sys.exit(main())
