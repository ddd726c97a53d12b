"""A/B: bge-m3 embed throughput with and without PyTorch TunableOp
(ROCm hipBLASLt GEMM autotuning)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from nornicdb_amd.models.bge_m3 import BgeM3Config, BgeM3Encoder


def bench(label):
    torch.manual_seed(0)
    m = BgeM3Encoder(BgeM3Config()).init_small().to("cuda", torch.bfloat16).eval()
    ids = torch.randint(0, 250000, (256, 256), device="cuda")
    mask = torch.ones_like(ids)
    with torch.no_grad():
        for _ in range(3):
            m(ids, mask)
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(10):
            m(ids, mask)
        torch.cuda.synchronize()
    dt = (time.time() - t0) / 10
    print(f"{label}: {dt*1e3:.1f} ms/batch  {256/dt:.0f} docs/s")
    return dt


if __name__ == "__main__":
    bench(sys.argv[1] if len(sys.argv) > 1 else "embed")
