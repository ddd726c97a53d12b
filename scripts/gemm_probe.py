"""GPU probe: hand-written MFMA GEMM (csrc/gemm.hip) vs hipBLASLt on the
bge-m3 encoder shapes. Run via gpurun; writes results under gpurun_out/.

Usage: python scripts/gemm_probe.py [--iters 50]
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from nornicdb_amd.ops import require_native  # noqa: E402

# encoder shapes at bench operating point: M = 256 batch x 256 seq tokens
def shapes(m):
    return [
        ("qkv", m, 3072, 1024, 0),
        ("attn_out", m, 1024, 1024, 0),
        ("ffn_up", m, 4096, 1024, 1),
        ("ffn_down", m, 1024, 4096, 0),
    ]


def bench(fn, iters, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--m", type=int, default=65536,
                    help="65280 = divisible by both 96 and 256 tiles")
    ap.add_argument("--refcheck", action="store_true", default=True)
    args = ap.parse_args()
    nat = require_native()
    dev = "cuda"
    out = {"shapes": []}

    for name, m, n, k, act in shapes(args.m):
        torch.manual_seed(1234)
        x = (torch.randn(m, k, device=dev) / k ** 0.25).to(torch.bfloat16)
        w = (torch.randn(n, k, device=dev) / k ** 0.25).to(torch.bfloat16)
        b = (torch.randn(n, device=dev) / 8).to(torch.bfloat16)

        # refcheck on a slice (full fp32 ref at M=65536 is heavy; use 4096 rows)
        xs = x[:4096]
        y = nat.gemm_nt(xs, w, b, act)
        ref = torch.nn.functional.linear(xs.float(), w.float(), b.float())
        if act == 1:
            ref = torch.nn.functional.gelu(ref)
        err = (y.float() - ref).abs().max().item()
        scale = ref.abs().max().item()
        rel = err / max(scale, 1e-6)
        ok = rel < 0.02
        del y, ref

        flops = 2.0 * m * n * k
        t_mine = bench(lambda: nat.gemm_nt(x, w, b, act), args.iters)
        tf_mine = flops / t_mine / 1e12

        # hipBLASLt path: matmul + separate bias(+gelu) like round-1 model code
        def lib():
            y = torch.nn.functional.linear(x, w, b)
            if act == 1:
                y = torch.nn.functional.gelu(y)
            return y

        t_lib = bench(lib, args.iters)
        tf_lib = flops / t_lib / 1e12
        # pure GEMM (no epilogue) library number too
        t_libraw = bench(lambda: x @ w.T, args.iters)
        tf_libraw = flops / t_libraw / 1e12

        r = dict(shape=name, m=m, n=n, k=k, act=act, rel_err=rel, ok=ok,
                 ms_mine=t_mine * 1e3, tf_mine=tf_mine,
                 ms_lib=t_lib * 1e3, tf_lib=tf_lib,
                 ms_libraw=t_libraw * 1e3, tf_libraw=tf_libraw,
                 speedup_vs_lib=t_lib / t_mine)
        out["shapes"].append(r)
        print(json.dumps(r), flush=True)

    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/gemm_probe.json", "w") as f:
        json.dump(out, f, indent=1)
    print("TOTAL mine ms:", sum(s["ms_mine"] for s in out["shapes"]),
          " lib ms:", sum(s["ms_lib"] for s in out["shapes"]))


if __name__ == "__main__":
    main()
