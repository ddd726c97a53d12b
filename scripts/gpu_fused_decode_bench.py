"""Decode throughput: eager vs hipGraph vs fused cooperative kernel."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from nornicdb_amd.models.heimdall import (FusedDecoder, GraphedDecoder,
                                          HeimdallConfig, HeimdallModel)


def main():
    torch.manual_seed(0)
    cfg = HeimdallConfig()  # Qwen2-0.5B shape
    m = HeimdallModel(cfg).init_small().to("cuda", torch.bfloat16).eval()
    prompt = torch.randint(0, cfg.vocab_size, (1, 16), device="cuda")
    N = 128

    fd = FusedDecoder(m, max_len=2048)
    fd.generate(prompt.clone(), max_new_tokens=8, temperature=0.0)  # warm
    torch.cuda.synchronize()
    t0 = time.time()
    out = fd.generate(prompt.clone(), max_new_tokens=N, temperature=0.0)
    torch.cuda.synchronize()
    dt = time.time() - t0
    print(f"fused:   {len(out)/dt:7.1f} tok/s  ({dt/len(out)*1e3:.2f} ms/tok)")

    gd = GraphedDecoder(m, max_len=2048).capture()
    gd.generate(prompt.clone(), max_new_tokens=8, temperature=0.0)
    torch.cuda.synchronize()
    t0 = time.time()
    out = gd.generate(prompt.clone(), max_new_tokens=N, temperature=0.0)
    torch.cuda.synchronize()
    dt = time.time() - t0
    print(f"graphed: {len(out)/dt:7.1f} tok/s  ({dt/len(out)*1e3:.2f} ms/tok)")

    t0 = time.time()
    out = m.generate(prompt.clone(), max_new_tokens=64, temperature=0.0)
    torch.cuda.synchronize()
    dt = time.time() - t0
    print(f"eager:   {len(out)/dt:7.1f} tok/s  ({dt/len(out)*1e3:.2f} ms/tok)")


if __name__ == "__main__":
    main()
