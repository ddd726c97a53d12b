"""Heimdall decode-path breakdown + multi-token kernel probe (GPU).

Measures, on the full Qwen2-0.5B-shape config with random weights:
  a) kernel-only cost of the per-layer cooperative step (decode_step)
  b) the round-1/2 greedy loop (step_logits + host argmax + .item())
  c) the new in-kernel greedy loop (decode_tokens) across grid sizes
and checks (c) emits exactly the tokens (b) would.

Usage: python scripts/decode_probe.py [--toks 64]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--toks", type=int, default=64)
    args = ap.parse_args()

    from nornicdb_amd.models.heimdall import (FusedDecoder, HeimdallConfig,
                                              HeimdallModel)

    torch.manual_seed(0)
    cfg = HeimdallConfig()
    m = HeimdallModel(cfg).init_small().to("cuda", torch.bfloat16).eval()
    fd = FusedDecoder(m, max_len=2048)
    nat = fd.nat

    prompt = torch.randint(0, cfg.vocab_size, (1, 8), device="cuda")
    s = prompt.shape[1]
    caches = [(None, None)] * len(m.layers)
    logits, caches = m.forward(prompt, kv_caches=caches, pos0=0)
    for li, (k, v) in enumerate(caches):
        fd.cache_k[li][:, :s] = k[0]
        fd.cache_v[li][:, :s] = v[0]
    first = int(logits[0, -1].float().argmax().item())

    T = args.toks
    c = cfg

    def sync():
        torch.cuda.synchronize()

    # ---- a) kernel-only (decode_step back to back, no host math) ----
    for _ in range(4):
        nat.decode_step(fd.layer_ptrs, fd.x, fd.qs, fd.attn, fd.hbuf,
                        fd.rope_cos, fd.rope_sin, c.num_layers,
                        c.hidden_size, c.num_heads, c.num_kv_heads, fd.hd,
                        c.intermediate_size, fd.max_len, c.rms_eps, s)
    sync()
    t0 = time.perf_counter()
    for i in range(T):
        nat.decode_step(fd.layer_ptrs, fd.x, fd.qs, fd.attn, fd.hbuf,
                        fd.rope_cos, fd.rope_sin, c.num_layers,
                        c.hidden_size, c.num_heads, c.num_kv_heads, fd.hd,
                        c.intermediate_size, fd.max_len, c.rms_eps, s + i)
    sync()
    a_ms = (time.perf_counter() - t0) / T * 1e3
    print(f"a) decode_step kernel only       : {a_ms:8.3f} ms/token")

    # ---- b) old greedy loop (step_logits + argmax + item) ----
    def old_greedy(n):
        toks = []
        cur = torch.tensor([first], device="cuda")
        pos = s
        for _ in range(n):
            lg = fd.step_logits(cur, pos)
            pos += 1
            cur = lg.argmax(-1, keepdim=True).view(-1)
            toks.append(int(cur.item()))
        return toks

    old_greedy(4)
    sync()
    t0 = time.perf_counter()
    ref_toks = old_greedy(T)
    sync()
    b_ms = (time.perf_counter() - t0) / T * 1e3
    print(f"b) host-loop greedy (r1 path)    : {b_ms:8.3f} ms/token "
          f"({1e3 / b_ms:6.1f} tok/s)")

    # ---- c) in-kernel greedy across grid sizes ----
    best = None
    for grid in (64, 96, 128, 192, 256):
        os.environ["NORNICDB_DECODE_GRID"] = str(grid)
        fd.decode_greedy(first, s, 4)
        sync()
        t0 = time.perf_counter()
        toks = fd.decode_greedy(first, s, T)
        sync()
        c_ms = (time.perf_counter() - t0) / T * 1e3
        match = toks == ref_toks
        print(f"c) decode_tokens grid={grid:<4d}      : {c_ms:8.3f} ms/token "
              f"({1e3 / c_ms:6.1f} tok/s)  match={match}")
        if match and (best is None or c_ms < best[1]):
            best = (grid, c_ms)
    os.environ.pop("NORNICDB_DECODE_GRID", None)

    if best is None:
        print("FAIL: no grid produced matching tokens")
        sys.exit(1)
    print(f"\nbest grid {best[0]}: {best[1]:.3f} ms/token = "
          f"{1e3 / best[1]:.1f} tok/s (was {1e3 / b_ms:.1f})")

    # ---- end-to-end generate() parity (greedy, with prefill) ----
    g1 = fd.generate(prompt.clone(), max_new_tokens=12, temperature=0.0)
    from nornicdb_amd.models.heimdall import GraphedDecoder
    gd = GraphedDecoder(m, max_len=2048).capture()
    g2 = gd.generate(prompt.clone(), max_new_tokens=12, temperature=0.0)
    print("generate parity (fused vs graphed):", g1 == g2, g1[:6], g2[:6])


if __name__ == "__main__":
    main()
