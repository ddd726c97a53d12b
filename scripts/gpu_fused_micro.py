"""Isolate fused decode kernel cost: full vs 1-layer vs kernel-only loop."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from nornicdb_amd import ops
from nornicdb_amd.models.heimdall import (FusedDecoder, HeimdallConfig,
                                          HeimdallModel)

nat = ops.require_native()
torch.manual_seed(0)
cfg = HeimdallConfig()
m = HeimdallModel(cfg).init_small().to("cuda", torch.bfloat16).eval()
fd = FusedDecoder(m, max_len=2048)


def t(nl, iters=50):
    for _ in range(5):
        nat.decode_step(fd.layer_ptrs, fd.x, fd.qs, fd.attn, fd.hbuf,
                        fd.rope_cos, fd.rope_sin, nl, cfg.hidden_size,
                        cfg.num_heads, cfg.num_kv_heads, 64,
                        cfg.intermediate_size, fd.max_len, cfg.rms_eps, 100)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        nat.decode_step(fd.layer_ptrs, fd.x, fd.qs, fd.attn, fd.hbuf,
                        fd.rope_cos, fd.rope_sin, nl, cfg.hidden_size,
                        cfg.num_heads, cfg.num_kv_heads, 64,
                        cfg.intermediate_size, fd.max_len, cfg.rms_eps, 100)
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


t24 = t(24)
t1 = t(1)
t0_ = t(0)
print(f"24 layers: {t24*1e3:.3f} ms  1 layer: {t1*1e3:.3f} ms  "
      f"0 layers: {t0_*1e3:.3f} ms")
print(f"per-layer: {(t24-t0_)/24*1e6:.0f} us; sync share estimate: "
      f"launch {t0_*1e6:.0f} us")
