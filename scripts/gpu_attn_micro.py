"""Micro-bench + PMC target for k_flash_attn_nc at the bge-m3 shape."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from nornicdb_amd.ops import attention as ops

B, H, S, D = 256, 16, 256, 64
q = torch.randn(B, S, H * D, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q)
v = torch.randn_like(q)

qv = q.view(B, S, H, D)
kv = k.view(B, S, H, D)
vv = v.view(B, S, H, D)

for _ in range(3):
    o = ops.flash_attention_bshd(qv, kv, vv)
torch.cuda.synchronize()
t0 = time.time()
for _ in range(20):
    o = ops.flash_attention_bshd(qv, kv, vv)
torch.cuda.synchronize()
dt = (time.time() - t0) / 20
fl = 4.0 * B * H * S * S * D  # QK^T + PV
print(f"flash_attn_nc B{B} H{H} S{S} D{D}: {dt*1e3:.2f} ms  {fl/dt/1e12:.0f} TF")
