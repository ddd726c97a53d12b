"""Debug fused encoder kernels: localize numerics errors + profile encoder."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch
import torch.nn.functional as F

from nornicdb_amd import ops
from nornicdb_amd.ops.encoder import add_layernorm, bias_gelu, mean_pool_l2norm
from nornicdb_amd.ops.attention import flash_attention_nc

torch.manual_seed(0)
dev = "cuda"

# --- add_layernorm ---
a = torch.randn(333, 1024, device=dev).to(torch.bfloat16)
b = torch.randn(333, 1024, device=dev).to(torch.bfloat16)
g = torch.randn(1024, device=dev)
be = torch.randn(1024, device=dev)
y = add_layernorm(a, b, g, be)
ref = F.layer_norm((a.float() + b.float()), (1024,), g, be)
d = (y.float() - ref).abs()
print(f"add_ln: max {d.max().item():.4f} mean {d.mean().item():.5f}")
bad = (d > 0.05)
print(f"  bad frac {bad.float().mean().item():.4f}; bad rows: {bad.any(-1).sum().item()}/333")
if bad.any():
    r = int(bad.any(-1).float().argmax())
    print(f"  first bad row {r}: cols {bad[r].nonzero().flatten()[:10].tolist()}")
    print("  y   ", y[r, :8].float().tolist())
    print("  ref ", ref[r, :8].tolist())

# --- bias_gelu ---
x = torch.randn(1000, 4096, device=dev).to(torch.bfloat16)
bb = torch.randn(4096, device=dev)
y2 = bias_gelu(x, bb)
ref2 = F.gelu(x.float() + bb.float())
print(f"bias_gelu: max {(y2.float()-ref2).abs().max().item():.4f}")

# --- pool ---
x3 = torch.randn(8, 128, 1024, device=dev).to(torch.bfloat16)
mask = torch.ones(8, 128, device=dev, dtype=torch.long); mask[:, 100:] = 0
y3 = mean_pool_l2norm(x3, mask)
m = mask[..., None].float()
ref3 = (x3.float() * m).sum(1) / m.sum(1)
ref3 = ref3 / torch.linalg.vector_norm(ref3, dim=-1, keepdim=True)
print(f"pool: max {(y3-ref3).abs().max().item():.5f}")

# --- flash attn ---
for (B, H, S) in [(2, 4, 128), (1, 16, 256)]:
    q = torch.randn(B, H, S, 64, device=dev).to(torch.bfloat16)
    k = torch.randn(B, H, S, 64, device=dev).to(torch.bfloat16)
    v = torch.randn(B, H, S, 64, device=dev).to(torch.bfloat16)
    with torch.no_grad():
        y4 = flash_attention_nc(q, k, v)
    ref4 = F.scaled_dot_product_attention(q.float(), k.float(), v.float())
    print(f"flash {B}x{H}x{S}: max {(y4.float()-ref4).abs().max().item():.4f}")

# --- encoder timing breakdown ---
from nornicdb_amd.models import BgeM3Config, BgeM3Encoder
mdl = BgeM3Encoder(BgeM3Config()).init_small().to(dev, torch.bfloat16).eval()
tok = torch.randint(0, 250002, (256, 256), device=dev)
with torch.no_grad():
    for _ in range(3):
        mdl(tok)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(5):
        mdl(tok)
    torch.cuda.synchronize()
    print(f"encoder fused: {(time.time()-t0)/5*1000:.1f} ms/batch(256x256)")
print("OK")
