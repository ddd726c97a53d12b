import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from nornicdb_amd.models.heimdall import GraphedDecoder, HeimdallConfig, HeimdallModel

cfg = HeimdallConfig()
m = HeimdallModel(cfg).init_small().to("cuda", torch.bfloat16).eval()
try:
    gd = GraphedDecoder(m, max_len=1024).capture()
    print("capture OK")
except Exception as e:
    print("capture FAILED:", repr(e)); sys.exit(1)

# pure replay rate
gd.tok.fill_(5); gd.pos.fill_(10)
for _ in range(5): gd.graph.replay()
torch.cuda.synchronize()
t0 = time.time()
for _ in range(200): gd.graph.replay()
torch.cuda.synchronize()
print(f"pure replay: {200/(time.time()-t0):.0f} tok/s")

# greedy generate (sync per token for argmax)
ids = torch.randint(0, cfg.vocab_size, (1, 16), device="cuda")
out = gd.generate(ids, max_new_tokens=64, temperature=0)
torch.cuda.synchronize()
t0 = time.time()
out = gd.generate(ids, max_new_tokens=128, temperature=0)
print(f"graphed greedy: {128/(time.time()-t0):.0f} tok/s")
t0 = time.time()
out = gd.generate(ids, max_new_tokens=128, temperature=0.8)
print(f"graphed sampled: {128/(time.time()-t0):.0f} tok/s")
# eager comparison
t0 = time.time()
m.generate(ids, max_new_tokens=64, temperature=0)
print(f"eager greedy: {64/(time.time()-t0):.0f} tok/s")

# 1M x 1024 Q=1 latency (reference parity: A100 1 ms)
from nornicdb_amd import ops
db1 = torch.empty(1_000_000, 1024, device="cuda", dtype=torch.bfloat16)
ops.fill_random_unit_(db1)
q1 = db1[:1].clone()
for _ in range(5): ops.knn_search(db1, q1, 10)
torch.cuda.synchronize()
t0 = time.time()
for _ in range(50): ops.knn_search(db1, q1, 10)
torch.cuda.synchronize()
print(f"kNN 1Mx1024 Q=1 k=10: {(time.time()-t0)/50*1000:.3f} ms/query")
q8b = db1[:8].clone()
t0 = time.time()
for _ in range(50): ops.knn_search(db1, q8b, 10)
torch.cuda.synchronize()
print(f"kNN 1Mx1024 Q=8 k=10: {(time.time()-t0)/50*1000:.3f} ms/batch")
