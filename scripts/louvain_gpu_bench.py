"""Louvain 10M-edge timing: CPU vs GPU local-moving."""
import os, sys, time
import numpy as np
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from nornicdb_amd.graph.csr import CSRGraph
from nornicdb_amd.graph.algos import louvain
rng = np.random.default_rng(2)
n, m = 1_000_000, 10_000_000
s = rng.integers(0, n, m); o = np.argsort(s, kind="stable")
s = s[o]; d = ((s + rng.integers(1, 50, m)) % n)
rp = np.searchsorted(s, np.arange(n + 1))
g = CSRGraph([str(i) for i in range(n)], rp.astype(np.int64), d.astype(np.int32))
for dev in ("cuda", "cuda", "cpu"):
    t0 = time.time()
    c = louvain(g, max_passes=5, max_levels=5, device=dev)
    print(f"louvain 10M edges device={dev}: {time.time()-t0:.1f}s, "
          f"{int(c.max())+1} communities")
