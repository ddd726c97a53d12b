"""Serving soak (GPU): boot the real server with the bge-m3 GPU
embedder, drive mixed load (store + embed + search + cypher + GraphQL)
for --seconds, and check the process stays healthy with bounded VRAM.

Usage: python scripts/soak_serve.py [--seconds 90] [--threads 4]
"""
import argparse
import json
import os
import re
import socket
import subprocess
import sys
import tempfile
import threading
import time
import urllib.request

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def vram_mb():
    try:
        out = subprocess.run(["rocm-smi", "--showmeminfo", "vram"],
                             capture_output=True, text=True, timeout=20).stdout
        m = re.findall(r"Used Memory.*?(\d+)", out)
        return int(m[0]) // (1 << 20) if m else None
    except Exception:
        return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=int, default=90)
    ap.add_argument("--threads", type=int, default=4)
    args = ap.parse_args()

    http_port, bolt_port = free_port(), free_port()
    data = tempfile.mkdtemp(prefix="soak_")
    env = dict(os.environ, PYTHONPATH=REPO, NORNICDB_EMBEDDER="bge-m3",
               NORNICDB_EMBEDDING_DIMS="1024",
               NORNICDB_DECAY_INTERVAL_S="20")
    proc = subprocess.Popen(
        [sys.executable, "-m", "nornicdb_amd", "serve", "--data-dir", data,
         "--bolt-port", str(bolt_port), "--http-port", str(http_port)],
        cwd=REPO, env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    base = f"http://127.0.0.1:{http_port}"
    t0 = time.time()
    while time.time() - t0 < 180:
        try:
            with urllib.request.urlopen(base + "/health", timeout=2) as r:
                if r.status == 200:
                    break
        except Exception:
            time.sleep(0.5)
    else:
        print("FAIL: server did not come up")
        print(proc.stdout.read(4000))
        sys.exit(1)
    print(f"server up in {time.time()-t0:.1f}s")

    def post(path, payload, timeout=60):
        req = urllib.request.Request(
            base + path, data=json.dumps(payload).encode(),
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=timeout) as r:
            return json.loads(r.read())

    counts = {"store": 0, "search": 0, "cypher": 0, "graphql": 0, "err": 0}
    lock = threading.Lock()
    stop = threading.Event()

    def worker(wid):
        i = 0
        while not stop.is_set():
            i += 1
            try:
                op = i % 4
                if op == 0:
                    post("/nornicdb/store",
                         {"content": f"memory {wid}-{i} about topic {i % 17}",
                          "title": f"t{wid}-{i}"})
                    k = "store"
                elif op == 1:
                    post("/nornicdb/search",
                         {"query": f"topic {i % 17}", "limit": 5})
                    k = "search"
                elif op == 2:
                    post("/db/neo4j/tx/commit", {"statements": [
                        {"statement": "MATCH (n:Memory) RETURN count(n)"}]})
                    k = "cypher"
                else:
                    post("/graphql", {"query":
                         'query { nodeCount(label: "Memory") }'})
                    k = "graphql"
                with lock:
                    counts[k] += 1
            except Exception:
                with lock:
                    counts["err"] += 1
                time.sleep(0.1)

    v0 = vram_mb()
    threads = [threading.Thread(target=worker, args=(w,), daemon=True)
               for w in range(args.threads)]
    for t in threads:
        t.start()
    t1 = time.time()
    mid_embed = 0
    while time.time() - t1 < args.seconds:
        time.sleep(5)
        try:
            post("/nornicdb/embed/trigger", {})
            mid_embed += 1
        except Exception:
            pass
        if proc.poll() is not None:
            print("FAIL: server died mid-soak")
            print(proc.stdout.read(4000))
            sys.exit(1)
    stop.set()
    for t in threads:
        t.join(timeout=10)
    v1 = vram_mb()
    dur = time.time() - t1
    total = sum(v for k, v in counts.items() if k != "err")
    print(f"soak {dur:.0f}s: {total} ops ({total/dur:.0f} ops/s) "
          f"{counts} embed_triggers={mid_embed}")
    print(f"vram {v0} -> {v1} MB")
    ok = counts["err"] <= total * 0.01 and proc.poll() is None
    if v0 and v1 and v1 - v0 > 4096:
        print("FAIL: VRAM grew >4 GB during soak")
        ok = False
    proc.terminate()
    try:
        proc.wait(timeout=20)
    except subprocess.TimeoutExpired:
        proc.kill()
    print("SOAK", "PASS" if ok else "FAIL")
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
