import sys, numpy as np
sys.path.insert(0, '/root/repo')
import grapehip
eng = grapehip.Engine(rank=0, world=1, master_port=29917, gpu=True)
for n in [1000, 2048, 2049, 100000, 4194304, 4194305, 8000000, 20000000]:
    rng = np.random.default_rng(1)
    a = rng.integers(0, 50, size=n, dtype=np.uint32)
    out = np.array(eng._debug_scan(a.tolist()), dtype=np.uint64)
    ref = np.zeros(n+1, dtype=np.uint64); ref[1:] = np.cumsum(a, dtype=np.uint64)
    ok = np.array_equal(out, ref)
    print(f"scan n={n}: {'OK' if ok else 'MISMATCH at '+str(np.argmax(out!=ref))}", flush=True)
print("now load_synthetic 10M/160M", flush=True)
g = eng.load_synthetic(num_vertices=10000000, num_edges=160000000, seed=42, weighted=True)
print("built", g.num_edges, flush=True)
r = eng.bfs(g, 0); print("bfs ok rounds", r["rounds"], flush=True)
r = eng.sssp(g, 0); print("sssp ok rounds", r["rounds"], flush=True)
r = eng.pagerank(g, 0.85, 3); print("pr ok sum", r["values"].sum(), flush=True)
r = eng.wcc(g); print("wcc ok", flush=True)
