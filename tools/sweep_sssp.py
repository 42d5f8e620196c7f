"""Per-graph SSSP delta sweep + per-round trace (GPU box tool)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import grapehip

SHAPES = {
    "soc-LiveJournal1": (4_847_571, 68_993_773),
    "soc-orkut": (2_997_166, 212_698_418),
    "datagen-9_0-fb": (404_817_003, 1_010_447_118),
}
eng = grapehip.Engine(rank=0, world=1, master_port=29917, gpu=True)
for name, (nv, ne) in SHAPES.items():
    g = eng.load_synthetic(num_vertices=nv, num_edges=ne, seed=42,
                           weighted=True)
    eng.sssp(g, 0, values=False)  # warmup
    best = None
    for delta in [0, 100, 200, 400, 800, 1600, 3200, 6400]:
        use = -1.0 if delta == 0 else float(delta)
        t = min(eng.sssp(g, 0, use, values=False)["seconds"]
                for _ in range(2))
        r = eng.sssp(g, 0, use, values=False)
        print(f"{name} delta={'auto' if delta==0 else delta}: "
              f"{t*1e3:.2f} ms rounds={r['rounds']}", flush=True)
        if best is None or t < best[1]:
            best = (delta, t)
    print(f"{name} BEST delta={best[0]} {best[1]*1e3:.2f} ms", flush=True)
    del g
