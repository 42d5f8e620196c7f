import sys, os, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import grapehip
eng = grapehip.Engine(rank=0, world=1, master_port=29921, gpu=True)
g = eng.load_synthetic(num_vertices=2_000_000_000, num_edges=8_000_000_000,
                       seed=42, weighted=False)
r = eng.bfs(g, 0, values=False)
print(json.dumps({"nv": 2_000_000_000, "stored": g.num_edges,
                  "bfs_ms": round(r["seconds"]*1e3, 2),
                  "rounds": r["rounds"]}))
