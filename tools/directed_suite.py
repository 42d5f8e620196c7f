import sys, os, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import grapehip
eng = grapehip.Engine(rank=0, world=1, master_port=29923, gpu=True)
g = eng.load_synthetic(num_vertices=404817003, num_edges=1010447118, seed=42,
                       weighted=True, directed=True, build_in_csr=True)
out = {"directed": True, "nv": 404817003}
for app, fn in (("bfs", lambda: eng.bfs(g, 0, values=False)),
                ("sssp", lambda: eng.sssp(g, 0, values=False)),
                ("pagerank", lambda: eng.pagerank(g, 0.85, 10, values=False)),
                ("wcc", lambda: eng.wcc(g, values=False)),
                ("cdlp", lambda: eng.cdlp(g, 10, values=False)),
                ("lcc", lambda: eng.lcc(g, values=False))):
    r = fn()
    out[app + "_ms"] = round(r["seconds"] * 1e3, 2)
print(json.dumps(out))
