import sys, os, time
sys.path.insert(0, '/root/repo')
import grapehip
eng = grapehip.Engine(rank=0, world=1, master_port=29917, gpu=True)
g = eng.load_synthetic(num_vertices=404817003, num_edges=1010447118, seed=42, weighted=True)
# warmup
eng.bfs(g, 0); eng.sssp(g, 0)
os.environ['GRAPEHIP_DEBUG'] = '1'
r = eng.bfs(g, 0); print("bfs ms", r["seconds"]*1e3, flush=True)
del os.environ['GRAPEHIP_DEBUG']
for delta in [50, 100, 200, 320, 640, 1280, 2560, 1e9]:
    r = eng.sssp(g, 0, delta)
    print(f"sssp delta={delta}: {r['seconds']*1e3:.1f} ms rounds={r['rounds']}", flush=True)
os.environ['GRAPEHIP_DEBUG'] = '1'
r = eng.sssp(g, 0, 640)
print("sssp(640) ms", r["seconds"]*1e3, flush=True)
