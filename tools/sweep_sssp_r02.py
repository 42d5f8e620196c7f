import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import grapehip
eng = grapehip.Engine(rank=0, world=1, master_port=29919, gpu=True)
g = eng.load_synthetic(num_vertices=404817003, num_edges=1010447118, seed=42,
                       weighted=True)
eng.sssp(g, 0, values=False)  # warmup
for d in (200, 400, 800, 1600, 3200, 6400):
    r = eng.sssp(g, 0, delta=float(d), values=False)
    print("delta %5d: %7.2f ms  rounds %d" % (d, r["seconds"]*1e3, r["rounds"]), flush=True)
r = eng.sssp(g, 0, values=False)
print("auto      : %7.2f ms  rounds %d" % (r["seconds"]*1e3, r["rounds"]), flush=True)
