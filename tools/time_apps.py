"""Time individual apps on a synthetic RMAT graph (GPU box tool)."""
import argparse, json, sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import grapehip

ap = argparse.ArgumentParser()
ap.add_argument("--nv", type=int, default=40_000_000)
ap.add_argument("--ne", type=int, default=101_000_000)
ap.add_argument("--apps", default="bfs,sssp,pagerank,wcc,cdlp,lcc")
ap.add_argument("--cdlp-iters", type=int, default=10)
ap.add_argument("--warmup", type=int, default=1,
                help="untimed runs per app first (absorbs one-time costs "
                     "like hipGraph capture, matching bench.py)")
args = ap.parse_args()

eng = grapehip.Engine(rank=0, world=1, master_port=29917, gpu=True)
g = eng.load_synthetic(num_vertices=args.nv, num_edges=args.ne, seed=42,
                       weighted=True)
def run_app(app):
    if app == "bfs":
        return eng.bfs(g, 0, values=False)
    if app == "sssp":
        return eng.sssp(g, 0, values=False)
    if app == "pagerank":
        return eng.pagerank(g, 0.85, 10, values=False)
    if app == "wcc":
        return eng.wcc(g, values=False)
    if app == "cdlp":
        return eng.cdlp(g, args.cdlp_iters, values=False)
    if app == "lcc":
        return eng.lcc(g, values=False)
    raise SystemExit("unknown app " + app)


out = {"nv": args.nv, "ne_input": args.ne, "ne_stored": g.num_edges}
for _ in range(args.warmup):
    for app in args.apps.split(","):
        run_app(app)
for app in args.apps.split(","):
    if app == "bfs":
        r = eng.bfs(g, 0, values=False)
    elif app == "sssp":
        r = eng.sssp(g, 0, values=False)
    elif app == "pagerank":
        r = eng.pagerank(g, 0.85, 10, values=False)
    elif app == "wcc":
        r = eng.wcc(g, values=False)
    elif app == "cdlp":
        r = eng.cdlp(g, args.cdlp_iters, values=False)
    elif app == "lcc":
        r = eng.lcc(g, values=False)
    out[app + "_ms"] = round(r["seconds"] * 1e3, 2)
    out[app + "_rounds"] = r["rounds"]
print(json.dumps(out))
