#!/usr/bin/env python3
"""Comm-volume evidence: per-rank data-plane bytes
per app across worlds. The mirror/halo/row-fetch redesign makes p2p
volume scale with the referenced boundary, not V*world — this driver
runs worlds on one box (TCP-staged data plane when ranks share the GPU)
and records each rank's bytes_p2p / bytes_coll per app.

  python tools/comm_volume.py --worlds 1,2,4 --nv 40000000 --ne 101000000
"""
import argparse
import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

WORKER = r'''
import json, os, sys
sys.path.insert(0, os.environ["GRAPEHIP_REPO"])
import grapehip
eng = grapehip.engine_from_env(gpu=True)
nv = int(os.environ["CV_NV"]); ne = int(os.environ["CV_NE"])
g = eng.load_synthetic(num_vertices=nv, num_edges=ne, seed=42, weighted=True)
out = {}
for app in ("bfs", "sssp", "pagerank", "wcc", "cdlp", "lcc"):
    r = (eng.bfs(g, 0, values=False) if app == "bfs" else
         eng.sssp(g, 0, values=False) if app == "sssp" else
         eng.pagerank(g, 0.85, 10, values=False) if app == "pagerank" else
         eng.wcc(g, values=False) if app == "wcc" else
         eng.cdlp(g, 10, values=False) if app == "cdlp" else
         eng.lcc(g, values=False))
    out[app] = {"ms": round(r["seconds"] * 1e3, 2), "rounds": r["rounds"],
                "bytes_p2p": int(r["bytes_p2p"]),
                "bytes_coll": int(r["bytes_coll"])}
path = os.path.join(os.environ["GRAPEHIP_OUT"],
                    "cv_rank%s.json" % os.environ["RANK"])
json.dump(out, open(path, "w"))
'''


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--worlds", default="1,2,4")
    ap.add_argument("--nv", type=int, default=40_000_000)
    ap.add_argument("--ne", type=int, default=101_000_000)
    ap.add_argument("--out", default="gpurun_out/comm_volume.json")
    args = ap.parse_args()
    result = {"nv": args.nv, "ne": args.ne, "worlds": {}}
    port = 29950
    for world in [int(w) for w in args.worlds.split(",")]:
        outdir = Path("gpurun_out") / ("cv_w%d" % world)
        outdir.mkdir(parents=True, exist_ok=True)
        procs = []
        for rank in range(world):
            env = dict(os.environ, GRAPEHIP_REPO=str(REPO),
                       GRAPEHIP_OUT=str(outdir), RANK=str(rank),
                       LOCAL_RANK=str(rank), WORLD_SIZE=str(world),
                       MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                       CV_NV=str(args.nv), CV_NE=str(args.ne))
            procs.append(subprocess.Popen([sys.executable, "-c", WORKER],
                                          env=env, stdout=subprocess.PIPE,
                                          stderr=subprocess.STDOUT))
        for p in procs:
            o, _ = p.communicate(timeout=1800)
            if p.returncode != 0:
                print(o.decode()[-2000:])
                sys.exit(1)
        port += 11
        ranks = [json.load(open(outdir / ("cv_rank%d.json" % r)))
                 for r in range(world)]
        result["worlds"][str(world)] = ranks
        for app in ranks[0]:
            p2p = max(r[app]["bytes_p2p"] for r in ranks)
            coll = max(r[app]["bytes_coll"] for r in ranks)
            print("w%-2d %-9s p2p/rank %10.1f MB  coll/rank %10.1f MB" %
                  (world, app, p2p / 1e6, coll / 1e6))
    json.dump(result, open(args.out, "w"), indent=1)
    print("wrote", args.out)


if __name__ == "__main__":
    main()
