#!/usr/bin/env python3
"""LDBC Graphalytics-style suite driver (reference: ldbc_driver/ Java
harness + run_ldbc.sh). Runs the six LDBC algorithms on a named dataset
shape, reports per-algorithm "run algorithm" makespan + TEPS, and writes a
Graphalytics-like JSON results file.

Datasets are synthetic stand-ins (no network): RMAT graphs with the |V|/|E|
of the named LDBC datagen graphs, random [1,100) weights.

  python tools/ldbc_bench.py --dataset datagen-9_0-fb --gpus 1
  python -m torch.distributed.run --nproc-per-node 8 --master-addr \
      127.0.0.1 tools/ldbc_bench.py --dataset datagen-9_0-fb --gpus 8
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# LDBC Graphalytics dataset shapes (|V|, |E|) — datagen sizes from the
# Graphalytics dataset table; graph500 scale-26 standard parameters.
DATASETS = {
    "datagen-7_5-fb": (633_432, 34_185_747),
    "datagen-7_6-fb": (754_147, 42_162_988),
    "datagen-8_4-fb": (3_809_084, 269_479_177),
    "datagen-9_0-fb": (404_817_003, 1_010_447_118),
    "graph500-26": (67_108_864, 1_073_741_824),
    "com-friendster-shaped": (65_608_366, 1_806_067_135),
    "graph500-27": (134_217_728, 2_147_483_648),
    "p2p-31-shaped": (6_300, 148_000),
    # reference GPU-table graphs (Performance.md:80-97, 8x V100)
    "soc-LiveJournal1-shaped": (4_847_571, 68_993_773),
    "soc-orkut-shaped": (2_997_166, 212_698_418),
    "soc-twitter-2010-shaped": (21_297_772, 530_051_090),
}

ALGOS = ("bfs", "sssp", "pagerank", "wcc", "cdlp", "lcc")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dataset", default="datagen-9_0-fb",
                    choices=sorted(DATASETS))
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--algorithms", default=",".join(ALGOS))
    ap.add_argument("--source", type=int, default=0)
    ap.add_argument("--pr-iters", type=int, default=10)
    ap.add_argument("--cdlp-iters", type=int, default=10)
    ap.add_argument("--seed", type=int, default=42)
    ap.add_argument("--out", default="")
    args = ap.parse_args()

    import grapehip

    rank = int(os.environ.get("RANK", "0"))
    eng = (grapehip.engine_from_env(gpu=True)
           if "RANK" in os.environ else
           grapehip.Engine(rank=0, world=1, master_port=29527, gpu=True))

    nv, ne = DATASETS[args.dataset]
    t0 = time.time()
    g = eng.load_synthetic(num_vertices=nv, num_edges=ne, seed=args.seed,
                           directed=False, weighted=True)
    load_s = time.time() - t0

    results = {"dataset": args.dataset, "nv": nv, "ne": ne,
               "data": "synthetic RMAT stand-in", "n_gpus": eng.world,
               "load_seconds": load_s, "algorithms": {}}
    for algo in args.algorithms.split(","):
        t_algo0 = time.time()
        if algo == "bfs":
            r = eng.bfs(g, args.source, values=False)
            edges = ne
        elif algo == "sssp":
            r = eng.sssp(g, args.source, values=False)
            edges = ne
        elif algo == "pagerank":
            r = eng.pagerank(g, 0.85, args.pr_iters, values=False)
            edges = ne * args.pr_iters
        elif algo == "wcc":
            r = eng.wcc(g, values=False)
            edges = ne
        elif algo == "cdlp":
            r = eng.cdlp(g, args.cdlp_iters, values=False)
            edges = ne * args.cdlp_iters
        elif algo == "lcc":
            r = eng.lcc(g, values=False)
            edges = ne
        else:
            raise SystemExit("unknown algorithm " + algo)
        # Graphalytics splits makespan (wall around the job incl. driver
        # overheads) from processing time (the algorithm phase proper)
        makespan = time.time() - t_algo0
        results["algorithms"][algo] = {
            "makespan_s": makespan,
            "processing_s": r["seconds"],
            "teps": edges / r["seconds"] if r["seconds"] > 0 else None,
            "rounds": r["rounds"],
            "bytes_p2p": int(r.get("bytes_p2p", 0)),
            "bytes_coll": int(r.get("bytes_coll", 0)),
        }
        if rank == 0:
            print("%-10s proc %10.2f ms  makespan %10.2f ms  %12.3g TEPS" %
                  (algo, r["seconds"] * 1e3, makespan * 1e3,
                   edges / max(r["seconds"], 1e-12)))

    if rank == 0:
        print(json.dumps(results))
        if args.out:
            with open(args.out, "w") as f:
                json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
