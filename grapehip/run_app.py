"""grapehip run_app — CLI driver with the reference's flag surface.

Reference parity: examples/analytical_apps/run_app.{h,cc} + flags.cc
(--application/--efile/--vfile/--directed/--out_prefix/per-app params,
--serialize/--deserialize) and timer.h's named-phase wall timers
("load graph" / "run algorithm" / "output"), which is what LDBC measures.

Single rank:   python -m grapehip.run_app --application bfs --efile g.e ...
Multi rank:    python -m torch.distributed.run --nproc-per-node N \
                   --master-addr 127.0.0.1 -m grapehip.run_app -- ...
(each rank partial-reads its slice of the .e/.v files and shuffles records
to owners during the distributed build, like EVFragmentLoader).
"""
from __future__ import annotations

import argparse
import os
import sys
import time

import numpy as np

import grapehip
from grapehip.io import read_ldbc_edges, read_ldbc_vertices

APPS = ("bfs", "sssp", "pagerank", "wcc", "cdlp", "lcc", "bc", "kcore",
        "core_decomposition", "kclique")


def build_parser():
    ap = argparse.ArgumentParser(prog="grapehip.run_app")
    ap.add_argument("--application", required=True, choices=APPS)
    ap.add_argument("--efile", default="")
    ap.add_argument("--vfile", default="")
    ap.add_argument("--out_prefix", default="")
    ap.add_argument("--directed", action="store_true")
    ap.add_argument("--weighted", action="store_true",
                    help="parse a third edge column (required by sssp)")
    ap.add_argument("--gpu", action="store_true")
    ap.add_argument("--app_concurrency", type=int, default=0)
    ap.add_argument("--idxer", default="hashmap",
                    choices=("hashmap", "mph"),
                    help="oid->lid index for --vfile graphs "
                         "(reference --idxer_type)")
    # per-app parameters (flags.cc:23-67)
    ap.add_argument("--bfs_source", type=int, default=0)
    ap.add_argument("--sssp_source", type=int, default=0)
    ap.add_argument("--bc_source", type=int, default=0)
    ap.add_argument("--pr_d", type=float, default=0.85)
    ap.add_argument("--pr_mr", type=int, default=10)
    ap.add_argument("--cdlp_mr", type=int, default=10)
    ap.add_argument("--kcore_k", type=int, default=3)
    ap.add_argument("--kclique_k", type=int, default=3)
    # delta mutation (LoadGraphAndMutate, loader.h:55-68 +
    # ev_fragment_mutator.h: base graph + additions/removals files)
    ap.add_argument("--efile_add", default="")
    ap.add_argument("--efile_remove", default="")
    # checkpoint (ev_fragment_loader.h:75-93)
    ap.add_argument("--serialize", action="store_true")
    ap.add_argument("--deserialize", action="store_true")
    ap.add_argument("--serialization_prefix", default="")
    # reference flags.cc parity
    ap.add_argument("--lb", default="", choices=("", "cm", "wm", "strict", "none"),
                    help="GPU edge-expansion scheduler (reference --lb)")
    ap.add_argument("--rebalance", action="store_true",
                    help="degree-balanced ownership (reference rebalancer)")
    ap.add_argument("--rebalance_vertex_factor", type=float, default=1.0)
    ap.add_argument("--vertex_num", type=int, default=0,
                    help="declared |V| (extends/validates the id space)")
    ap.add_argument("--edge_num", type=int, default=0,
                    help="declared |E| (accepted for reference parity)")
    ap.add_argument("--jobid", default="", help="echoed in the timer header")
    return ap


def fmt_value(v):
    if isinstance(v, np.floating) or isinstance(v, float):
        return repr(float(v))
    return str(v)


def main(argv=None):
    args = build_parser().parse_args(argv)
    if args.lb:
        os.environ["GRAPEHIP_LB"] = args.lb
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    eng = grapehip.engine_from_env(n_threads=args.app_concurrency,
                                   gpu=args.gpu)

    timers = []
    t = time.time()

    weighted = args.weighted or args.application == "sssp"
    if args.deserialize:
        g = eng.load_serialized(args.serialization_prefix)
    else:
        if not args.efile:
            raise SystemExit("--efile is required (or --deserialize)")
        src, dst, w = read_ldbc_edges(args.efile, weighted=weighted,
                                      rank=rank, world=world)
        kw = {}
        if args.rebalance:
            from grapehip.io import rebalance_partition
            nv_hint = args.vertex_num or int(
                max(src.max(initial=0), dst.max(initial=0)) + 1)
            owned = rebalance_partition(eng, src, dst, nv_hint,
                                        factor=args.rebalance_vertex_factor)
            kw["vertex_oids"] = owned
            kw["partitioner"] = "map"
            kw["idxer"] = args.idxer
        elif args.vfile:
            oids = read_ldbc_vertices(args.vfile, rank=rank, world=world)
            kw["vertex_oids"] = oids
            kw["idxer"] = args.idxer
        else:
            hi = max(src.max(initial=0), dst.max(initial=0)) + 1
            kw["num_vertices"] = int(max(hi, args.vertex_num))
        g = eng.load_edges(src, dst, weights=w, directed=args.directed,
                           build_in_csr=args.directed, **kw)
        if args.vertex_num and g.num_vertices != args.vertex_num and rank == 0:
            print("[warn] --vertex_num %d but loaded %d" %
                  (args.vertex_num, g.num_vertices))
        if args.serialize:
            os.makedirs(args.serialization_prefix, exist_ok=True)
            eng.save_graph(g, args.serialization_prefix)
    if args.efile_add or args.efile_remove:
        empty = np.array([], dtype=np.int64)
        a_s = a_d = r_s = r_d = empty
        a_w = None
        if args.efile_add:
            a_s, a_d, a_w = read_ldbc_edges(args.efile_add,
                                            weighted=weighted,
                                            rank=rank, world=world)
        if args.efile_remove:
            r_s, r_d, _ = read_ldbc_edges(args.efile_remove, rank=rank,
                                          world=world)
        g = eng.mutate_graph(g, add_src=a_s, add_dst=a_d, add_weights=a_w,
                             remove_src=r_s, remove_dst=r_d,
                             remove_vertices=empty)
    timers.append(("load graph", time.time() - t))

    t = time.time()
    app = args.application
    if app == "bfs":
        res = eng.bfs(g, args.bfs_source)
    elif app == "sssp":
        res = eng.sssp(g, args.sssp_source)
    elif app == "pagerank":
        res = eng.pagerank(g, args.pr_d, args.pr_mr)
    elif app == "wcc":
        res = eng.wcc(g)
    elif app == "cdlp":
        res = eng.cdlp(g, args.cdlp_mr)
    elif app == "lcc":
        res = eng.lcc(g)
    elif app == "bc":
        res = eng.bc(g, args.bc_source)
    elif app == "kcore":
        res = eng.kcore(g, args.kcore_k)
    elif app == "core_decomposition":
        res = eng.core_decomposition(g)
    elif app == "kclique":
        res = eng.kclique(g, args.kclique_k)
    timers.append(("run algorithm", time.time() - t))

    t = time.time()
    if args.out_prefix:
        os.makedirs(args.out_prefix, exist_ok=True)
        path = os.path.join(args.out_prefix,
                            "result_frag_%d" % rank)
        with open(path, "w") as f:
            if app == "kclique":
                if rank == 0:
                    f.write("%d\n" % res["clique_count"])
            else:
                for oid, val in zip(res["oids"], res["values"]):
                    f.write("%d %s\n" % (oid, fmt_value(val)))
    timers.append(("output", time.time() - t))

    if rank == 0:
        if args.jobid:
            print("[job] %s" % args.jobid)
        for name, secs in timers:
            print("[timer] %-14s %.6f s" % (name, secs))
        # reference GetMemoryUsage line printed with the timer table
        mi = eng.memory_info()
        parts = ["%s=%.1fMB" % (k, mi[k] / 1048576.0)
                 for k in ("VmHWM", "VmRSS", "hip_alloc_peak", "hip_used")
                 if k in mi]
        print("[memory] " + " ".join(parts))
        if app == "kclique":
            print("clique_num = %d" % res["clique_count"])
    return 0


if __name__ == "__main__":
    sys.exit(main())
