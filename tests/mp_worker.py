"""Subprocess worker for multi-rank CPU tests.

Launched by test_multiproc.py with RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT
set; loads a deterministic slice of the shared synthetic graph, runs the
requested app, writes this rank's (oids, values) to OUT_DIR/rank<k>.npz.
"""
import json
import os
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import grapehip  # noqa: E402


def main():
    cfg = json.loads(os.environ["GRAPEHIP_TEST_CFG"])
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    eng = grapehip.engine_from_env(n_threads=2)

    rng = np.random.default_rng(cfg["seed"])
    src = rng.integers(0, cfg["num_v"], size=cfg["num_e"], dtype=np.int64)
    dst = rng.integers(0, cfg["num_v"], size=cfg["num_e"], dtype=np.int64)
    keep = src != dst
    src, dst = src[keep], dst[keep]
    w = None
    if cfg["weighted"]:
        w = rng.random(len(src), dtype=np.float32) * 9 + 1
    # each rank takes a strided slice of the same global edge list
    sl = slice(rank, None, world)
    kw = {}
    if cfg.get("rebalance"):
        oids = grapehip.rebalance_partition(eng, src[sl], dst[sl],
                                            cfg["num_v"])
        kw["vertex_oids"] = oids
        kw["partitioner"] = "map"
    elif cfg.get("vertex_oids"):
        # split vertex list round-robin too
        all_oids = np.arange(cfg["num_v"], dtype=np.int64) * 3 + 1
        kw["vertex_oids"] = all_oids[sl]
        if cfg.get("idxer"):
            kw["idxer"] = cfg["idxer"]
        src = all_oids[src]
        dst = all_oids[dst]
    else:
        kw["num_vertices"] = cfg["num_v"]
    g = eng.load_edges(src[sl], dst[sl],
                       weights=None if w is None else w[sl],
                       directed=cfg["directed"],
                       build_in_csr=cfg["in_csr"], **kw)

    app = cfg["app"]
    if cfg.get("serialize_roundtrip"):
        # per-fragment checkpoint files in a shared dir; reload replaces g
        ckpt = os.path.join(cfg["out_dir"], "ckpt")
        if rank == 0:
            os.makedirs(ckpt, exist_ok=True)
        eng.barrier()
        eng.save_graph(g, ckpt)
        g = eng.load_serialized(ckpt)
    if cfg.get("mutate"):
        # collective delta: each rank contributes a slice of adds/removes
        rng2 = np.random.default_rng(cfg["seed"] + 1)
        ad_s = rng2.integers(0, cfg["num_v"], 300).astype(np.int64)
        ad_d = rng2.integers(0, cfg["num_v"], 300).astype(np.int64)
        k = ad_s != ad_d
        ad_s, ad_d = ad_s[k], ad_d[k]
        msl = slice(rank, None, world)
        g = eng.mutate_graph(g, add_src=ad_s[msl], add_dst=ad_d[msl],
                             remove_src=src[sl][:50], remove_dst=dst[sl][:50],
                             remove_vertices=np.array([], dtype=np.int64))
    if app == "sssp_auto":
        res = eng.sssp_auto(g, cfg["source"])
    elif app == "bfs":
        res = eng.bfs(g, cfg["source"])
    elif app == "sssp":
        res = eng.sssp(g, cfg["source"])
    elif app == "pagerank":
        res = eng.pagerank(g, 0.85, 10)
    elif app == "wcc":
        res = eng.wcc(g)
    elif app == "cdlp":
        res = eng.cdlp(g, 10)
    elif app == "lcc":
        res = eng.lcc(g)
    elif app == "bc":
        res = eng.bc(g, cfg["source"])
    elif app == "kcore":
        res = eng.kcore(g, cfg.get("k", 3))
    elif app == "core_decomposition":
        res = eng.core_decomposition(g)
    elif app == "pagerank_vc":
        gvc = eng.load_vertexcut(src[sl], dst[sl],
                                 num_vertices=cfg["num_v"])
        res = eng.pagerank_vc(gvc, 0.85, 10)
    elif app == "sample":
        r = eng.sample(g, np.arange(cfg.get("n_walks", 20), dtype=np.int64),
                       hops=cfg.get("hops", 3), seed=5)
        res = {"oids": r["walk_ids"], "values": r["paths"]}
    elif app == "force_terminate":
        # cooperative abort: rank 1 aborts; EVERY rank must raise with
        # the aborting rank's info string at the round boundary
        try:
            eng._test_force_terminate(g, 1)
            res = {"oids": np.array([rank], dtype=np.int64),
                   "values": np.array([0], dtype=np.int64)}
        except RuntimeError as ex:
            ok = "boom from rank 1" in str(ex)
            res = {"oids": np.array([rank], dtype=np.int64),
                   "values": np.array([1 if ok else -1], dtype=np.int64)}
    elif app == "kclique":
        res = eng.kclique(g, cfg.get("k", 3))
        res = dict(res, oids=np.array([0], dtype=np.int64),
                   values=np.array([res["clique_count"]], dtype=np.int64))
    else:
        raise ValueError(app)

    np.savez(os.path.join(cfg["out_dir"], f"rank{rank}.npz"),
             oids=res["oids"], values=res["values"])


if __name__ == "__main__":
    main()
