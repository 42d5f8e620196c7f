"""Multi-rank (world_size 2/3) CPU-path tests: distributed results must merge
to exactly the oracle's answer. Exercises the TCP control plane, the
distributed loader shuffle, outer-vertex halo messaging, mirror exchange and
BSP termination — the same machinery the RCCL path scales on."""
import json
import os
import subprocess
import sys
from pathlib import Path

import numpy as np
import pytest

from oracles import (bc_oracle, bfs_oracle, cdlp_oracle, coreness_oracle,
                     kclique_oracle, kcore_oracle, lcc_oracle,
                     pagerank_oracle, sssp_oracle, wcc_oracle)

REPO = Path(__file__).resolve().parent.parent
WORKER = REPO / "tests" / "mp_worker.py"


def run_world(world, cfg, free_port, tmp_path, timeout=120):
    cfg = dict(cfg, out_dir=str(tmp_path))
    procs = []
    for rank in range(world):
        env = dict(os.environ,
                   RANK=str(rank), WORLD_SIZE=str(world),
                   MASTER_ADDR="127.0.0.1", MASTER_PORT=str(free_port - 17),
                   GRAPEHIP_TEST_CFG=json.dumps(cfg))
        procs.append(subprocess.Popen([sys.executable, str(WORKER)], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for p in procs:
        out, _ = p.communicate(timeout=timeout)
        assert p.returncode == 0, out.decode()
    oids, vals = [], []
    for rank in range(world):
        z = np.load(tmp_path / f"rank{rank}.npz")
        oids.append(z["oids"])
        vals.append(z["values"])
    oids = np.concatenate(oids)
    vals = np.concatenate(vals)
    order = np.argsort(oids)
    return oids[order], vals[order]


def graph_arrays(cfg):
    rng = np.random.default_rng(cfg["seed"])
    src = rng.integers(0, cfg["num_v"], size=cfg["num_e"], dtype=np.int64)
    dst = rng.integers(0, cfg["num_v"], size=cfg["num_e"], dtype=np.int64)
    keep = src != dst
    src, dst = src[keep], dst[keep]
    w = rng.random(len(src), dtype=np.float32) * 9 + 1 if cfg["weighted"] else None
    return src, dst, w


BASE = dict(num_v=300, num_e=1800, seed=101, weighted=False, directed=True,
            in_csr=False, source=3, vertex_oids=False)


@pytest.mark.parametrize("world", [2, 3, 4])
def test_bfs_mp(world, free_port, tmp_path):
    cfg = dict(BASE, app="bfs")
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    assert np.array_equal(vals, bfs_oracle(cfg["num_v"], src, dst, 3))


@pytest.mark.parametrize("world", [2, 4])
def test_sssp_mp(world, free_port, tmp_path):
    cfg = dict(BASE, app="sssp", weighted=True)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, w = graph_arrays(cfg)
    assert np.allclose(vals, sssp_oracle(cfg["num_v"], src, dst, w, 3),
                       rtol=1e-9)


@pytest.mark.parametrize("world", [2, 4])
def test_pagerank_mp(world, free_port, tmp_path):
    cfg = dict(BASE, app="pagerank")
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    assert np.allclose(vals, pagerank_oracle(cfg["num_v"], src, dst),
                       rtol=1e-9)


@pytest.mark.parametrize("world", [2])
def test_wcc_mp(world, free_port, tmp_path):
    cfg = dict(BASE, app="wcc", directed=False, num_e=250, num_v=400)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    assert np.array_equal(vals, wcc_oracle(cfg["num_v"], src, dst))


@pytest.mark.parametrize("world", [2])
def test_cdlp_mp(world, free_port, tmp_path):
    cfg = dict(BASE, app="cdlp", directed=False, num_v=100, num_e=400)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    assert np.array_equal(vals, cdlp_oracle(cfg["num_v"], src, dst))


@pytest.mark.parametrize("world", [2, 3])
def test_lcc_mp(world, free_port, tmp_path):
    cfg = dict(BASE, app="lcc", directed=False, num_v=80, num_e=600)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    # engine treats each undirected input edge once per side; duplicate
    # (u,v)/(v,u) pairs in the random input collapse in both impls
    assert np.allclose(vals, lcc_oracle(cfg["num_v"], src, dst,
                                        directed=False), rtol=1e-12)


@pytest.mark.parametrize("world", [2])
def test_bfs_mp_hashmap_oids(world, free_port, tmp_path):
    cfg = dict(BASE, app="bfs", vertex_oids=True, source=3 * 3 + 1)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    expect = bfs_oracle(cfg["num_v"], src, dst, 3)
    assert np.array_equal(oids, np.arange(cfg["num_v"]) * 3 + 1)
    assert np.array_equal(vals, expect)


@pytest.mark.parametrize("world", [2])
def test_bc_mp(world, free_port, tmp_path):
    cfg = dict(BASE, app="bc", directed=False, num_v=250, num_e=1200)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    delta, _, depth = bc_oracle(cfg["num_v"], src, dst, cfg["source"],
                                directed=False)
    reach = depth < 1e300
    assert np.allclose(vals[reach], delta[reach], rtol=1e-9, atol=1e-9)


@pytest.mark.parametrize("world", [2, 3])
def test_kcore_mp(world, free_port, tmp_path):
    cfg = dict(BASE, app="kcore", directed=False, num_v=300, num_e=2000, k=4)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    assert np.array_equal(vals, kcore_oracle(cfg["num_v"], src, dst, 4))


@pytest.mark.parametrize("world", [2])
def test_core_decomposition_mp(world, free_port, tmp_path):
    cfg = dict(BASE, app="core_decomposition", directed=False, num_v=250,
               num_e=1500)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    assert np.array_equal(vals, coreness_oracle(cfg["num_v"], src, dst))


@pytest.mark.parametrize("world", [2, 3])
def test_kclique_mp(world, free_port, tmp_path):
    cfg = dict(BASE, app="kclique", directed=False, num_v=100, num_e=900, k=4)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    # every rank reports the same global count
    assert (vals == kclique_oracle(cfg["num_v"], src, dst, 4)).all()


@pytest.mark.parametrize("world", [2, 3])
def test_sampler_mp(world, free_port, tmp_path):
    # walks hop across fragments; every consecutive pair must be an edge
    cfg = dict(BASE, app="sample", directed=True, num_v=200, num_e=1600,
               n_walks=20, hops=3)
    wids, paths = run_world(world, cfg, free_port, tmp_path)
    assert np.array_equal(np.sort(wids), np.arange(20))
    src, dst, _ = graph_arrays(cfg)
    adj = {}
    for s, d in zip(src, dst):
        adj.setdefault(int(s), set()).add(int(d))
    for wid, path in zip(wids, paths):
        assert path[0] == wid
        for h in range(3):
            a, b = int(path[h]), int(path[h + 1])
            if a == -1 or b == -1:
                continue
            assert b in adj.get(a, set()), (wid, h)


@pytest.mark.parametrize("world", [2, 3])
def test_pagerank_vc_mp(world, free_port, tmp_path):
    cfg = dict(BASE, app="pagerank_vc", num_v=400, num_e=2600)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    assert np.allclose(vals, pagerank_oracle(cfg["num_v"], src, dst),
                       rtol=1e-9)


@pytest.mark.parametrize("world", [2, 3])
def test_bfs_mp_rebalanced(world, free_port, tmp_path):
    # degree-balanced contiguous map partition (reference rebalancer.h)
    cfg = dict(BASE, app="bfs", rebalance=True)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    assert np.array_equal(np.sort(oids), np.arange(cfg["num_v"]))
    assert np.array_equal(vals, bfs_oracle(cfg["num_v"], src, dst, 3))


@pytest.mark.parametrize("world", [2])
def test_sssp_auto_mp(world, free_port, tmp_path):
    cfg = dict(BASE, app="sssp_auto", weighted=True)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, w = graph_arrays(cfg)
    expect = sssp_oracle(cfg["num_v"], src, dst, w, 3)
    finite = expect < 1e300
    assert np.allclose(vals[finite], expect[finite], rtol=1e-9)


@pytest.mark.parametrize("world", [2, 3])
def test_bfs_mp_mph_idxer(world, free_port, tmp_path):
    cfg = dict(BASE, app="bfs", vertex_oids=True, idxer="mph",
               source=3 * 3 + 1)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    assert np.array_equal(oids, np.arange(cfg["num_v"]) * 3 + 1)
    assert np.array_equal(vals, bfs_oracle(cfg["num_v"], src, dst, 3))


@pytest.mark.parametrize("world", [8])
def test_bfs_mp_world8(world, free_port, tmp_path):
    # the SCALE tier's rank count: full 8-way TCP mesh + shuffle + halo
    cfg = dict(BASE, app="bfs", num_v=500, num_e=3000)
    oids, vals = run_world(world, cfg, free_port, tmp_path, timeout=240)
    src, dst, _ = graph_arrays(cfg)
    assert np.array_equal(vals, bfs_oracle(cfg["num_v"], src, dst, 3))


@pytest.mark.parametrize("world", [2, 3])
def test_mutation_mp(world, free_port, tmp_path):
    # collective mutate: per-rank delta slices must merge identically
    cfg = dict(BASE, app="bfs", mutate=True, directed=True)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    rng2 = np.random.default_rng(cfg["seed"] + 1)
    ad_s = rng2.integers(0, cfg["num_v"], 300).astype(np.int64)
    ad_d = rng2.integers(0, cfg["num_v"], 300).astype(np.int64)
    k = ad_s != ad_d
    ad_s, ad_d = ad_s[k], ad_d[k]
    # removals: every rank removed the first 50 of ITS slice = a global set
    rm = set()
    for r in range(world):
        sl = slice(r, None, world)
        for a, b in zip(src[sl][:50], dst[sl][:50]):
            rm.add((a, b))
    keep = np.array([(a, b) not in rm for a, b in zip(src, dst)])
    e_src = np.concatenate([src[keep], ad_s])
    e_dst = np.concatenate([dst[keep], ad_d])
    assert np.array_equal(vals, bfs_oracle(cfg["num_v"], e_src, e_dst, 3))


@pytest.mark.parametrize("world", [2, 4])
def test_serialize_roundtrip_mp(world, free_port, tmp_path):
    # per-rank checkpoint + reload mid-run (reference app_tests.sh:53-66)
    cfg = dict(BASE, app="bfs", serialize_roundtrip=True)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    assert np.array_equal(vals, bfs_oracle(cfg["num_v"], src, dst, 3))


@pytest.mark.parametrize("world", [2])
def test_serialize_roundtrip_mp_hashmap(world, free_port, tmp_path):
    cfg = dict(BASE, app="bfs", serialize_roundtrip=True, vertex_oids=True,
               source=3 * 3 + 1)
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    src, dst, _ = graph_arrays(cfg)
    assert np.array_equal(vals, bfs_oracle(cfg["num_v"], src, dst, 3))


@pytest.mark.parametrize("world", [2, 3])
def test_force_terminate_mp(world, free_port, tmp_path):
    """Cooperative abort (reference ForceTerminate + TerminateInfo,
    default_message_manager.h:156-166): rank 1 aborts in PEval; every
    rank raises at the round boundary carrying rank 1's info string."""
    cfg = dict(BASE, app="force_terminate")
    oids, vals = run_world(world, cfg, free_port, tmp_path)
    assert len(vals) == world
    assert (vals == 1).all()  # 1 = raised with the right info on that rank


@pytest.mark.parametrize("world", [2, 3, 8])
def test_large_messages_eager_flush(world, free_port, tmp_path):
    # boundary messages exceeding the flush block force the mid-round
    # flush (background sender thread); results must still merge exactly
    cfg = dict(BASE, app="sssp", num_v=60000, num_e=360000, weighted=True,
               directed=False, source=11)
    oids, vals = run_world(world, cfg, free_port, tmp_path, timeout=300)
    src, dst, w = graph_arrays(cfg)
    expect = sssp_oracle(cfg["num_v"], src, dst, w, 11, directed=False)
    finite = expect < 1e300
    assert np.array_equal(vals >= 1e300, ~finite)
    assert np.allclose(vals[finite], expect[finite], rtol=1e-4)
