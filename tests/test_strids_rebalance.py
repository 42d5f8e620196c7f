"""String vertex ids (IdIndexer parity) + degree-balanced rebalancer."""
import numpy as np
import pytest

import grapehip
from oracles import bfs_oracle


@pytest.fixture(scope="module")
def eng():
    return grapehip.Engine(rank=0, world=1, master_port=29681)


def test_string_oids(eng):
    rng = np.random.default_rng(113)
    names = ["node_%04d" % i for i in range(300)]
    si = rng.integers(0, 300, 2000)
    di = rng.integers(0, 300, 2000)
    keep = si != di
    si, di = si[keep], di[keep]
    src_strs = [names[i] for i in si]
    dst_strs = [names[i] for i in di]
    g, i2s = grapehip.load_string_edges(eng, src_strs, dst_strs,
                                        directed=False)
    assert sorted(i2s) == sorted(set(src_strs) | set(dst_strs))
    r = eng.bfs(g, i2s.index("node_0005") if "node_0005" in i2s else 0)
    # map back: dense ids are the sorted-string order
    sid = np.array([i2s.index("node_%04d" % i) for i in range(300)
                    if ("node_%04d" % i) in i2s])
    # oracle over the dense-id graph
    s2pos = {s: i for i, s in enumerate(i2s)}
    osrc = np.array([s2pos[s] for s in src_strs])
    odst = np.array([s2pos[s] for s in dst_strs])
    source = s2pos.get("node_0005", 0)
    r = eng.bfs(g, source)
    order = np.argsort(r["oids"])
    expect = bfs_oracle(len(i2s), osrc, odst, source, directed=False)
    assert np.array_equal(r["values"][order], expect)


def test_rebalance_partition_single(eng):
    rng = np.random.default_rng(117)
    nv = 500
    src = rng.integers(0, nv, 4000).astype(np.int64)
    dst = rng.integers(0, nv, 4000).astype(np.int64)
    oids = grapehip.rebalance_partition(eng, src, dst, nv)
    assert np.array_equal(oids, np.arange(nv))  # world=1: everything
    g = eng.load_edges(src, dst, directed=True, vertex_oids=oids)
    assert g.num_vertices == nv


def test_mph_idxer(eng):
    # minimal-perfect-hash idxer (pthash parity): same results as the
    # hashmap idxer on sparse oids, including checkpoint round-trip
    rng = np.random.default_rng(223)
    nv = 4000
    oids = np.sort(rng.choice(10**12, size=nv, replace=False)).astype(
        np.int64)
    si = rng.integers(0, nv, 30000)
    di = rng.integers(0, nv, 30000)
    keep = si != di
    si, di = si[keep], di[keep]
    g_h = eng.load_edges(oids[si], oids[di], directed=False,
                         vertex_oids=oids, idxer="hashmap")
    g_m = eng.load_edges(oids[si], oids[di], directed=False,
                         vertex_oids=oids, idxer="mph")
    src_oid = int(oids[17])
    rh = eng.bfs(g_h, src_oid)
    rm = eng.bfs(g_m, src_oid)
    oh, om = np.argsort(rh["oids"]), np.argsort(rm["oids"])
    assert np.array_equal(rh["oids"][oh], rm["oids"][om])
    assert np.array_equal(rh["values"][oh], rm["values"][om])
    rh = eng.pagerank(g_h, 0.85, 5)
    rm = eng.pagerank(g_m, 0.85, 5)
    oh, om = np.argsort(rh["oids"]), np.argsort(rm["oids"])
    assert np.allclose(rh["values"][oh], rm["values"][om], rtol=1e-12)


def test_mph_idxer_roundtrip(eng, tmp_path):
    rng = np.random.default_rng(227)
    nv = 800
    oids = np.sort(rng.choice(10**9, size=nv, replace=False)).astype(
        np.int64)
    si = rng.integers(0, nv, 5000)
    di = rng.integers(0, nv, 5000)
    keep = si != di
    g = eng.load_edges(oids[si[keep]], oids[di[keep]], directed=False,
                       vertex_oids=oids, idxer="mph")
    before = eng.wcc(g)
    eng.save_graph(g, str(tmp_path))
    g2 = eng.load_serialized(str(tmp_path))
    after = eng.wcc(g2)
    o1, o2 = np.argsort(before["oids"]), np.argsort(after["oids"])
    assert np.array_equal(before["oids"][o1], after["oids"][o2])
    assert np.array_equal(before["values"][o1], after["values"][o2])
