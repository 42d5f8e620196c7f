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
