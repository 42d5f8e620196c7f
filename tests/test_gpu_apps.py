"""GPU-path validation (single MI355X): HIP kernels vs the same NumPy
oracles the CPU path is validated against, on identical graphs."""
import numpy as np
import pytest

import grapehip
from oracles import (bfs_oracle, cdlp_oracle, lcc_oracle, pagerank_oracle,
                     sssp_oracle, wcc_oracle, INT64_MAX)

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    return grapehip.Engine(rank=0, world=1, master_port=29711, gpu=True)


def random_graph(num_v=5000, num_e=60000, seed=7, weighted=False):
    rng = np.random.default_rng(seed)
    src = rng.integers(0, num_v, size=num_e, dtype=np.int64)
    dst = rng.integers(0, num_v, size=num_e, dtype=np.int64)
    keep = src != dst
    src, dst = src[keep], dst[keep]
    w = (rng.random(len(src), dtype=np.float32) * 9 + 1) if weighted else None
    return src, dst, w


def by_oid(res):
    order = np.argsort(res["oids"])
    return res["oids"][order], res["values"][order]


def test_gpu_bfs_directed(eng):
    src, dst, _ = random_graph()
    g = eng.load_edges(src, dst, directed=True, num_vertices=5000)
    _, vals = by_oid(eng.bfs(g, 3))
    assert np.array_equal(vals, bfs_oracle(5000, src, dst, 3, directed=True))


def test_gpu_bfs_undirected_sparse(eng):
    # sparse: exercises unreachable vertices + many levels
    src, dst, _ = random_graph(num_v=20000, num_e=30000, seed=11)
    g = eng.load_edges(src, dst, directed=False, num_vertices=20000)
    _, vals = by_oid(eng.bfs(g, 5))
    assert np.array_equal(vals, bfs_oracle(20000, src, dst, 5, directed=False))


def test_gpu_sssp(eng):
    src, dst, w = random_graph(weighted=True)
    g = eng.load_edges(src, dst, weights=w, directed=True, num_vertices=5000)
    _, vals = by_oid(eng.sssp(g, 3))
    expect = sssp_oracle(5000, src, dst, w, 3, directed=True)
    # fp32 relaxation vs fp64 oracle
    finite = expect < 1e300
    assert np.array_equal(vals >= 1e300, ~finite)
    assert np.allclose(vals[finite], expect[finite], rtol=1e-4)


def test_gpu_sssp_undirected(eng):
    src, dst, w = random_graph(num_v=3000, num_e=20000, seed=13, weighted=True)
    g = eng.load_edges(src, dst, weights=w, directed=False, num_vertices=3000)
    _, vals = by_oid(eng.sssp(g, 7))
    expect = sssp_oracle(3000, src, dst, w, 7, directed=False)
    finite = expect < 1e300
    assert np.allclose(vals[finite], expect[finite], rtol=1e-4)


def test_gpu_pagerank(eng):
    src, dst, _ = random_graph()
    g = eng.load_edges(src, dst, directed=True, num_vertices=5000)
    _, vals = by_oid(eng.pagerank(g, 0.85, 10))
    expect = pagerank_oracle(5000, src, dst, 0.85, 10, directed=True)
    # fp32 contributions (the reference's GPU precision) vs fp64 oracle
    assert np.allclose(vals, expect, rtol=3e-5, atol=1e-12)
    assert abs(vals.sum() - 1.0) < 1e-5


def test_gpu_wcc(eng):
    src, dst, _ = random_graph(num_v=8000, num_e=6000, seed=17)
    g = eng.load_edges(src, dst, directed=False, num_vertices=8000)
    _, vals = by_oid(eng.wcc(g))
    assert np.array_equal(vals, wcc_oracle(8000, src, dst))


def test_gpu_wcc_directed_sparse(eng):
    # regression: on a directed graph the Afforest giant-skip must not
    # drop edges stored only in skipped giant rows — an in-degree-only
    # vertex reachable solely via such an edge was left unmerged
    src, dst, _ = random_graph(num_v=60000, num_e=200000, seed=7)
    g = eng.load_edges(src, dst, directed=True, num_vertices=60000,
                       build_in_csr=True)
    _, vals = by_oid(eng.wcc(g))
    assert np.array_equal(vals, wcc_oracle(60000, src, dst))


def test_gpu_wcc_directed_no_incsr(eng):
    # without an in-CSR the skip is disabled entirely; result must agree
    src, dst, _ = random_graph(num_v=30000, num_e=90000, seed=23)
    g = eng.load_edges(src, dst, directed=True, num_vertices=30000)
    _, vals = by_oid(eng.wcc(g))
    assert np.array_equal(vals, wcc_oracle(30000, src, dst))


def test_gpu_synthetic(eng):
    g = eng.load_synthetic(num_vertices=100000, num_edges=1600000, seed=1,
                           weighted=True)
    assert g.num_vertices == 100000
    assert g.input_edges == 1600000
    # undirected storage holds both orientations minus self-loop dupes
    assert 1600000 < g.num_edges <= 3200000
    # rows come back with an explicit oid list (hub renumbering reorders
    # rows internally) — index results by oid
    _, d = by_oid(eng.bfs(g, 0))
    assert d[0] == 0
    reached = d < INT64_MAX
    assert reached.sum() > 1000  # hub-connected RMAT core
    r2 = eng.pagerank(g, 0.85, 5)
    assert abs(r2["values"].sum() - 1.0) < 1e-5
    _, sd = by_oid(eng.sssp(g, 0))
    # any BFS-reachable vertex must be SSSP-reachable and dist >= depth
    assert (sd[reached] < 1e300).all()
    _, labs = by_oid(eng.wcc(g))
    # all BFS-reachable vertices share vertex 0's component label
    assert (labs[reached] == labs[0]).all()


def test_gpu_cdlp_undirected(eng):
    src, dst, _ = random_graph(num_v=4000, num_e=30000, seed=23)
    g = eng.load_edges(src, dst, directed=False, num_vertices=4000)
    _, vals = by_oid(eng.cdlp(g, 5))
    assert np.array_equal(vals, cdlp_oracle(4000, src, dst, 5, directed=False))


def test_gpu_cdlp_directed(eng):
    src, dst, _ = random_graph(num_v=2000, num_e=15000, seed=29)
    g = eng.load_edges(src, dst, directed=True, num_vertices=2000,
                       build_in_csr=True)
    _, vals = by_oid(eng.cdlp(g, 6))
    assert np.array_equal(vals, cdlp_oracle(2000, src, dst, 6, directed=True))


def test_gpu_cdlp_hub_tiers(eng):
    # exercises every CDLP tier: tiny (<=16), ballot (<=64), wave-hash
    # (<=512), block LDS hash (<=4096), global hash (hub)
    rng = np.random.default_rng(31)
    nv = 30000
    hub_e = np.stack([np.zeros(12000, np.int64),
                      rng.integers(1, nv, 12000)], 1)
    mid_e = np.stack([np.ones(2000, np.int64),
                      rng.integers(2, nv, 2000)], 1)
    wave_rows = []
    for v in range(2, 12):  # ten rows with degree ~100-400 (wave tier)
        d = int(rng.integers(80, 400))
        wave_rows.append(np.stack([np.full(d, v, np.int64),
                                   rng.integers(0, nv, d)], 1))
    ballot_rows = []
    for v in range(12, 30):  # rows with degree ~20-60 (ballot tier)
        d = int(rng.integers(18, 60))
        ballot_rows.append(np.stack([np.full(d, v, np.int64),
                                     rng.integers(0, nv, d)], 1))
    rest = np.stack([rng.integers(0, nv, 40000),
                     rng.integers(0, nv, 40000)], 1)
    e = np.concatenate([hub_e, mid_e, *wave_rows, *ballot_rows, rest])
    keep = e[:, 0] != e[:, 1]
    src, dst = e[keep, 0], e[keep, 1]
    g = eng.load_edges(src, dst, directed=False, num_vertices=nv)
    _, vals = by_oid(eng.cdlp(g, 4))
    assert np.array_equal(vals, cdlp_oracle(nv, src, dst, 4, directed=False))


def test_gpu_lcc_undirected(eng):
    src, dst, _ = random_graph(num_v=2000, num_e=40000, seed=37)
    g = eng.load_edges(src, dst, directed=False, num_vertices=2000)
    _, vals = by_oid(eng.lcc(g))
    expect = lcc_oracle(2000, src, dst, directed=False)
    assert np.allclose(vals, expect, rtol=1e-12)


def test_gpu_lcc_directed(eng):
    src, dst, _ = random_graph(num_v=1500, num_e=20000, seed=41)
    g = eng.load_edges(src, dst, directed=True, num_vertices=1500,
                       build_in_csr=True)
    _, vals = by_oid(eng.lcc(g))
    expect = lcc_oracle(1500, src, dst, directed=True)
    assert np.allclose(vals, expect, rtol=1e-12)


def test_gpu_lcc_directed_hub(eng):
    rng = np.random.default_rng(47)
    nv = 8000
    hub = np.stack([np.zeros(6000, np.int64), rng.integers(1, nv, 6000)], 1)
    rev = np.stack([rng.integers(1, nv, 3000), np.zeros(3000, np.int64)], 1)
    rest = np.stack([rng.integers(0, nv, 30000),
                     rng.integers(0, nv, 30000)], 1)
    e = np.concatenate([hub, rev, rest])
    keep = e[:, 0] != e[:, 1]
    src, dst = e[keep, 0], e[keep, 1]
    g = eng.load_edges(src, dst, directed=True, num_vertices=nv,
                       build_in_csr=True)
    _, vals = by_oid(eng.lcc(g))
    expect = lcc_oracle(nv, src, dst, directed=True)
    assert np.allclose(vals, expect, rtol=1e-12)


def test_gpu_lcc_hub(eng):
    # hub row exercises the global-hash tier + big-row bitonic sort
    rng = np.random.default_rng(43)
    nv = 20000
    hub = np.stack([np.zeros(9000, np.int64), rng.integers(1, nv, 9000)], 1)
    rest = np.stack([rng.integers(0, nv, 60000),
                     rng.integers(0, nv, 60000)], 1)
    e = np.concatenate([hub, rest])
    keep = e[:, 0] != e[:, 1]
    src, dst = e[keep, 0], e[keep, 1]
    g = eng.load_edges(src, dst, directed=False, num_vertices=nv)
    _, vals = by_oid(eng.lcc(g))
    expect = lcc_oracle(nv, src, dst, directed=False)
    assert np.allclose(vals, expect, rtol=1e-12)


def test_gpu_synthetic_directed_incsr(eng):
    # directed synthetic with in-CSR: CDLP runs (needs in+out multiset) and
    # in/out edge totals agree
    g = eng.load_synthetic(num_vertices=50000, num_edges=400000, seed=3,
                           directed=True, weighted=True, build_in_csr=True)
    r = eng.cdlp(g, 3)
    assert len(r["values"]) == 50000
    r2 = eng.pagerank(g, 0.85, 5)
    assert abs(r2["values"].sum() - 1.0) < 1e-5


def test_gpu_arbitrary_oids(eng):
    # hashmap vertex map: sparse non-dense oids, densely renumbered on
    # upload (reference needs cuda_hashmap's DeviceVertexMap for this)
    rng = np.random.default_rng(71)
    nv = 3000
    oids = np.sort(rng.choice(10**9, size=nv, replace=False)).astype(np.int64)
    si = rng.integers(0, nv, 25000)
    di = rng.integers(0, nv, 25000)
    keep = si != di
    si, di = si[keep], di[keep]
    w = (rng.random(len(si), dtype=np.float32) * 9 + 1)
    g = eng.load_edges(oids[si], oids[di], weights=w, directed=False,
                       vertex_oids=oids)
    src_oid = int(oids[7])
    r = eng.bfs(g, src_oid)
    got = dict(zip(r["oids"].tolist(), r["values"].tolist()))
    exp = bfs_oracle(nv, si, di, 7, directed=False)
    for j in range(nv):
        assert got[int(oids[j])] == exp[j], j
    # pagerank sums to 1 over REAL vertices (padding must not leak mass)
    rp = eng.pagerank(g, 0.85, 10)
    assert abs(rp["values"].sum() - 1.0) < 1e-5
    expect_pr = pagerank_oracle(nv, si, di, 0.85, 10, directed=False)
    gotp = dict(zip(rp["oids"].tolist(), rp["values"].tolist()))
    for j in range(0, nv, 7):
        assert abs(gotp[int(oids[j])] - expect_pr[j]) < 3e-5 * max(
            expect_pr[j], 1e-9), j
    # wcc labels come back as REAL oids forming consistent components
    rw = eng.wcc(g)
    labs = dict(zip(rw["oids"].tolist(), rw["values"].tolist()))
    exp_w = wcc_oracle(nv, si, di)
    fwd = {}
    for j in range(nv):
        assert labs[int(oids[j])] in labs  # label is a real vertex oid
        assert fwd.setdefault(exp_w[j], labs[int(oids[j])]) == \
            labs[int(oids[j])], j
    # cdlp labels are oids too
    rc = eng.cdlp(g, 4)
    labc = dict(zip(rc["oids"].tolist(), rc["values"].tolist()))
    oid_to_dense = {int(o): j for j, o in enumerate(oids)}
    exp_c = cdlp_oracle(nv, si, di, 4, directed=False)
    for j in range(nv):
        assert labc[int(oids[j])] == int(oids[exp_c[j]]), j
    # lcc values
    rl = eng.lcc(g)
    gotl = dict(zip(rl["oids"].tolist(), rl["values"].tolist()))
    exp_l = lcc_oracle(nv, si, di, directed=False)
    for j in range(0, nv, 11):
        assert abs(gotl[int(oids[j])] - exp_l[j]) < 1e-9, j


def test_gpu_serialized_roundtrip(eng, tmp_path):
    # checkpoint -> reload -> upload on the GPU engine
    src, dst, w = random_graph(num_v=3000, num_e=25000, seed=83,
                               weighted=True)
    g = eng.load_edges(src, dst, weights=w, directed=False,
                       num_vertices=3000)
    before = eng.sssp(g, 5)
    eng.save_graph(g, str(tmp_path))
    g2 = eng.load_serialized(str(tmp_path))
    after = eng.sssp(g2, 5)
    o1, o2 = np.argsort(before["oids"]), np.argsort(after["oids"])
    assert np.array_equal(before["oids"][o1], after["oids"][o2])
    assert np.allclose(before["values"][o1], after["values"][o2],
                       rtol=1e-6)


def test_gpu_exclusive_scan_sizes(eng):
    # hand-rolled hierarchical scan vs numpy at boundary sizes (block
    # chunk = 2048; recursion levels at multiples)
    rng = np.random.default_rng(251)
    for n in (0, 1, 2, 2047, 2048, 2049, 4096, 100000, 4194304, 4194305):
        a = rng.integers(0, 50, size=n, dtype=np.uint32)
        got = np.array(eng._debug_scan(a.tolist()), dtype=np.uint64)
        exp = np.concatenate([[0], np.cumsum(a, dtype=np.uint64)])
        assert np.array_equal(got, exp), n


def test_memory_tracking(eng):
    # reference memory_tracker parity: DeviceBuffer bytes are accounted
    # with a high-water mark
    g = eng.load_synthetic(num_vertices=100000, num_edges=800000, seed=3)
    eng.bfs(g, 0)
    mi = eng.memory_info()
    assert mi["hip_alloc_current"] > 0
    assert mi["hip_alloc_peak"] >= mi["hip_alloc_current"]
    assert mi["VmHWM"] > 0


def test_gpu_mutate_then_run(eng):
    # in-place delta mutation must compact slack and re-upload: results on
    # the mutated device graph equal a fresh build of the mutated edges
    rng = np.random.default_rng(53)
    nv = 3000
    src = rng.integers(0, nv, 20000)
    dst = rng.integers(0, nv, 20000)
    k = src != dst
    src, dst = src[k].astype(np.int64), dst[k].astype(np.int64)
    g = eng.load_edges(src, dst, directed=False, num_vertices=nv)
    ad_s = rng.integers(0, nv, 500).astype(np.int64)
    ad_d = rng.integers(0, nv, 500).astype(np.int64)
    ka = ad_s != ad_d
    ad_s, ad_d = ad_s[ka], ad_d[ka]
    empty = np.array([], dtype=np.int64)
    g = eng.mutate_graph(g, add_src=ad_s, add_dst=ad_d,
                         remove_src=src[:300].copy(),
                         remove_dst=dst[:300].copy(),
                         remove_vertices=empty)
    pairs = set()
    for a, b in zip(src[:300], dst[:300]):
        pairs.add((a, b))
        pairs.add((b, a))
    keep = np.array([(a, b) not in pairs for a, b in zip(src, dst)])
    e_s = np.concatenate([src[keep], ad_s])
    e_d = np.concatenate([dst[keep], ad_d])
    _, vals = by_oid(eng.bfs(g, 7))
    assert np.array_equal(vals, bfs_oracle(nv, e_s, e_d, 7, directed=False))
    _, w = by_oid(eng.wcc(g))
    assert np.array_equal(w, wcc_oracle(nv, e_s, e_d))
