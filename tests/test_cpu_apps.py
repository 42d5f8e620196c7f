"""Single-rank CPU-path validation of all six LDBC kernels vs NumPy oracles."""
import numpy as np
import pytest

import grapehip
from oracles import (bc_oracle, bfs_oracle, cdlp_oracle, coreness_oracle,
                     kclique_oracle, kcore_oracle, lcc_oracle,
                     pagerank_oracle, sssp_oracle, wcc_oracle, INT64_MAX,
                     DBL_MAX)


@pytest.fixture(scope="module")
def eng():
    return grapehip.Engine(rank=0, world=1)


def random_graph(num_v=250, num_e=1500, seed=7, weighted=False):
    rng = np.random.default_rng(seed)
    src = rng.integers(0, num_v, size=num_e, dtype=np.int64)
    dst = rng.integers(0, num_v, size=num_e, dtype=np.int64)
    keep = src != dst  # LDBC inputs have no self loops
    src, dst = src[keep], dst[keep]
    w = (rng.random(len(src), dtype=np.float32) * 9 + 1) if weighted else None
    return src, dst, w


def sorted_by_oid(res):
    order = np.argsort(res["oids"])
    return res["oids"][order], res["values"][order]


def test_bfs_directed(eng):
    src, dst, _ = random_graph()
    g = eng.load_edges(src, dst, directed=True, num_vertices=250)
    oids, vals = sorted_by_oid(eng.bfs(g, 3))
    expect = bfs_oracle(250, src, dst, 3, directed=True)
    assert np.array_equal(vals, expect)


def test_bfs_undirected(eng):
    src, dst, _ = random_graph(num_v=300, num_e=900, seed=9)
    g = eng.load_edges(src, dst, directed=False, num_vertices=300)
    oids, vals = sorted_by_oid(eng.bfs(g, 0))
    expect = bfs_oracle(300, src, dst, 0, directed=False)
    assert np.array_equal(vals, expect)


def test_sssp_directed(eng):
    src, dst, w = random_graph(weighted=True)
    g = eng.load_edges(src, dst, weights=w, directed=True, num_vertices=250)
    oids, vals = sorted_by_oid(eng.sssp(g, 3))
    expect = sssp_oracle(250, src, dst, w, 3, directed=True)
    assert np.allclose(vals, expect, rtol=1e-9)


def test_sssp_undirected(eng):
    src, dst, w = random_graph(num_v=120, num_e=500, seed=11, weighted=True)
    g = eng.load_edges(src, dst, weights=w, directed=False, num_vertices=120)
    oids, vals = sorted_by_oid(eng.sssp(g, 5))
    expect = sssp_oracle(120, src, dst, w, 5, directed=False)
    assert np.allclose(vals, expect, rtol=1e-9)


def test_pagerank_directed(eng):
    src, dst, _ = random_graph()
    g = eng.load_edges(src, dst, directed=True, num_vertices=250)
    oids, vals = sorted_by_oid(eng.pagerank(g, 0.85, 10))
    expect = pagerank_oracle(250, src, dst, 0.85, 10, directed=True)
    assert np.allclose(vals, expect, rtol=1e-9)
    # ranks of a stochastic-with-dangling matrix sum to ~1
    assert abs(vals.sum() - 1.0) < 1e-9


def test_wcc(eng):
    # sparse graph so several components exist
    src, dst, _ = random_graph(num_v=400, num_e=300, seed=13)
    g = eng.load_edges(src, dst, directed=False, num_vertices=400)
    oids, vals = sorted_by_oid(eng.wcc(g))
    expect = wcc_oracle(400, src, dst)
    assert np.array_equal(vals, expect)


def test_cdlp_undirected(eng):
    src, dst, _ = random_graph(num_v=100, num_e=400, seed=17)
    g = eng.load_edges(src, dst, directed=False, num_vertices=100)
    oids, vals = sorted_by_oid(eng.cdlp(g, 10))
    expect = cdlp_oracle(100, src, dst, 10)
    assert np.array_equal(vals, expect)


def test_cdlp_directed(eng):
    src, dst, _ = random_graph(num_v=100, num_e=400, seed=19)
    g = eng.load_edges(src, dst, directed=True, num_vertices=100,
                       build_in_csr=True)
    oids, vals = sorted_by_oid(eng.cdlp(g, 10))
    expect = cdlp_oracle(100, src, dst, 10)
    assert np.array_equal(vals, expect)


def test_lcc_undirected(eng):
    src, dst, _ = random_graph(num_v=80, num_e=600, seed=23)
    # dedup for clean undirected semantics
    pairs = {(min(s, d), max(s, d)) for s, d in zip(src, dst)}
    src = np.array([p[0] for p in pairs], dtype=np.int64)
    dst = np.array([p[1] for p in pairs], dtype=np.int64)
    g = eng.load_edges(src, dst, directed=False, num_vertices=80)
    oids, vals = sorted_by_oid(eng.lcc(g))
    expect = lcc_oracle(80, src, dst, directed=False)
    assert np.allclose(vals, expect, rtol=1e-12)


def test_lcc_directed(eng):
    src, dst, _ = random_graph(num_v=80, num_e=500, seed=29)
    pairs = {(s, d) for s, d in zip(src, dst)}
    src = np.array([p[0] for p in pairs], dtype=np.int64)
    dst = np.array([p[1] for p in pairs], dtype=np.int64)
    g = eng.load_edges(src, dst, directed=True, num_vertices=80,
                       build_in_csr=True)
    oids, vals = sorted_by_oid(eng.lcc(g))
    expect = lcc_oracle(80, src, dst, directed=True)
    assert np.allclose(vals, expect, rtol=1e-12)


def test_bfs_unreachable(eng):
    src = np.array([0, 1], dtype=np.int64)
    dst = np.array([1, 2], dtype=np.int64)
    g = eng.load_edges(src, dst, directed=True, num_vertices=5)
    _, vals = sorted_by_oid(eng.bfs(g, 0))
    assert vals[3] == INT64_MAX and vals[4] == INT64_MAX
    assert list(vals[:3]) == [0, 1, 2]


def test_hashmap_oids(eng):
    # non-contiguous oids via explicit vertex list (hashmap idxer)
    oids_in = np.array([10, 20, 30, 40, 77], dtype=np.int64)
    src = np.array([10, 20, 30], dtype=np.int64)
    dst = np.array([20, 30, 77], dtype=np.int64)
    g = eng.load_edges(src, dst, directed=True, vertex_oids=oids_in)
    oids, vals = sorted_by_oid(eng.bfs(g, 10))
    assert list(oids) == [10, 20, 30, 40, 77]
    assert list(vals) == [0, 1, 2, INT64_MAX, 3]


# ---- extras: BC / k-core / core decomposition / k-clique -------------------

def test_bc_undirected(eng):
    src, dst, _ = random_graph(num_v=800, num_e=4000, seed=51)
    g = eng.load_edges(src, dst, directed=False, num_vertices=800)
    r = eng.bc(g, 5)
    order = np.argsort(r["oids"])
    vals = r["values"][order]
    sig = r["path_num"][order]
    dep = r["depth"][order]
    e_delta, e_sigma, e_depth = bc_oracle(800, src, dst, 5, directed=False)
    reach = e_depth < 1e300
    assert np.array_equal(dep[reach], e_depth[reach].astype(np.int64))
    assert np.allclose(sig[reach], e_sigma[reach])
    assert np.allclose(vals[reach], e_delta[reach], rtol=1e-9, atol=1e-9)


def test_bc_directed(eng):
    src, dst, _ = random_graph(num_v=600, num_e=3000, seed=53)
    g = eng.load_edges(src, dst, directed=True, num_vertices=600)
    r = eng.bc(g, 2)
    order = np.argsort(r["oids"])
    e_delta, e_sigma, e_depth = bc_oracle(600, src, dst, 2, directed=True)
    reach = e_depth < 1e300
    assert np.allclose(r["values"][order][reach], e_delta[reach], rtol=1e-9)


def test_kcore(eng):
    src, dst, _ = random_graph(num_v=500, num_e=3000, seed=57)
    g = eng.load_edges(src, dst, directed=False, num_vertices=500)
    for k in (2, 4, 7):
        r = eng.kcore(g, k)
        order = np.argsort(r["oids"])
        assert np.array_equal(r["values"][order],
                              kcore_oracle(500, src, dst, k)), k


def test_core_decomposition(eng):
    src, dst, _ = random_graph(num_v=400, num_e=2500, seed=59)
    g = eng.load_edges(src, dst, directed=False, num_vertices=400)
    r = eng.core_decomposition(g)
    order = np.argsort(r["oids"])
    assert np.array_equal(r["values"][order], coreness_oracle(400, src, dst))


def test_kclique(eng):
    src, dst, _ = random_graph(num_v=120, num_e=1400, seed=61)
    g = eng.load_edges(src, dst, directed=False, num_vertices=120)
    for k in (3, 4, 5):
        r = eng.kclique(g, k)
        assert r["clique_count"] == kclique_oracle(120, src, dst, k), k


def test_sssp_auto(eng):
    # auto-app (sync-buffer) variant must agree with the explicit app
    src, dst, w = random_graph(num_v=400, num_e=2500, seed=67, weighted=True)
    g = eng.load_edges(src, dst, weights=w, directed=True, num_vertices=400)
    r_auto = eng.sssp_auto(g, 3)
    r_par = eng.sssp(g, 3)
    o1, o2 = np.argsort(r_auto["oids"]), np.argsort(r_par["oids"])
    a, b = r_auto["values"][o1], r_par["values"][o2]
    finite = b < 1e300
    assert np.allclose(a[finite], b[finite], rtol=1e-9)
    assert (a[~finite] > 1e300).all()


def test_pagerank_convergence_mode(eng):
    # pagerank_local parity: tol-based early stop reaches the fixpoint
    src, dst, _ = random_graph(num_v=300, num_e=2000, seed=69)
    g = eng.load_edges(src, dst, directed=True, num_vertices=300)
    r_conv = eng.pagerank(g, 0.85, 1000, tol=1e-12)
    r_long = eng.pagerank(g, 0.85, 200)
    assert r_conv["rounds"] < 1000  # stopped early
    o1, o2 = np.argsort(r_conv["oids"]), np.argsort(r_long["oids"])
    assert np.allclose(r_conv["values"][o1], r_long["values"][o2],
                       rtol=1e-9)


def test_undirected_self_loop_semantics(eng):
    # reference parity: undirected storage holds self loops twice (both
    # orientations), which flips CDLP's mode tie here
    src = np.array([0, 0, 0], dtype=np.int64)
    dst = np.array([0, 1, 1], dtype=np.int64)
    g = eng.load_edges(src, dst, directed=False, num_vertices=2)
    r = eng.cdlp(g, 1)
    got = r["values"][np.argsort(r["oids"])]
    assert np.array_equal(got, cdlp_oracle(2, src, dst, 1, directed=False))
    rp = eng.pagerank(g, 0.85, 6)
    exp = pagerank_oracle(2, src, dst, 0.85, 6, directed=False)
    assert np.allclose(rp["values"][np.argsort(rp["oids"])], exp, rtol=1e-9)
