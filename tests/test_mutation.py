"""Graph mutation (add/remove edges and vertices): results on the mutated
graph must equal a fresh build of the mutated edge list. Reference parity:
LoadGraphAndMutate + MutableEdgecutFragment::Mutate + mutation app tests
(misc/app_tests.sh:114-165)."""
import numpy as np
import pytest

import grapehip
from oracles import bfs_oracle, pagerank_oracle, sssp_oracle, wcc_oracle


@pytest.fixture(scope="module")
def eng():
    return grapehip.Engine(rank=0, world=1, master_port=29651)


def base_graph(seed=81, num_v=600, num_e=3000):
    rng = np.random.default_rng(seed)
    src = rng.integers(0, num_v, size=num_e, dtype=np.int64)
    dst = rng.integers(0, num_v, size=num_e, dtype=np.int64)
    keep = src != dst
    return src[keep], dst[keep]


def test_add_remove_edges(eng):
    src, dst = base_graph()
    nv = 600
    g = eng.load_edges(src, dst, directed=False, num_vertices=nv)
    # remove the first 200 edges, add 300 new ones
    rm_s, rm_d = src[:200].copy(), dst[:200].copy()
    rng = np.random.default_rng(83)
    ad_s = rng.integers(0, nv, 300).astype(np.int64)
    ad_d = rng.integers(0, nv, 300).astype(np.int64)
    k = ad_s != ad_d
    ad_s, ad_d = ad_s[k], ad_d[k]
    g2 = eng.mutate_graph(g, add_src=ad_s, add_dst=ad_d,
                          remove_src=rm_s, remove_dst=rm_d,
                          remove_vertices=np.array([], dtype=np.int64))
    # expected edge multiset: drop ALL copies of removed pairs, then add
    pairs = set()
    for a, b in zip(rm_s, rm_d):
        pairs.add((a, b))
        pairs.add((b, a))
    keep = np.array([(a, b) not in pairs for a, b in zip(src, dst)])
    e_src = np.concatenate([src[keep], ad_s])
    e_dst = np.concatenate([dst[keep], ad_d])
    r = eng.bfs(g2, 5)
    order = np.argsort(r["oids"])
    assert np.array_equal(r["values"][order],
                          bfs_oracle(nv, e_src, e_dst, 5, directed=False))
    r = eng.wcc(g2)
    assert np.array_equal(r["values"][np.argsort(r["oids"])],
                          wcc_oracle(nv, e_src, e_dst))


def test_remove_vertices(eng):
    src, dst = base_graph(seed=91)
    nv = 600
    g = eng.load_edges(src, dst, directed=True, num_vertices=nv,
                       build_in_csr=True)
    dead = np.array([3, 77, 200, 401], dtype=np.int64)
    g2 = eng.mutate_graph(g, add_src=np.array([], dtype=np.int64),
                          add_dst=np.array([], dtype=np.int64),
                          remove_src=np.array([], dtype=np.int64),
                          remove_dst=np.array([], dtype=np.int64),
                          remove_vertices=dead)
    deadset = set(dead.tolist())
    keep = np.array([s not in deadset and d not in deadset
                     for s, d in zip(src, dst)])
    r = eng.pagerank(g2, 0.85, 10)
    order = np.argsort(r["oids"])
    expect = pagerank_oracle(nv, src[keep], dst[keep], 0.85, 10,
                             directed=True)
    assert np.allclose(r["values"][order], expect, rtol=1e-9)


def test_mutate_weighted(eng):
    src, dst = base_graph(seed=95)
    nv = 600
    rng = np.random.default_rng(97)
    w = rng.random(len(src), dtype=np.float32) * 9 + 1
    g = eng.load_edges(src, dst, weights=w, directed=True, num_vertices=nv)
    ad_s = np.array([1, 2, 3], dtype=np.int64)
    ad_d = np.array([4, 5, 6], dtype=np.int64)
    ad_w = np.array([0.5, 0.25, 0.125], dtype=np.float32)
    g2 = eng.mutate_graph(g, add_src=ad_s, add_dst=ad_d, add_weights=ad_w,
                          remove_src=np.array([], dtype=np.int64),
                          remove_dst=np.array([], dtype=np.int64),
                          remove_vertices=np.array([], dtype=np.int64))
    e_src = np.concatenate([src, ad_s])
    e_dst = np.concatenate([dst, ad_d])
    e_w = np.concatenate([w, ad_w])
    r = eng.sssp(g2, 1)
    order = np.argsort(r["oids"])
    expect = sssp_oracle(nv, e_src, e_dst, e_w, 1, directed=True)
    finite = expect < 1e300
    assert np.allclose(r["values"][order][finite], expect[finite], rtol=1e-6)


def test_repeated_deltas_accumulate(eng):
    # in-place path: successive deltas reuse slack, grow rows into the
    # end arena, and add new outer vertices; final state must equal a
    # fresh build of the accumulated edge list
    src, dst = base_graph(seed=101, num_v=400, num_e=1200)
    nv = 400
    g = eng.load_edges(src, dst, directed=False, num_vertices=nv)
    cur_s, cur_d = src.copy(), dst.copy()
    rng = np.random.default_rng(103)
    empty = np.array([], dtype=np.int64)
    for step in range(5):
        ad_s = rng.integers(0, nv, 150).astype(np.int64)
        ad_d = rng.integers(0, nv, 150).astype(np.int64)
        k = ad_s != ad_d
        ad_s, ad_d = ad_s[k], ad_d[k]
        rm_s, rm_d = cur_s[:20].copy(), cur_d[:20].copy()
        g = eng.mutate_graph(g, add_src=ad_s, add_dst=ad_d,
                             remove_src=rm_s, remove_dst=rm_d,
                             remove_vertices=empty)
        pairs = set()
        for a, b in zip(rm_s, rm_d):
            pairs.add((a, b))
            pairs.add((b, a))
        keep = np.array([(a, b) not in pairs
                         for a, b in zip(cur_s, cur_d)])
        cur_s = np.concatenate([cur_s[keep], ad_s])
        cur_d = np.concatenate([cur_d[keep], ad_d])
    r = eng.bfs(g, 5)
    order = np.argsort(r["oids"])
    assert np.array_equal(r["values"][order],
                          bfs_oracle(nv, cur_s, cur_d, 5, directed=False))
    r = eng.wcc(g)
    assert np.array_equal(r["values"][np.argsort(r["oids"])],
                          wcc_oracle(nv, cur_s, cur_d))


def test_serialize_after_mutate(eng, tmp_path):
    # mutable-mode slack must compact into a canonical checkpoint
    src, dst = base_graph(seed=107)
    nv = 600
    g = eng.load_edges(src, dst, directed=False, num_vertices=nv)
    rng = np.random.default_rng(109)
    ad_s = rng.integers(0, nv, 200).astype(np.int64)
    ad_d = rng.integers(0, nv, 200).astype(np.int64)
    k = ad_s != ad_d
    ad_s, ad_d = ad_s[k], ad_d[k]
    empty = np.array([], dtype=np.int64)
    g = eng.mutate_graph(g, add_src=ad_s, add_dst=ad_d, remove_src=empty,
                         remove_dst=empty, remove_vertices=empty)
    ck = tmp_path / "ck"
    ck.mkdir()
    eng.save_graph(g, str(ck))
    g2 = eng.load_serialized(str(ck))
    e_src = np.concatenate([src, ad_s])
    e_dst = np.concatenate([dst, ad_d])
    r = eng.bfs(g2, 5)
    order = np.argsort(r["oids"])
    assert np.array_equal(r["values"][order],
                          bfs_oracle(nv, e_src, e_dst, 5, directed=False))


def test_mutate_after_deserialize(eng, tmp_path):
    # in-place delta applied to a checkpoint-restored fragment (FromParts
    # state must enter mutable mode cleanly), then re-checkpointed
    src, dst = base_graph(seed=113)
    nv = 600
    g = eng.load_edges(src, dst, directed=False, num_vertices=nv)
    ck1 = tmp_path / "ck1"
    ck1.mkdir()
    eng.save_graph(g, str(ck1))
    g2 = eng.load_serialized(str(ck1))
    rng = np.random.default_rng(127)
    ad_s = rng.integers(0, nv, 120).astype(np.int64)
    ad_d = rng.integers(0, nv, 120).astype(np.int64)
    k = ad_s != ad_d
    ad_s, ad_d = ad_s[k], ad_d[k]
    empty = np.array([], dtype=np.int64)
    g2 = eng.mutate_graph(g2, add_src=ad_s, add_dst=ad_d,
                          remove_src=src[:50].copy(),
                          remove_dst=dst[:50].copy(),
                          remove_vertices=empty)
    ck2 = tmp_path / "ck2"
    ck2.mkdir()
    eng.save_graph(g2, str(ck2))
    g3 = eng.load_serialized(str(ck2))
    pairs = set()
    for a, b in zip(src[:50], dst[:50]):
        pairs.add((a, b))
        pairs.add((b, a))
    keep = np.array([(a, b) not in pairs for a, b in zip(src, dst)])
    e_s = np.concatenate([src[keep], ad_s])
    e_d = np.concatenate([dst[keep], ad_d])
    for gx in (g2, g3):
        r = eng.bfs(gx, 5)
        order = np.argsort(r["oids"])
        assert np.array_equal(
            r["values"][order], bfs_oracle(nv, e_s, e_d, 5, directed=False))
