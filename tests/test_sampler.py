"""GNN random-walk sampler (reference examples/gnn_sampler): walk validity,
determinism, strategy semantics."""
import numpy as np
import pytest

import grapehip


@pytest.fixture(scope="module")
def eng():
    return grapehip.Engine(rank=0, world=1, master_port=29661)


def make_graph(eng, num_v=300, num_e=2400, seed=7, weighted=True,
               directed=True):
    rng = np.random.default_rng(seed)
    src = rng.integers(0, num_v, size=num_e, dtype=np.int64)
    dst = rng.integers(0, num_v, size=num_e, dtype=np.int64)
    keep = src != dst
    src, dst = src[keep], dst[keep]
    w = rng.random(len(src), dtype=np.float32) * 9 + 1 if weighted else None
    g = eng.load_edges(src, dst, weights=w, directed=directed,
                       num_vertices=num_v)
    adj = {}
    for s, d in zip(src, dst):
        adj.setdefault(int(s), set()).add(int(d))
        if not directed:
            adj.setdefault(int(d), set()).add(int(s))
    return g, adj, (src, dst, w)


def test_walks_are_paths(eng):
    g, adj, _ = make_graph(eng)
    starts = np.arange(50, dtype=np.int64)
    r = eng.sample(g, starts, hops=3, strategy="random", seed=11)
    assert len(r["walk_ids"]) == 50
    completed = 0
    for wid, path in zip(r["walk_ids"], r["paths"]):
        assert path[0] == starts[wid]
        if path[-1] >= 0:
            completed += 1
        for h in range(3):
            a, b = int(path[h]), int(path[h + 1])
            if b == -1:
                # a dead end must really be a dead end
                assert a == -1 or not adj.get(a), (wid, h, a)
                continue
            assert b in adj.get(a, set()), (wid, h, a, b)
    # dense random graph: nearly every walk must run its full length
    assert completed >= 45, completed


def test_deterministic(eng):
    g, _, _ = make_graph(eng, seed=9)
    starts = np.arange(30, dtype=np.int64)
    r1 = eng.sample(g, starts, hops=4, strategy="edge_weight", seed=42)
    r2 = eng.sample(g, starts, hops=4, strategy="edge_weight", seed=42)
    assert np.array_equal(r1["paths"], r2["paths"])
    r3 = eng.sample(g, starts, hops=4, strategy="edge_weight", seed=43)
    assert not np.array_equal(r1["paths"], r3["paths"])


def test_top_k_strategy(eng):
    # star: vertex 0 -> 1..20 with known weights; top_k=3 must only ever
    # pick the 3 heaviest targets
    src = np.zeros(20, dtype=np.int64)
    dst = np.arange(1, 21, dtype=np.int64)
    w = np.arange(1, 21, dtype=np.float32)
    g = eng.load_edges(src, dst, weights=w, directed=True, num_vertices=21)
    hits = set()
    for seed in range(40):
        r = eng.sample(g, np.zeros(1, dtype=np.int64), hops=1,
                       strategy="top_k", top_k=3, seed=seed)
        hits.add(int(r["paths"][0][1]))
    assert hits <= {18, 19, 20}
    assert len(hits) > 1  # uniform over the top 3, not argmax only


def test_edge_weight_bias(eng):
    # 0 -> 1 (weight 99), 0 -> 2 (weight 1): overwhelmingly picks 1
    src = np.array([0, 0], dtype=np.int64)
    dst = np.array([1, 2], dtype=np.int64)
    w = np.array([99.0, 1.0], dtype=np.float32)
    g = eng.load_edges(src, dst, weights=w, directed=True, num_vertices=3)
    picks = [int(eng.sample(g, np.zeros(1, dtype=np.int64), hops=1,
                            strategy="edge_weight", seed=s)["paths"][0][1])
             for s in range(50)]
    assert picks.count(1) > 40
