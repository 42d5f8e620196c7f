"""Fragment checkpoint (serialize/deserialize) round-trips: reloaded graphs
must produce identical app results without rerunning the build pipeline.
Reference parity: immutable_edgecut_fragment.h:508-584 + --serialize /
--deserialize (ev_fragment_loader.h:75-93)."""
import numpy as np
import pytest

import grapehip


@pytest.fixture(scope="module")
def eng():
    return grapehip.Engine(rank=0, world=1, master_port=29641)


def random_graph(num_v=2000, num_e=12000, seed=71):
    rng = np.random.default_rng(seed)
    src = rng.integers(0, num_v, size=num_e, dtype=np.int64)
    dst = rng.integers(0, num_v, size=num_e, dtype=np.int64)
    keep = src != dst
    w = rng.random(keep.sum(), dtype=np.float32) * 9 + 1
    return src[keep], dst[keep], w


def test_roundtrip_identity(eng, tmp_path):
    src, dst, w = random_graph()
    g = eng.load_edges(src, dst, weights=w, directed=False,
                       num_vertices=2000)
    before_bfs = eng.bfs(g, 7)
    before_sssp = eng.sssp(g, 7)
    eng.save_graph(g, str(tmp_path))
    g2 = eng.load_serialized(str(tmp_path))
    assert g2.num_vertices == g.num_vertices
    assert g2.num_edges == g.num_edges
    after_bfs = eng.bfs(g2, 7)
    after_sssp = eng.sssp(g2, 7)
    assert np.array_equal(before_bfs["values"], after_bfs["values"])
    assert np.array_equal(before_sssp["values"], after_sssp["values"])


def test_roundtrip_hashmap_oids(eng, tmp_path):
    src, dst, w = random_graph(num_v=500, num_e=3000, seed=73)
    oids = np.arange(500, dtype=np.int64) * 7 + 3
    g = eng.load_edges(oids[src], oids[dst], weights=w, directed=True,
                       vertex_oids=oids, build_in_csr=True)
    before = eng.pagerank(g, 0.85, 5)
    eng.save_graph(g, str(tmp_path))
    g2 = eng.load_serialized(str(tmp_path))
    after = eng.pagerank(g2, 0.85, 5)
    o1 = np.argsort(before["oids"])
    o2 = np.argsort(after["oids"])
    assert np.array_equal(before["oids"][o1], after["oids"][o2])
    assert np.allclose(before["values"][o1], after["values"][o2], rtol=1e-14)


def test_missing_checkpoint_raises(eng, tmp_path):
    with pytest.raises(RuntimeError, match="cannot open"):
        eng.load_serialized(str(tmp_path / "nope"))


def test_bad_magic_raises(eng, tmp_path):
    p = tmp_path / "frag_0.s"
    p.write_bytes(b"\x00" * 64)
    with pytest.raises(RuntimeError, match="bad magic"):
        eng.load_serialized(str(tmp_path))


def test_truncated_checkpoint_raises(eng, tmp_path):
    src, dst, w = random_graph(num_v=300, num_e=2000, seed=77)
    g = eng.load_edges(src, dst, weights=w, directed=False,
                       num_vertices=300)
    eng.save_graph(g, str(tmp_path))
    p = tmp_path / "frag_0.s"
    data = p.read_bytes()
    p.write_bytes(data[: len(data) // 2])
    with pytest.raises(RuntimeError, match="short read"):
        eng.load_serialized(str(tmp_path))


def test_roundtrip_mph_oids(eng, tmp_path):
    src, dst, w = random_graph(num_v=400, num_e=2500, seed=79)
    oids = np.arange(400, dtype=np.int64) * 11 + 5
    g = eng.load_edges(oids[src], oids[dst], weights=w, directed=False,
                       vertex_oids=oids, idxer="mph")
    before = eng.sssp(g, oids[3])
    eng.save_graph(g, str(tmp_path))
    g2 = eng.load_serialized(str(tmp_path))
    after = eng.sssp(g2, oids[3])
    o1, o2 = np.argsort(before["oids"]), np.argsort(after["oids"])
    assert np.array_equal(before["oids"][o1], after["oids"][o2])
    assert np.array_equal(before["values"][o1], after["values"][o2])
