"""Error-path and malformed-input behavior. The loader drops edges whose
endpoints no rank resolves (reference parity: basic loaders skip records
absent from the vertex map) and apps treat an unresolvable source as
globally unreached; genuinely inconsistent inputs raise."""
import numpy as np
import pytest

import grapehip

INF = np.iinfo(np.int64).max


@pytest.fixture(scope="module")
def eng():
    return grapehip.Engine(rank=0, world=1, master_port=29643)


def test_weights_length_mismatch_raises(eng):
    src = np.array([0, 1], dtype=np.int64)
    dst = np.array([1, 2], dtype=np.int64)
    with pytest.raises(RuntimeError, match="weights size mismatch"):
        eng.load_edges(src, dst, weights=np.ones(1, dtype=np.float32),
                       directed=False, num_vertices=3)


def test_src_dst_length_mismatch_raises(eng):
    src = np.array([0, 1], dtype=np.int64)
    dst = np.array([1], dtype=np.int64)
    with pytest.raises(RuntimeError, match="size mismatch"):
        eng.load_edges(src, dst, directed=False, num_vertices=3)


def test_identity_requires_num_vertices(eng):
    e = np.array([0], dtype=np.int64)
    with pytest.raises(RuntimeError, match="num_vertices"):
        eng.load_edges(e, e, directed=False)


def test_out_of_range_vid_edge_dropped(eng):
    src = np.array([0, 5], dtype=np.int64)
    dst = np.array([1, 0], dtype=np.int64)
    g = eng.load_edges(src, dst, directed=False, num_vertices=3)
    assert g.num_edges == 2  # only 0-1, stored both directions


def test_unknown_oid_edge_dropped(eng):
    oids = np.array([10, 20, 30], dtype=np.int64)
    src = np.array([10, 20, 99], dtype=np.int64)
    dst = np.array([20, 30, 10], dtype=np.int64)
    g = eng.load_edges(src, dst, directed=False, vertex_oids=oids)
    assert g.num_edges == 4  # two surviving edges, both directions


def test_out_of_range_source_unreached(eng):
    src = np.array([0, 1], dtype=np.int64)
    dst = np.array([1, 2], dtype=np.int64)
    g = eng.load_edges(src, dst, directed=False, num_vertices=3)
    for bad in (99, -1):
        r = eng.bfs(g, bad)
        assert (r["values"] == INF).all()


def test_unknown_oid_source_unreached(eng):
    oids = np.array([10, 20], dtype=np.int64)
    g = eng.load_edges(np.array([10], dtype=np.int64),
                       np.array([20], dtype=np.int64),
                       directed=False, vertex_oids=oids)
    r = eng.sssp(g, 77)
    assert (r["values"] == np.finfo(np.float64).max).all()


def test_byte_slice_partition_exact(tmp_path):
    """Every line lands on exactly one rank for any world size, including
    worlds where a slice is smaller than one line (regression: a negative
    read count returned the rest of the file, duplicating edges)."""
    from grapehip.io import _byte_slice
    lines = [f"{i} {i*7} 1.25\n" for i in range(17)]
    p = tmp_path / "t.e"
    p.write_text("".join(lines))
    for world in (1, 2, 3, 5, 8, 16, 64):
        got = []
        for r in range(world):
            got.append(_byte_slice(str(p), r, world).decode())
        assert "".join(got) == "".join(lines), world
        for g in got:  # each slice is whole lines
            assert g == "" or g.endswith("\n")


def test_read_ldbc_edges_many_ranks(tmp_path):
    import grapehip.io as io
    lines = [f"{i} {i+1} 0.5\n" for i in range(10)]
    p = tmp_path / "w.e"
    p.write_text("".join(lines))
    seen = []
    for r in range(32):
        src, dst, w = io.read_ldbc_edges(str(p), weighted=True, rank=r,
                                         world=32)
        assert len(src) == len(dst) == len(w)
        seen.extend(src.tolist())
    assert sorted(seen) == list(range(10))


def test_engine_thread_count_restart(tmp_path):
    """Recreating engines with different thread counts in one process
    restarts the shared pool; work after a restart must be correct
    (regression: restarted workers started at epoch 0 and could call a
    null task on a spurious wakeup)."""
    src = np.array([0, 1, 2, 3], dtype=np.int64)
    dst = np.array([1, 2, 3, 0], dtype=np.int64)
    for i, nt in enumerate((2, 4, 1, 3)):
        eng2 = grapehip.Engine(rank=0, world=1, master_port=29770 + i,
                               n_threads=nt)
        g = eng2.load_edges(src, dst, directed=False, num_vertices=4)
        r = eng2.bfs(g, 0)
        assert sorted(r["values"].tolist()) == [0, 1, 1, 2]
