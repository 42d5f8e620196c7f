"""Vertex-cut fragment + gather-scatter PageRank (reference --vc path:
immutable_vertexcut_fragment.h + gather_scatter_message_manager.h +
pagerank_vc.h)."""
import numpy as np
import pytest

import grapehip
from oracles import pagerank_oracle


@pytest.fixture(scope="module")
def eng():
    return grapehip.Engine(rank=0, world=1, master_port=29671)


def test_pagerank_vc_matches_oracle(eng):
    rng = np.random.default_rng(103)
    nv, ne = 800, 5000
    src = rng.integers(0, nv, ne).astype(np.int64)
    dst = rng.integers(0, nv, ne).astype(np.int64)
    g = eng.load_vertexcut(src, dst, num_vertices=nv)
    assert g.num_edges == ne
    r = eng.pagerank_vc(g, 0.85, 10)
    order = np.argsort(r["oids"])
    expect = pagerank_oracle(nv, src, dst, 0.85, 10, directed=True)
    assert np.allclose(r["values"][order], expect, rtol=1e-9)


def test_pagerank_vc_matches_edgecut(eng):
    rng = np.random.default_rng(107)
    nv, ne = 1200, 9000
    src = rng.integers(0, nv, ne).astype(np.int64)
    dst = rng.integers(0, nv, ne).astype(np.int64)
    gvc = eng.load_vertexcut(src, dst, num_vertices=nv)
    gec = eng.load_edges(src, dst, directed=True, num_vertices=nv,
                         build_in_csr=True)
    rvc = eng.pagerank_vc(gvc, 0.85, 8)
    rec = eng.pagerank(gec, 0.85, 8)
    o1, o2 = np.argsort(rvc["oids"]), np.argsort(rec["oids"])
    assert np.allclose(rvc["values"][o1], rec["values"][o2], rtol=1e-9)
