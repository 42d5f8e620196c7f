"""Parity check against LDBC golden outputs of the reference's bundled
p2p-31 dataset (runs only where /root/reference is mounted; skipped
elsewhere, e.g. on GPU boxes). Validates BFS/SSSP/PR/WCC/CDLP/LCC end to end
on a real LDBC-format input including the .v/.e loaders."""
import os
from pathlib import Path

import numpy as np
import pytest

import grapehip
from grapehip.io import read_ldbc_edges, read_ldbc_vertices

DATASET = Path("/root/reference/dataset")

pytestmark = pytest.mark.skipif(
    not DATASET.exists(), reason="reference dataset not available")


def load_golden(name):
    path = DATASET / name
    oids, vals = [], []
    with open(path) as f:
        for line in f:
            a, b = line.split()
            oids.append(int(a))
            vals.append(b)
    return np.array(oids), vals


@pytest.fixture(scope="module")
def eng():
    return grapehip.Engine(rank=0, world=1, master_port=29601)


@pytest.fixture(scope="module")
def graphs(eng):
    oids = read_ldbc_vertices(str(DATASET / "p2p-31.v"))
    src, dst, w = read_ldbc_edges(str(DATASET / "p2p-31.e"), weighted=True)
    out = {}
    out["undirected"] = eng.load_edges(src, dst, weights=w, directed=False,
                                       vertex_oids=oids)
    out["arrays"] = (oids, src, dst, w)
    return out


def result_map(res):
    return dict(zip(res["oids"].tolist(), res["values"].tolist()))


def test_bfs_golden(eng, graphs):
    # golden p2p-31-BFS was produced with source 6 (depth 0 at oid 6)
    res = eng.bfs(graphs["undirected"], 6)
    got = result_map(res)
    oids, vals = load_golden("p2p-31-BFS")
    for o, v in zip(oids, vals):
        assert got[o] == int(v), (o, got[o], v)


def test_sssp_golden(eng, graphs):
    res = eng.sssp(graphs["undirected"], 6)
    got = result_map(res)
    oids, vals = load_golden("p2p-31-SSSP")
    for o, v in zip(oids, vals):
        assert abs(got[o] - float(v)) <= 1e-9 * max(1.0, abs(float(v))), o


def test_pagerank_golden(eng, graphs):
    res = eng.pagerank(graphs["undirected"], 0.85, 10)
    got = result_map(res)
    oids, vals = load_golden("p2p-31-PR")
    for o, v in zip(oids, vals):
        assert abs(got[o] - float(v)) <= 1e-6 * max(1e-12, abs(float(v))), o


def test_wcc_golden(eng, graphs):
    res = eng.wcc(graphs["undirected"])
    got = result_map(res)
    oids, vals = load_golden("p2p-31-WCC")
    # WCC labels must match up to isomorphism
    fwd, bwd = {}, {}
    for o, v in zip(oids, vals):
        mine, ref = got[o], int(v)
        assert fwd.setdefault(mine, ref) == ref, o
        assert bwd.setdefault(ref, mine) == mine, o


def test_cdlp_golden(eng, graphs):
    res = eng.cdlp(graphs["undirected"], 10)
    got = result_map(res)
    oids, vals = load_golden("p2p-31-CDLP")
    for o, v in zip(oids, vals):
        assert got[o] == int(v), (o, got[o], v)


def test_lcc_golden(eng, graphs):
    res = eng.lcc(graphs["undirected"])
    got = result_map(res)
    oids, vals = load_golden("p2p-31-LCC")
    for o, v in zip(oids, vals):
        assert abs(got[o] - float(v)) <= 1e-9 * max(1.0, abs(float(v))), o
