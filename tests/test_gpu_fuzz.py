"""GPU seed-sweep fuzz: many random graph shapes x all six kernels vs the
NumPy oracles, plus the reference's LB-strategy sweep (misc/
cuda_app_tests.sh runs every app under --lb none|cm|strict — grapehip's
GRAPEHIP_LB env selects the expansion kernel)."""
import json
import os
import subprocess
import sys
from pathlib import Path

import numpy as np
import pytest

import grapehip
from oracles import (bfs_oracle, cdlp_oracle, lcc_oracle, pagerank_oracle,
                     sssp_oracle, wcc_oracle)

REPO = Path(__file__).resolve().parent.parent
pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    return grapehip.Engine(rank=0, world=1, master_port=29771, gpu=True)


@pytest.mark.parametrize("seed", [101, 202, 303, 404, 505])
def test_gpu_fuzz_suite(eng, seed):
    rng = np.random.default_rng(seed)
    nv = int(rng.integers(50, 20000))
    ne = int(rng.integers(nv, nv * 12))
    src = rng.integers(0, nv, ne).astype(np.int64)
    dst = rng.integers(0, nv, ne).astype(np.int64)
    w = (rng.random(ne, dtype=np.float32) * 9 + 1)
    g = eng.load_edges(src, dst, weights=w, directed=False,
                       num_vertices=nv)
    source = int(rng.integers(0, nv))

    r = eng.bfs(g, source)
    order = np.argsort(r["oids"])
    assert np.array_equal(r["values"][order],
                          bfs_oracle(nv, src, dst, source, directed=False))

    r = eng.sssp(g, source)
    exp = sssp_oracle(nv, src, dst, w, source, directed=False)
    got = r["values"][np.argsort(r["oids"])]
    finite = exp < 1e300
    assert np.allclose(got[finite], exp[finite], rtol=1e-4)
    assert (got[~finite] > 1e300).all()

    r = eng.pagerank(g, 0.85, 6)
    assert np.allclose(r["values"][np.argsort(r["oids"])],
                       pagerank_oracle(nv, src, dst, 0.85, 6,
                                       directed=False), rtol=3e-5,
                       atol=1e-12)

    r = eng.wcc(g)
    got = r["values"][np.argsort(r["oids"])]
    exp = wcc_oracle(nv, src, dst)
    fwd, bwd = {}, {}
    for a, b in zip(got, exp):
        assert fwd.setdefault(a, b) == b
        assert bwd.setdefault(b, a) == a

    r = eng.cdlp(g, 5)
    assert np.array_equal(r["values"][np.argsort(r["oids"])],
                          cdlp_oracle(nv, src, dst, 5, directed=False))

    r = eng.lcc(g)
    assert np.allclose(r["values"][np.argsort(r["oids"])],
                       lcc_oracle(nv, src, dst, directed=False),
                       rtol=1e-12)


LB_WORKER = r'''
import json, os, sys
sys.path.insert(0, os.environ["GRAPEHIP_REPO"])
import numpy as np
import grapehip
rng = np.random.default_rng(907)
nv, ne = 8000, 90000
src = rng.integers(0, nv, ne).astype(np.int64)
dst = rng.integers(0, nv, ne).astype(np.int64)
w = (rng.random(ne, dtype=np.float32) * 9 + 1)
eng = grapehip.Engine(rank=0, world=1, master_port=29781, gpu=True)
g = eng.load_edges(src, dst, weights=w, directed=False, num_vertices=nv)
b = eng.bfs(g, 3)
s = eng.sssp(g, 3)
out = {"bfs": b["values"][np.argsort(b["oids"])].tolist(),
       "sssp": s["values"][np.argsort(s["oids"])].tolist()}
json.dump(out, open(os.environ["GRAPEHIP_OUT"], "w"))
'''


def test_lb_strategies_agree(tmp_path):
    # np x lb matrix (reference cuda_app_tests.sh): every LB strategy must
    # produce identical results
    results = {}
    for lb in ("cm", "strict", "none"):
        out = tmp_path / ("lb_%s.json" % lb)
        env = dict(os.environ, GRAPEHIP_REPO=str(REPO),
                   GRAPEHIP_OUT=str(out), GRAPEHIP_LB=lb)
        r = subprocess.run([sys.executable, "-c", LB_WORKER], env=env,
                           capture_output=True, text=True, timeout=300)
        assert r.returncode == 0, (lb, r.stdout + r.stderr)
        results[lb] = json.load(open(out))
    for lb in ("strict", "none"):
        assert results[lb]["bfs"] == results["cm"]["bfs"], lb
        assert np.allclose(results[lb]["sssp"], results["cm"]["sssp"],
                           rtol=1e-6), lb
