"""CLI driver (grapehip.run_app) end-to-end on the reference's bundled
p2p-31 dataset, verified against the LDBC golden outputs like the
reference's misc/app_tests.sh ExactVerify/EpsVerify."""
import os
import subprocess
import sys
from pathlib import Path

import numpy as np
import pytest

REPO = Path(__file__).resolve().parent.parent
DATASET = Path("/root/reference/dataset")

pytestmark = pytest.mark.skipif(
    not DATASET.exists(), reason="reference dataset not available")


def run_cli(tmp_path, *extra):
    out = tmp_path / "out"
    cmd = [sys.executable, "-m", "grapehip.run_app",
           "--efile", str(DATASET / "p2p-31.e"),
           "--vfile", str(DATASET / "p2p-31.v"),
           "--out_prefix", str(out), *extra]
    env = dict(os.environ, PYTHONPATH=str(REPO))
    r = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    return out / "result_frag_0", r.stdout


def load_pairs(path):
    oids, vals = [], []
    for line in open(path):
        a, b = line.split()
        oids.append(int(a))
        vals.append(b)
    return np.array(oids), vals


def test_cli_bfs_golden(tmp_path):
    res, out = run_cli(tmp_path, "--application", "bfs", "--bfs_source", "6")
    assert "run algorithm" in out
    g_oids, g_vals = load_pairs(DATASET / "p2p-31-BFS")
    oids, vals = load_pairs(res)
    order, gorder = np.argsort(oids), np.argsort(g_oids)
    assert np.array_equal(oids[order], g_oids[gorder])
    got = np.array([int(v) for v in vals], dtype=np.int64)[order]
    exp = np.array([int(v) for v in g_vals], dtype=np.int64)[gorder]
    assert np.array_equal(got, exp)


def test_cli_sssp_golden(tmp_path):
    res, _ = run_cli(tmp_path, "--application", "sssp", "--sssp_source", "6")
    g_oids, g_vals = load_pairs(DATASET / "p2p-31-SSSP")
    oids, vals = load_pairs(res)
    order, gorder = np.argsort(oids), np.argsort(g_oids)
    got = np.array([float(v) for v in vals])[order]
    exp = np.array([float(v) for v in g_vals])[gorder]
    finite = exp < 1e300
    assert np.allclose(got[finite], exp[finite], rtol=1e-5)
    assert (got[~finite] > 1e300).all()


def test_cli_serialize_roundtrip(tmp_path):
    ser = tmp_path / "ckpt"
    res1, _ = run_cli(tmp_path, "--application", "wcc",
                      "--serialize", "--serialization_prefix", str(ser))
    out2 = tmp_path / "out2"
    cmd = [sys.executable, "-m", "grapehip.run_app",
           "--application", "wcc", "--deserialize",
           "--serialization_prefix", str(ser),
           "--out_prefix", str(out2)]
    env = dict(os.environ, PYTHONPATH=str(REPO))
    r = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                       text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    a = open(res1).read()
    b = open(out2 / "result_frag_0").read()
    assert a == b


@pytest.mark.parametrize("world", [2, 4])
def test_cli_multirank_bfs_golden(tmp_path, world):
    # torchrun-style env; each rank partial-reads its file slice
    out = tmp_path / "out"
    procs = []
    port = 29755
    for rank in range(world):
        env = dict(os.environ, PYTHONPATH=str(REPO), RANK=str(rank),
                   WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                   MASTER_PORT=str(port - 17))
        cmd = [sys.executable, "-m", "grapehip.run_app",
               "--application", "bfs", "--bfs_source", "6",
               "--efile", str(DATASET / "p2p-31.e"),
               "--vfile", str(DATASET / "p2p-31.v"),
               "--out_prefix", str(out)]
        procs.append(subprocess.Popen(cmd, cwd=REPO, env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    for p in procs:
        o, _ = p.communicate(timeout=300)
        assert p.returncode == 0, o.decode()
    oids, vals = [], []
    for rank in range(world):
        o, v = load_pairs(out / ("result_frag_%d" % rank))
        oids.extend(o)
        vals.extend(int(x) for x in v)
    oids = np.array(oids)
    vals = np.array(vals)
    g_oids, g_vals = load_pairs(DATASET / "p2p-31-BFS")
    order, gorder = np.argsort(oids), np.argsort(g_oids)
    assert np.array_equal(oids[order], g_oids[gorder])
    exp = np.array([int(v) for v in g_vals], dtype=np.int64)[gorder]
    assert np.array_equal(vals[order], exp)


def test_cli_delta_mutation(tmp_path):
    # LoadGraphAndMutate parity: base + delta efiles, verified against a
    # single-file load of the merged edge list
    rng = np.random.default_rng(211)
    nv = 500
    base_s = rng.integers(0, nv, 2500)
    base_d = rng.integers(0, nv, 2500)
    add_s = rng.integers(0, nv, 400)
    add_d = rng.integers(0, nv, 400)
    k1, k2 = base_s != base_d, add_s != add_d
    base_s, base_d = base_s[k1], base_d[k1]
    add_s, add_d = add_s[k2], add_d[k2]
    base = tmp_path / "base.e"
    delta = tmp_path / "delta.e"
    merged = tmp_path / "merged.e"
    np.savetxt(base, np.stack([base_s, base_d], 1), fmt="%d")
    np.savetxt(delta, np.stack([add_s, add_d], 1), fmt="%d")
    np.savetxt(merged, np.stack([np.concatenate([base_s, add_s]),
                                 np.concatenate([base_d, add_d])], 1),
               fmt="%d")
    env = dict(os.environ, PYTHONPATH=str(REPO))

    def run(efile, out, extra=()):
        cmd = [sys.executable, "-m", "grapehip.run_app", "--application",
               "wcc", "--efile", str(efile), "--out_prefix", str(out),
               *extra]
        r = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                           text=True, timeout=300)
        assert r.returncode == 0, r.stdout + r.stderr
        return open(out / "result_frag_0").read()

    a = run(base, tmp_path / "o1", ("--efile_add", str(delta)))
    b = run(merged, tmp_path / "o2")
    assert a == b


def test_cli_bfs_directed_golden(tmp_path):
    res, _ = run_cli(tmp_path, "--application", "bfs", "--bfs_source", "6",
                     "--directed")
    g_oids, g_vals = load_pairs(DATASET / "p2p-31-BFS-directed")
    oids, vals = load_pairs(res)
    order, gorder = np.argsort(oids), np.argsort(g_oids)
    assert np.array_equal(oids[order], g_oids[gorder])
    got = np.array([int(v) for v in vals], dtype=np.int64)[order]
    exp = np.array([int(v) for v in g_vals], dtype=np.int64)[gorder]
    assert np.array_equal(got, exp)


def test_cli_sssp_directed_golden(tmp_path):
    res, _ = run_cli(tmp_path, "--application", "sssp", "--sssp_source",
                     "6", "--directed")
    g_oids, g_vals = load_pairs(DATASET / "p2p-31-SSSP-directed")
    oids, vals = load_pairs(res)
    order, gorder = np.argsort(oids), np.argsort(g_oids)
    got = np.array([float(v) for v in vals])[order]
    exp = np.array([float(v) for v in g_vals])[gorder]
    finite = exp < 1e300
    assert np.allclose(got[finite], exp[finite], rtol=1e-5)
    assert (got[~finite] > 1e300).all()


def test_cli_pagerank_directed_golden(tmp_path):
    """The reference's directed PageRank (pagerank_parallel.h:129-199) PINS
    dangling vertices to `base` and estimates the dangling mass as
    base * n_dangling — a deviation from the LDBC formula that is baked
    into dataset/p2p-31-PR-directed (its values sum to 0.462, not 1).
    grapehip implements the mass-conserving formula on both engines, so
    this test (a) proves the golden matches the reference's pinned
    recurrence to 1e-12 (the difference is understood, not a bug) and
    (b) validates our CLI output against the standard formula."""
    import numpy as np
    from grapehip.io import read_ldbc_edges, read_ldbc_vertices
    oids = read_ldbc_vertices(str(DATASET / "p2p-31.v"))
    src, dst, _ = read_ldbc_edges(str(DATASET / "p2p-31.e"))
    o2i = {int(o): i for i, o in enumerate(oids)}
    s = np.array([o2i[int(x)] for x in src])
    d = np.array([o2i[int(x)] for x in dst])
    nv = len(oids)
    outdeg = np.bincount(s, minlength=nv).astype(float)
    dang = outdeg == 0
    delta, p = 0.85, 1.0 / nv

    gold = np.zeros(nv)
    for line in open(DATASET / "p2p-31-PR-directed"):
        a, b = line.split()
        gold[o2i[int(a)]] = float(b)

    # (a) the reference's pinned-dangling recurrence reproduces its golden
    r = np.full(nv, p)
    dangling_sum = p * dang.sum()
    for _ in range(10):
        base = (1 - delta) / nv + delta * dangling_sum / nv
        contrib = np.where(outdeg > 0, r / np.maximum(outdeg, 1), 0.0)
        acc = np.zeros(nv)
        np.add.at(acc, d, contrib[s])
        r = np.where(dang, base, base + delta * acc)
        dangling_sum = base * dang.sum()
    assert np.allclose(r, gold, rtol=1e-9)

    # (b) our CLI computes the mass-conserving LDBC formula
    res, _ = run_cli(tmp_path, "--application", "pagerank", "--directed")
    o, v = load_pairs(res)
    order = np.argsort(o)
    got = np.array([float(x) for x in v])[order]
    rs = np.full(nv, p)
    for _ in range(10):
        dsum = rs[dang].sum()
        contrib = np.where(outdeg > 0, rs / np.maximum(outdeg, 1), 0.0)
        acc = np.zeros(nv)
        np.add.at(acc, d, contrib[s])
        rs = (1 - delta) / nv + delta * (acc + dsum / nv)
    idx = np.argsort(oids)
    assert np.allclose(got, rs[idx], rtol=1e-9)
    assert abs(got.sum() - 1.0) < 1e-9
