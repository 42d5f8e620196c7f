#!/usr/bin/env python3
"""app_tests.sh-equivalent correctness matrix.

Reference parity: misc/app_tests.sh (CPU: every app x {1,2,4,6,8} ranks x
serialize round-trips, ExactVerify/EpsVerify/WCCVerify against goldens)
and misc/cuda_app_tests.sh (GPU: x load-balancer sweep). This driver runs
the grapehip CLI over ranks x lb x directed x serialize cells on ONE
machine (multi-process, TCP control plane; on a GPU box ranks share the
device through the TCP data plane) and validates every cell against the
world-1 output with the matching verifier:

  exact  : bfs, cdlp                (integer outputs)
  eps    : sssp, pagerank, lcc      (float outputs, rtol)
  wcc    : wcc                      (component isomorphism)

  python tools/app_matrix.py                    # CPU matrix
  python tools/app_matrix.py --gpu              # GPU matrix (on a box)
  python tools/app_matrix.py --worlds 1,2,4 --apps bfs,wcc
"""
import argparse
import os
import subprocess
import sys
import tempfile
import time
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

APPS = ("bfs", "sssp", "pagerank", "wcc", "cdlp", "lcc")
VERIFY = {"bfs": "exact", "cdlp": "exact", "sssp": "eps",
          "pagerank": "eps", "lcc": "eps", "wcc": "wcc"}


def gen_dataset(d: Path, num_v=4000, num_e=40000, seed=5):
    rng = np.random.default_rng(seed)
    src = rng.integers(0, num_v, size=num_e, dtype=np.int64)
    dst = rng.integers(0, num_v, size=num_e, dtype=np.int64)
    keep = src != dst
    src, dst = src[keep], dst[keep]
    w = rng.random(len(src), dtype=np.float32) * 99 + 1
    with open(d / "m.v", "w") as f:
        for v in range(num_v):
            f.write("%d\n" % v)
    with open(d / "m.e", "w") as f:
        for s, t, ww in zip(src, dst, w):
            f.write("%d %d %.6f\n" % (s, t, ww))
    return d / "m.v", d / "m.e"


def run_cell(app, world, efile, vfile, outdir, port, directed, gpu, lb,
             serialize=None, deser_prefix=None, timeout=300):
    outdir.mkdir(parents=True, exist_ok=True)
    cmd = [sys.executable, "-m", "grapehip.run_app",
           "--application", app, "--efile", str(efile),
           "--vfile", str(vfile), "--out_prefix", str(outdir),
           "--weighted"]
    if directed:
        cmd.append("--directed")
    if gpu:
        cmd.append("--gpu")
    if serialize == "save":
        cmd += ["--serialize", "--serialization_prefix", str(deser_prefix)]
    elif serialize == "load":
        cmd += ["--deserialize", "--serialization_prefix",
                str(deser_prefix)]
    procs = []
    for rank in range(world):
        env = dict(os.environ, RANK=str(rank), LOCAL_RANK=str(rank),
                   WORLD_SIZE=str(world), MASTER_ADDR="127.0.0.1",
                   MASTER_PORT=str(port), PYTHONPATH=str(REPO))
        if lb:
            env["GRAPEHIP_LB"] = lb
        procs.append(subprocess.Popen(cmd, env=env, cwd=str(REPO),
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    ok = True
    logs = []
    for p in procs:
        try:
            out, _ = p.communicate(timeout=timeout)
        except subprocess.TimeoutExpired:
            p.kill()
            out = b"TIMEOUT"
        logs.append(out.decode(errors="replace"))
        ok = ok and p.returncode == 0
    return ok, logs


def load_dir(d):
    import glob
    oids, vals = [], []
    for path in sorted(glob.glob(str(d) + "/result_frag_*")):
        for line in open(path):
            a, b = line.split()
            oids.append(int(a))
            vals.append(b)
    order = np.argsort(oids)
    return np.array(oids)[order], [vals[i] for i in order]


def verify(mode, got_dir, golden_dir, rtol=1e-4):
    o1, v1 = load_dir(got_dir)
    o2, v2 = load_dir(golden_dir)
    if len(o1) != len(o2) or not np.array_equal(o1, o2):
        return False, "oid sets differ (%d vs %d)" % (len(o1), len(o2))
    if mode == "exact":
        return (v1 == v2), "value mismatch"
    if mode == "eps":
        a = np.array([float(x) for x in v1])
        b = np.array([float(x) for x in v2])
        big = (np.abs(a) > 1e300) | (np.abs(b) > 1e300)
        if not np.array_equal(np.abs(a) > 1e300, np.abs(b) > 1e300):
            return False, "inf pattern differs"
        ok = np.allclose(a[~big], b[~big], rtol=rtol)
        return ok, "eps mismatch"
    if mode == "wcc":
        fwd, bwd = {}, {}
        for a, b in zip(v1, v2):
            if fwd.setdefault(a, b) != b or bwd.setdefault(b, a) != a:
                return False, "partition differs"
        return True, ""
    raise ValueError(mode)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--apps", default=",".join(APPS))
    ap.add_argument("--worlds", default="1,2,4")
    ap.add_argument("--lbs", default="")
    ap.add_argument("--gpu", action="store_true")
    ap.add_argument("--num-v", type=int, default=4000)
    ap.add_argument("--num-e", type=int, default=40000)
    ap.add_argument("--skip-serialize", action="store_true")
    ap.add_argument("--port", type=int, default=29810)
    args = ap.parse_args()
    apps = args.apps.split(",")
    worlds = [int(w) for w in args.worlds.split(",")]
    lbs = args.lbs.split(",") if args.lbs else (
        ["cm", "wm", "strict", "none"] if args.gpu else [""])

    tmp = Path(tempfile.mkdtemp(prefix="grapehip_matrix_"))
    vfile, efile = gen_dataset(tmp, args.num_v, args.num_e)
    port = args.port
    failures = []
    cells = 0
    t0 = time.time()
    for directed in (False, True):
        dtag = "dir" if directed else "und"
        for app in apps:
            golden = tmp / ("golden_%s_%s" % (app, dtag))
            ok, logs = run_cell(app, 1, efile, vfile, golden, port,
                                directed, args.gpu, lbs[0])
            port += 7
            if not ok:
                failures.append(("golden", app, dtag, logs[-1][-800:]))
                continue
            for world in worlds:
                for lb in lbs:
                    cell = tmp / ("cell_%s_%s_w%d_%s" % (app, dtag, world,
                                                         lb or "def"))
                    ok, logs = run_cell(app, world, efile, vfile, cell,
                                        port, directed, args.gpu, lb)
                    port += 7
                    cells += 1
                    if not ok:
                        failures.append((app, dtag, world, lb,
                                         logs[-1][-800:]))
                        continue
                    good, why = verify(VERIFY[app], cell, golden)
                    if not good:
                        failures.append((app, dtag, world, lb, why))
            if not args.skip_serialize:
                # serialize on world max, reload, verify
                world = worlds[-1]
                ck = tmp / ("ckpt_%s_%s" % (app, dtag))
                ck.mkdir(exist_ok=True)
                c1 = tmp / ("cell_%s_%s_save" % (app, dtag))
                ok1, l1 = run_cell(app, world, efile, vfile, c1, port,
                                   directed, args.gpu, lbs[0],
                                   serialize="save", deser_prefix=ck)
                port += 7
                c2 = tmp / ("cell_%s_%s_load" % (app, dtag))
                ok2, l2 = run_cell(app, world, efile, vfile, c2, port,
                                   directed, args.gpu, lbs[0],
                                   serialize="load", deser_prefix=ck)
                port += 7
                cells += 2
                if not (ok1 and ok2):
                    failures.append((app, dtag, "serialize",
                                     (l1 + l2)[-1][-800:]))
                else:
                    good, why = verify(VERIFY[app], c2, golden)
                    if not good:
                        failures.append((app, dtag, "serialize", why))
    dt = time.time() - t0
    print("matrix: %d cells in %.1fs, %d failures" % (cells, dt,
                                                      len(failures)))
    for f in failures:
        print("FAIL:", *[str(x)[:200] for x in f])
    sys.exit(1 if failures else 0)


if __name__ == "__main__":
    main()
