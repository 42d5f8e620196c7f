#!/usr/bin/env python3
"""Verify run_app outputs against reference results (the reference's
misc/app_tests.sh verifiers: ExactVerify = sorted compare, EpsVerify =
relative-epsilon float compare (eps_check.cc:43-56), WCCVerify =
component-mapping isomorphism (wcc_check.cc)).

  python tools/verify_output.py exact  out_dir/ golden_file
  python tools/verify_output.py eps    out_dir/ golden_file [rtol]
  python tools/verify_output.py wcc    out_dir/ golden_file
"""
import glob
import sys

import numpy as np


def load_dir(d):
    oids, vals = [], []
    for path in sorted(glob.glob(d.rstrip("/") + "/result_frag_*")):
        for line in open(path):
            a, b = line.split()
            oids.append(int(a))
            vals.append(b)
    return np.array(oids), vals


def load_file(p):
    oids, vals = [], []
    for line in open(p):
        a, b = line.split()
        oids.append(int(a))
        vals.append(b)
    return np.array(oids), vals


def main():
    if len(sys.argv) < 4:
        sys.exit("usage: verify_output.py {exact,eps,wcc} <out_dir> <golden> [rtol]")
    mode, out_dir, golden = sys.argv[1], sys.argv[2], sys.argv[3]
    o1, v1 = load_dir(out_dir)
    o2, v2 = load_file(golden)
    if len(o1) != len(o2):
        sys.exit("FAIL: %d vs %d vertices" % (len(o1), len(o2)))
    i1, i2 = np.argsort(o1, kind="stable"), np.argsort(o2, kind="stable")
    if not np.array_equal(o1[i1], o2[i2]):
        sys.exit("FAIL: oid sets differ")
    a = [v1[i] for i in i1]
    b = [v2[i] for i in i2]
    if mode == "exact":
        bad = sum(1 for x, y in zip(a, b) if x != y)
        if bad:
            sys.exit("FAIL: %d mismatching values" % bad)
    elif mode == "eps":
        rtol = float(sys.argv[4]) if len(sys.argv) > 4 else 1e-4
        x = np.array([float(v) for v in a])
        y = np.array([float(v) for v in b])
        inf = y > 1e300
        if not (x[inf] > 1e300).all():
            sys.exit("FAIL: reachability differs")
        if not np.allclose(x[~inf], y[~inf], rtol=rtol):
            worst = np.abs(x[~inf] - y[~inf]) / np.maximum(1e-300,
                                                           np.abs(y[~inf]))
            sys.exit("FAIL: max rel err %.3g" % worst.max())
    elif mode == "wcc":
        fwd, bwd = {}, {}
        for x, y in zip(a, b):
            if fwd.setdefault(x, y) != y or bwd.setdefault(y, x) != x:
                sys.exit("FAIL: component mapping not a bijection")
    else:
        sys.exit("unknown mode " + mode)
    print("OK (%s, %d vertices)" % (mode, len(o1)))


if __name__ == "__main__":
    main()
