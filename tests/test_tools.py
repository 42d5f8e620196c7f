"""Smoke tests for the repo tools (verifier, profile summarizers)."""
import subprocess
import sys
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent


def test_verify_output_exact_and_eps(tmp_path):
    out = tmp_path / "out"
    out.mkdir()
    (out / "result_frag_0").write_text("1 5\n3 7\n")
    (out / "result_frag_1").write_text("2 6\n")
    golden = tmp_path / "golden"
    golden.write_text("3 7\n1 5\n2 6\n")
    for mode in ("exact", "eps"):
        r = subprocess.run([sys.executable, "tools/verify_output.py", mode,
                            str(out), str(golden)], cwd=REPO,
                           capture_output=True, text=True)
        assert r.returncode == 0, (mode, r.stdout, r.stderr)
    bad = tmp_path / "bad"
    bad.write_text("3 7\n1 9\n2 6\n")
    r = subprocess.run([sys.executable, "tools/verify_output.py", "exact",
                        str(out), str(bad)], cwd=REPO, capture_output=True,
                       text=True)
    assert r.returncode != 0


def test_verify_output_wcc_isomorphism(tmp_path):
    out = tmp_path / "out"
    out.mkdir()
    (out / "result_frag_0").write_text("1 10\n2 10\n3 20\n")
    golden = tmp_path / "golden"
    golden.write_text("1 7\n2 7\n3 9\n")       # same partition, new labels
    r = subprocess.run([sys.executable, "tools/verify_output.py", "wcc",
                        str(out), str(golden)], cwd=REPO,
                       capture_output=True, text=True)
    assert r.returncode == 0, r.stdout + r.stderr
    broken = tmp_path / "broken"
    broken.write_text("1 7\n2 9\n3 9\n")       # different partition
    r = subprocess.run([sys.executable, "tools/verify_output.py", "wcc",
                        str(out), str(broken)], cwd=REPO,
                       capture_output=True, text=True)
    assert r.returncode != 0


def test_app_matrix_smoke():
    # app_tests.sh-equivalent matrix (reference misc/app_tests.sh): a
    # reduced CPU matrix — full sweep via `python tools/app_matrix.py`
    r = subprocess.run([sys.executable, "tools/app_matrix.py",
                        "--apps", "bfs,wcc", "--worlds", "1,2",
                        "--skip-serialize", "--port", "29890"],
                       cwd=REPO, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
