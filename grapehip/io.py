"""LDBC Graphalytics TSV input (.v / .e files).

Reference parity: grape/io/local_io_adaptor.cc partial-read loader — each
rank reads a byte-range of the file aligned to line boundaries and parses
its slice; the engine's distributed builder shuffles records to owners.
"""
from __future__ import annotations

import os

import numpy as np


def _byte_slice(path: str, rank: int, world: int):
    size = os.path.getsize(path)
    lo = size * rank // world
    hi = size * (rank + 1) // world
    with open(path, "rb") as f:
        if lo > 0:
            f.seek(lo - 1)
            # advance to the start of the next full line
            chunk = f.read(1)
            if chunk != b"\n":
                f.readline()
            lo = f.tell()
        f.seek(lo)
        data = f.read(hi - lo)
        if hi < size and (not data or data[-1:] != b"\n"):
            # finish the straddling line
            with open(path, "rb") as g:
                g.seek(lo + len(data))
                data += g.readline()
    return data


def _ncols(path: str) -> int:
    with open(path, "rb") as f:
        line = f.readline()
    return len(line.split())


def read_ldbc_edges(efile: str, weighted: bool = False, rank: int = 0,
                    world: int = 1):
    """Parse this rank's slice of an LDBC .e file (src dst [weight]).

    The file's column count is probed from its first line: a weight column
    present in the file is parsed (and dropped if weighted=False), so
    unweighted apps work on weighted .e files, like the reference loader.
    """
    data = _byte_slice(efile, rank, world)
    if not data.strip():
        empty = np.zeros(0, dtype=np.int64)
        return (empty, empty.copy(),
                np.zeros(0, dtype=np.float32) if weighted else None)
    ncols = _ncols(efile)
    if weighted and ncols < 3:
        raise ValueError("%s has no weight column" % efile)
    fields = np.array(data.split(),
                      dtype=np.float64 if ncols >= 3 else np.int64)
    fields = fields.reshape(-1, ncols)
    src = fields[:, 0].astype(np.int64)
    dst = fields[:, 1].astype(np.int64)
    w = fields[:, 2].astype(np.float32) if weighted else None
    return src, dst, w


def read_ldbc_vertices(vfile: str, rank: int = 0, world: int = 1):
    """Parse this rank's slice of an LDBC .v file (oid [vdata])."""
    data = _byte_slice(vfile, rank, world)
    if not data.strip():
        return np.zeros(0, dtype=np.int64)
    oids = [int(line.split()[0]) for line in data.splitlines() if line.strip()]
    return np.array(oids, dtype=np.int64)
