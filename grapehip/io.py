"""LDBC Graphalytics TSV input (.v / .e files).

Reference parity: grape/io/local_io_adaptor.cc partial-read loader — each
rank reads a byte-range of the file aligned to line boundaries and parses
its slice; the engine's distributed builder shuffles records to owners.
"""
from __future__ import annotations

import os

import numpy as np


def _line_start(f, pos: int, size: int) -> int:
    """First line-start byte offset at or after pos."""
    if pos <= 0:
        return 0
    if pos >= size:
        return size
    f.seek(pos - 1)
    if f.read(1) != b"\n":
        f.readline()  # skip the rest of the straddling line
    return f.tell()


def _byte_slice(path: str, rank: int, world: int):
    # a rank owns the lines whose first byte falls in its raw byte range;
    # aligning BOTH ends with the same rule gives no gaps, overlaps, or
    # duplicates even when a slice is smaller than one line
    size = os.path.getsize(path)
    with open(path, "rb") as f:
        lo = _line_start(f, size * rank // world, size)
        hi = _line_start(f, size * (rank + 1) // world, size)
        if lo >= hi:
            return b""
        f.seek(lo)
        return f.read(hi - lo)


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
    fields = data.split()
    if len(fields) % ncols:
        raise ValueError("%s: ragged row (column count varies)" % efile)
    # parse id columns directly as int64 — going through float64 would
    # silently corrupt oids above 2^53 (sparse LDBC id spaces)
    src = np.array(fields[0::ncols], dtype=np.int64)
    dst = np.array(fields[1::ncols], dtype=np.int64)
    w = (np.array(fields[2::ncols], dtype=np.float32) if weighted else None)
    return src, dst, w


def read_ldbc_vertices(vfile: str, rank: int = 0, world: int = 1):
    """Parse this rank's slice of an LDBC .v file (oid [vdata])."""
    data = _byte_slice(vfile, rank, world)
    if not data.strip():
        return np.zeros(0, dtype=np.int64)
    oids = [int(line.split()[0]) for line in data.splitlines() if line.strip()]
    return np.array(oids, dtype=np.int64)


def build_string_dictionary(eng, local_strings):
    """Distributed string-oid dictionary (reference: IdIndexer over string
    oids, grape/graph/id_indexer.h). Every rank contributes the strings it
    read; the replicated dictionary maps each distinct string to a dense
    int64 id (global sorted order, so every rank derives identical ids
    without a second exchange). Returns (dict str->id, list id->str)."""
    blob = b"\0".join(s.encode() if isinstance(s, str) else bytes(s)
                      for s in sorted(set(local_strings)))
    blobs = eng._exchange_all([blob] * eng.world)
    seen = set()
    for b in blobs:
        if b:
            seen.update(b.split(b"\0"))
    ordered = sorted(seen)
    s2i = {s.decode(): i for i, s in enumerate(ordered)}
    return s2i, [s.decode() for s in ordered]


def load_string_edges(eng, src_strs, dst_strs, weights=None, directed=False,
                      **kw):
    """Load a graph whose vertex ids are strings. The dense int64 mapping
    is built collectively; returns (graph, id->string list)."""
    import numpy as np
    s2i, i2s = build_string_dictionary(
        eng, list(src_strs) + list(dst_strs))
    src = np.array([s2i[s] for s in src_strs], dtype=np.int64)
    dst = np.array([s2i[s] for s in dst_strs], dtype=np.int64)
    g = eng.load_edges(src, dst, weights=weights, directed=directed,
                       num_vertices=len(i2s), **kw)
    return g, i2s


def rebalance_partition(eng, src, dst, num_vertices, factor=1.0):
    """Degree-balanced contiguous partition (reference: rebalancer.h —
    vertex ownership reassigned by degree-weighted cost). Each rank counts
    degrees over its edge slice; counts are summed collectively; the oid
    space is split into fnum contiguous ranges of equal cost
    (vertex_weight + factor * degree). Returns this rank's oid array, to
    pass as load_edges(vertex_oids=...)."""
    import numpy as np
    deg = np.bincount(src, minlength=num_vertices).astype(np.int64)
    deg += np.bincount(dst, minlength=num_vertices)
    blobs = eng._exchange_all([deg.tobytes()] * eng.world)
    total = np.zeros(num_vertices, dtype=np.int64)
    for b in blobs:
        total += np.frombuffer(b, dtype=np.int64)
    cost = 1.0 + factor * total
    csum = np.cumsum(cost)
    bound = csum[-1] / eng.world
    splits = [0]
    for f in range(1, eng.world):
        splits.append(int(np.searchsorted(csum, bound * f)))
    splits.append(num_vertices)
    lo, hi = splits[eng.rank], splits[eng.rank + 1]
    return np.arange(lo, hi, dtype=np.int64)
