"""Synthetic graph generators (no network — datasets are generated).

Used by tests (small graphs vs. NumPy oracles) and by bench.py as the
stand-in for LDBC datagen inputs (same |V|, |E| shape, random weights).
Deterministic per (seed, rank): each rank produces its slice of the global
edge list, so a multi-rank load sees exactly the same graph.
"""
from __future__ import annotations

import numpy as np


def uniform_edges(num_vertices: int, num_edges: int, seed: int = 42,
                  rank: int = 0, world: int = 1, weighted: bool = False):
    """Uniform random directed edges; rank gets a contiguous slice."""
    lo = num_edges * rank // world
    hi = num_edges * (rank + 1) // world
    # per-rank child seed derived from (seed, rank, world): ranks are
    # independent but the union is deterministic for a fixed world size
    rng = np.random.default_rng([seed, rank, world])
    n = hi - lo
    src = rng.integers(0, num_vertices, size=n, dtype=np.int64)
    dst = rng.integers(0, num_vertices, size=n, dtype=np.int64)
    if weighted:
        w = rng.random(size=n, dtype=np.float32) * 99.0 + 1.0
        return src, dst, w
    return src, dst, None


def rmat_edges(scale: int, edge_factor: int = 16, seed: int = 42,
               rank: int = 0, world: int = 1, weighted: bool = False,
               a: float = 0.57, b: float = 0.19, c: float = 0.19):
    """R-MAT/Kronecker edges (Graph500-style skew), sliced per rank."""
    num_vertices = 1 << scale
    num_edges = num_vertices * edge_factor
    lo = num_edges * rank // world
    hi = num_edges * (rank + 1) // world
    n = hi - lo
    rng = np.random.default_rng([seed, rank, world])
    src = np.zeros(n, dtype=np.int64)
    dst = np.zeros(n, dtype=np.int64)
    for level in range(scale):
        r1 = rng.random(n)
        r2 = rng.random(n)
        src_bit = r1 > (a + b)
        dst_bit = np.where(
            src_bit,
            r2 > (c / (c + (1 - a - b - c)) if (c + (1 - a - b - c)) > 0 else 0.5),
            r2 > (b / (a + b)),
        )
        src |= src_bit.astype(np.int64) << level
        dst |= dst_bit.astype(np.int64) << level
    if weighted:
        w = rng.random(size=n, dtype=np.float32) * 99.0 + 1.0
        return src, dst, w, num_vertices
    return src, dst, None, num_vertices
