"""Pure NumPy/SciPy oracle implementations of the LDBC kernels.

These are independent reference implementations used to validate the C++
engine (CPU path) and, transitively, the HIP kernels (GPU tests compare
against the same oracles or against the CPU path on identical graphs).
"""
from __future__ import annotations

import numpy as np
from scipy.sparse import csr_matrix
from scipy.sparse.csgraph import connected_components, shortest_path

INT64_MAX = np.iinfo(np.int64).max
DBL_MAX = np.finfo(np.float64).max


def build_matrix(num_v, src, dst, w=None, directed=True):
    data = np.ones(len(src)) if w is None else np.asarray(w, dtype=np.float64)
    m = csr_matrix((data, (src, dst)), shape=(num_v, num_v))
    if not directed:
        m = m.maximum(m.T)
    return m


def bfs_oracle(num_v, src, dst, source, directed=True):
    m = build_matrix(num_v, src, dst, None, directed)
    d = shortest_path(m, method="D", unweighted=True, directed=True,
                      indices=source)
    out = np.full(num_v, INT64_MAX, dtype=np.int64)
    finite = np.isfinite(d)
    out[finite] = d[finite].astype(np.int64)
    return out


def sssp_oracle(num_v, src, dst, w, source, directed=True):
    # Multigraph semantics: parallel edges relax independently == min-dedup.
    s = np.asarray(src, dtype=np.int64)
    d = np.asarray(dst, dtype=np.int64)
    ww = np.asarray(w, dtype=np.float64)
    if not directed:
        s, d, ww = (np.concatenate([s, d]), np.concatenate([d, s]),
                    np.concatenate([ww, ww]))
    order = np.lexsort((ww, d, s))
    s, d, ww = s[order], d[order], ww[order]
    keep = np.ones(len(s), dtype=bool)
    keep[1:] = (s[1:] != s[:-1]) | (d[1:] != d[:-1])
    m = csr_matrix((ww[keep], (s[keep], d[keep])), shape=(num_v, num_v))
    dist = shortest_path(m, method="D", directed=True, indices=source)
    out = np.full(num_v, DBL_MAX)
    finite = np.isfinite(dist)
    out[finite] = dist[finite]
    return out


def pagerank_oracle(num_v, src, dst, damping=0.85, iters=10, directed=True):
    if not directed:
        src, dst = np.concatenate([src, dst]), np.concatenate([dst, src])
    outdeg = np.bincount(src, minlength=num_v).astype(np.float64)
    r = np.full(num_v, 1.0 / num_v)
    for _ in range(iters):
        dangling = r[outdeg == 0].sum()
        contrib = np.zeros(num_v)
        c = np.where(outdeg > 0, r / np.maximum(outdeg, 1), 0.0)
        np.add.at(contrib, dst, c[src])
        r = (1 - damping) / num_v + damping * (contrib + dangling / num_v)
    return r


def wcc_oracle(num_v, src, dst):
    m = build_matrix(num_v, src, dst, None, True)
    n, labels = connected_components(m, directed=True, connection="weak")
    # canonical label = min vertex id in component
    out = np.zeros(num_v, dtype=np.int64)
    for comp in range(n):
        members = np.where(labels == comp)[0]
        out[members] = members.min()
    return out


def cdlp_oracle(num_v, src, dst, iters=10, directed=True):
    """Synchronous label propagation; directed uses in+out multiset."""
    if directed:
        nbr_src = np.concatenate([src, dst])
        nbr_dst = np.concatenate([dst, src])
    else:
        nbr_src = np.concatenate([src, dst])
        nbr_dst = np.concatenate([dst, src])
    # adjacency list: for vertex v, multiset of neighbor ids
    order = np.argsort(nbr_src, kind="stable")
    ns, nd = nbr_src[order], nbr_dst[order]
    starts = np.searchsorted(ns, np.arange(num_v + 1))
    labels = np.arange(num_v, dtype=np.int64)
    for _ in range(iters):
        new = labels.copy()
        for v in range(num_v):
            lo, hi = starts[v], starts[v + 1]
            if lo == hi:
                continue
            labs = np.sort(labels[nd[lo:hi]])
            # mode with min tie-break
            vals, counts = np.unique(labs, return_counts=True)
            new[v] = vals[np.argmax(counts)]
        labels = new
    return labels


def lcc_oracle(num_v, src, dst, directed=True):
    nbrs = [set() for _ in range(num_v)]
    outs = [set() for _ in range(num_v)]
    for s, d in zip(src, dst):
        if s == d:
            continue
        nbrs[s].add(d)
        nbrs[d].add(s)
        outs[s].add(d)
        if not directed:
            outs[d].add(s)
    out = np.zeros(num_v)
    for v in range(num_v):
        nv = nbrs[v]
        deg = len(nv)
        if deg < 2:
            continue
        cnt = sum(len(nv & outs[u]) for u in nv)
        out[v] = cnt / (deg * (deg - 1))
    return out
