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


def bc_oracle(num_v, src, dst, source, directed=True):
    """Single-source Brandes dependency delta over the stored adjacency
    (undirected: both orientations; duplicates count as parallel paths)."""
    adj = [[] for _ in range(num_v)]
    for s, d in zip(src, dst):
        adj[s].append(d)
        if not directed and s != d:
            adj[d].append(s)
    INF = float("inf")
    depth = np.full(num_v, INF)
    sigma = np.zeros(num_v)
    depth[source] = 0
    sigma[source] = 1.0
    frontier = [source]
    order = []
    while frontier:
        order.extend(frontier)
        nxt = set()
        for v in frontier:
            for u in adj[v]:
                if depth[u] == INF or depth[u] == depth[v] + 1:
                    if depth[u] == INF:
                        depth[u] = depth[v] + 1
                        nxt.add(u)
                    sigma[u] += sigma[v]
        frontier = sorted(nxt)
    delta = np.zeros(num_v)
    for v in reversed(order):
        for u in adj[v]:
            if depth[u] == depth[v] - 1:
                delta[u] += sigma[u] / sigma[v] * (1.0 + delta[v])
    return delta, sigma, depth


def _stored_adj(num_v, src, dst, directed):
    """Multiset adjacency matching Fragment storage (self-loops kept on the
    forward orientation, reverse of self-loops dropped for undirected)."""
    adj = [[] for _ in range(num_v)]
    for s, d in zip(src, dst):
        adj[s].append(d)
        if not directed and s != d:
            adj[d].append(s)
    return adj


def coreness_oracle(num_v, src, dst, directed=False):
    """Peeling coreness; degree = stored multiplicity, self-loops excluded."""
    adj = _stored_adj(num_v, src, dst, directed)
    rem = np.array([sum(1 for u in a if u != v)
                    for v, a in enumerate(adj)], dtype=np.int64)
    core = np.zeros(num_v, dtype=np.int64)
    alive = np.ones(num_v, bool)
    for k in range(0, int(rem.max(initial=0)) + 2):
        changed = True
        while changed:
            changed = False
            for v in range(num_v):
                if alive[v] and rem[v] <= k:
                    alive[v] = False
                    core[v] = k
                    for u in adj[v]:
                        if u != v and alive[u]:
                            rem[u] -= 1
                    changed = True
        if not alive.any():
            break
    return core


def kcore_oracle(num_v, src, dst, k, directed=False):
    adj = _stored_adj(num_v, src, dst, directed)
    rem = np.array([sum(1 for u in a if u != v)
                    for v, a in enumerate(adj)], dtype=np.int64)
    alive = np.ones(num_v, bool)
    changed = True
    while changed:
        changed = False
        for v in range(num_v):
            if alive[v] and rem[v] < k:
                alive[v] = False
                for u in adj[v]:
                    if u != v and alive[u]:
                        rem[u] -= 1
                changed = True
    return alive.astype(np.int64)


def kclique_oracle(num_v, src, dst, k, directed=False):
    """Count k-cliques on the simple undirected graph."""
    nbrs = [set() for _ in range(num_v)]
    for s, d in zip(src, dst):
        if s == d:
            continue
        nbrs[s].add(d)
        nbrs[d].add(s)
    oriented = [sorted(u for u in nbrs[v] if u > v) for v in range(num_v)]

    def rec(cand, depth):
        if depth == k - 1:
            return len(cand)
        total = 0
        for c in cand:
            nxt = [x for x in cand if x in nbrs[c] and x > c]
            if nxt:
                total += rec(nxt, depth + 1)
        return total

    if k == 2:
        return sum(len(o) for o in oriented)
    return sum(rec(o, 1) for o in oriented if o)
