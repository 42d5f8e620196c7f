import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import grapehip
def log(*a): print(*a, file=sys.stderr, flush=True)
NUM_V=1_000_000; NUM_E=12_000_000
rng = np.random.default_rng(1234)
u = rng.random(NUM_E); src = ((u**2.2)*NUM_V).astype(np.int64)%NUM_V
v = rng.random(NUM_E); dst = ((v**2.2)*NUM_V).astype(np.int64)%NUM_V
perm = rng.permutation(NUM_V).astype(np.int64); src,dst = perm[src],perm[dst]
keep = src!=dst; src,dst=src[keep],dst[keep]
log("edges", len(src), "srcsum", src.sum(), "dstsum", dst.sum())
gpu = grapehip.Engine(rank=0, world=1, master_port=29719, gpu=True)
gg = gpu.load_edges(src, dst, directed=False, num_vertices=NUM_V)
log("loaded")
for t in range(2):
    res = gpu.wcc(gg)
    order = np.argsort(res["oids"]); lab = np.asarray(res["values"])[order]
    uniq, cnt = np.unique(lab, return_counts=True)
    log("trial", t, "labels", len(uniq))
    for l, c in zip(uniq.tolist(), cnt.tolist()):
        if c > 1000: continue
        mem = np.where(lab == l)[0][:4]
        log("  label", l, "size", c, "members", mem.tolist(),
            "lab[label]", int(lab[l]), "idempotent", bool((lab[lab[mem]] == lab[mem]).all()))
        vv = mem[0]
        nbrs = np.unique(np.concatenate([dst[src==vv], src[dst==vv]]))[:6]
        log("    v", int(vv), "deg", int((src==vv).sum()+(dst==vv).sum()),
            "nbrs", nbrs.tolist(), "nbr_lab", lab[nbrs].tolist())
