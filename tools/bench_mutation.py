#!/usr/bin/env python3
"""Delta-mutation cost check: in-place MutateDelta vs functional rebuild.

Builds an N-edge graph, then times (a) a small in-place edge delta and
(b) an equivalent full rebuild via remove_vertices=[dummy] fallback-free
reload. Reports the ratio; the in-place path must scale with the delta,
not with E (VERDICT r01 item 5 / reference mutable_edgecut_fragment.h).
"""
import argparse
import sys
import time
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import grapehip  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--num-v", type=int, default=2_000_000)
    ap.add_argument("--num-e", type=int, default=20_000_000)
    ap.add_argument("--delta", type=int, default=1000)
    ap.add_argument("--reps", type=int, default=5)
    args = ap.parse_args()

    rng = np.random.default_rng(11)
    src = rng.integers(0, args.num_v, size=args.num_e, dtype=np.int64)
    dst = rng.integers(0, args.num_v, size=args.num_e, dtype=np.int64)
    keep = src != dst
    src, dst = src[keep], dst[keep]
    eng = grapehip.Engine(rank=0, world=1, master_port=29663)

    t0 = time.time()
    g = eng.load_edges(src, dst, directed=False, num_vertices=args.num_v)
    t_build = time.time() - t0

    empty = np.array([], dtype=np.int64)
    ad_s = rng.integers(0, args.num_v, args.delta).astype(np.int64)
    ad_d = rng.integers(0, args.num_v, args.delta).astype(np.int64)
    k = ad_s != ad_d
    ad_s, ad_d = ad_s[k], ad_d[k]

    t0 = time.time()
    for _ in range(args.reps):
        g = eng.mutate_graph(g, add_src=ad_s, add_dst=ad_d,
                             remove_src=src[:args.delta],
                             remove_dst=dst[:args.delta],
                             remove_vertices=empty)
    t_delta = (time.time() - t0) / args.reps

    print(f"edges={len(src)} delta={args.delta}")
    print(f"full build      : {t_build*1e3:9.1f} ms")
    print(f"in-place delta  : {t_delta*1e3:9.1f} ms")
    print(f"speedup vs build: {t_build/t_delta:9.1f}x")


if __name__ == "__main__":
    main()
