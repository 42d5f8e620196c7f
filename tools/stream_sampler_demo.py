#!/usr/bin/env python3
"""Streaming GNN sampler demo (reference examples/gnn_sampler
run_sampler.cc:92-141: consume edge stream -> Query -> produce walks).
Kafka isn't available in this environment, so the stream is replayed from
an in-memory batch source through mutate_graph; the sampler app itself is
identical to the batch path.

  python tools/stream_sampler_demo.py
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import grapehip


def edge_stream(rng, nv, batches, batch_size):
    for _ in range(batches):
        src = rng.integers(0, nv, batch_size).astype(np.int64)
        dst = rng.integers(0, nv, batch_size).astype(np.int64)
        keep = src != dst
        yield src[keep], dst[keep]


def main():
    nv = 5000
    rng = np.random.default_rng(7)
    eng = grapehip.Engine(rank=0, world=1, master_port=29697)
    # bootstrap graph
    src, dst = next(edge_stream(rng, nv, 1, 20000))
    g = eng.load_edges(src, dst, directed=True, num_vertices=nv)
    starts = np.arange(64, dtype=np.int64)
    empty = np.array([], dtype=np.int64)
    for i, (bs, bd) in enumerate(edge_stream(rng, nv, 5, 4000)):
        # "consume": apply the next edge batch
        g = eng.mutate_graph(g, add_src=bs, add_dst=bd,
                             remove_src=empty, remove_dst=empty,
                             remove_vertices=empty)
        # "query": multi-hop walks over the updated graph
        r = eng.sample(g, starts, hops=3, strategy="edge_weight",
                       seed=100 + i)
        # "produce": emit walks (stdout stands in for the Kafka producer)
        done = int((r["paths"][:, -1] >= 0).sum())
        print("batch %d: graph |E|=%d, %d/%d walks completed 3 hops"
              % (i, g.num_edges, done, len(starts)))
    print("stream demo OK")


if __name__ == "__main__":
    main()
