"""grapehip — an MI355X-native PIE graph-processing engine.

Brand-new implementation of the GRAPE PIE model (PEval/IncEval fixpoint over
edge-cut CSR fragments) with the capabilities of alibaba/libgrape-lite:
the six LDBC Graphalytics kernels (BFS, SSSP, PageRank, WCC, CDLP, LCC) on a
multithreaded CPU path and a hand-written HIP/CDNA4 GPU path, one process per
GPU, TCP control plane + RCCL-over-xGMI data plane.
"""
from __future__ import annotations

import os

from grapehip._core import Engine, Graph, WITH_HIP  # noqa: F401

from grapehip.generate import rmat_edges, uniform_edges  # noqa: F401
from grapehip.io import (read_ldbc_edges, read_ldbc_vertices,  # noqa: F401
                         load_string_edges, rebalance_partition)  # noqa: F401

__version__ = "0.1.0"


def engine_from_env(n_threads: int = 0, gpu: bool = False,
                    port_offset: int = 17) -> Engine:
    """Create an Engine from torchrun-style env (RANK/WORLD_SIZE/MASTER_*).

    The engine's rendezvous port is MASTER_PORT + port_offset so it never
    collides with the launcher's own store.
    """
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    addr = os.environ.get("MASTER_ADDR", "127.0.0.1")
    port = int(os.environ.get("MASTER_PORT", "29500")) + port_offset
    return Engine(rank=rank, world=world, master_addr=addr,
                  master_port=port, n_threads=n_threads, gpu=gpu)
