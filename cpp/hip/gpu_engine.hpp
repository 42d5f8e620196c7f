// grapehip — GPU engine interface (host side).
//
// Reference parity: grape/cuda/worker/gpu_worker.h + gpu_message_manager.h
// reimagined: TCP control plane (lengths, termination, scalar collectives) +
// RCCL-over-xGMI data plane, dual HIP streams. One GpuContext per process ==
// one MI355X GPU == one fragment.
#pragma once

#include <cstdint>
#include <memory>
#include <vector>

#include "../core/fragment.hpp"
#include "../core/net.hpp"
#include "../core/types.hpp"
#include "hip_common.hpp"

struct ncclComm;  // fwd (rccl.h included in the .hip TU)

namespace grapehip {

struct DeviceGraph {
  uint32_t nv_global = 0;   // device id space (padded for dense renumber)
  uint64_t nv_real = 0;     // true vertex count (PageRank N etc.)
  uint32_t v_begin = 0, v_end = 0;
  uint32_t owned_real = 0;  // rows with real vertices (<= owned(); dense
                            // renumbering pads the tail of each slice)
  bool directed = false, weighted = false, has_in = false;
  uint64_t local_edges = 0, total_edges = 0, input_edges = 0;
  std::vector<uint32_t> seg_host;
  DeviceBuffer<uint32_t> seg;
  DeviceBuffer<uint64_t> oe_off;
  DeviceBuffer<uint32_t> oe_dst;
  DeviceBuffer<float> oe_w;
  DeviceBuffer<uint64_t> ie_off;
  DeviceBuffer<uint32_t> ie_dst;
  DeviceBuffer<float> ie_w;
  // degree buckets over owned rows (built lazily; row-per-thread/wave/block
  // scheduling for whole-graph sweeps — the CTA scheduler, preprocessed)
  DeviceBuffer<uint32_t> rows_small, rows_mid, rows_large;
  uint64_t n_small = 0, n_mid = 0, n_large = 0;
  bool buckets_built = false;
  // mirror topology (multi-GPU, built lazily): the reference's mirror-info /
  // BatchShuffle design (edgecut_fragment_base.h:569+, cuda batch_shuffle
  // :81-103) recast for xGMI — per-peer sorted lists of the REMOTE vertices
  // this rank's edges actually reference (recv side) and of the OWNED
  // vertices each peer references (send side). Dense per-round refreshes
  // then ship referenced values point-to-point on distinct xGMI links
  // instead of allgathering whole slices around the ring.
  DeviceBuffer<uint32_t> mr_recv_idx, mr_send_idx;   // concatenated, global ids
  std::vector<uint64_t> mr_recv_off, mr_send_off;    // [world+1] element offs
  DeviceBuffer<uint8_t> mr_sendbuf, mr_recvbuf;      // 4B-element staging
  bool mirrors_built = false;
  // CDLP label order (non-identity maps only): dev id -> rank in global
  // sorted-OID order, so min-label tie-breaks match the reference's
  // oid-space semantics under dense renumbering. Empty for identity maps.
  DeviceBuffer<uint32_t> oid_order;
  // hub-clustering renumber (synthetic graphs): new = perm[old],
  // old = inv[new]; slices preserved. inv_host caches the owned slice
  // for output oid translation (filled on first fetch).
  DeviceBuffer<uint32_t> perm, inv;
  bool permuted = false;
  std::vector<uint32_t> inv_host;
  // PageRank tiled-pull stream (built lazily; 8 B per stored edge):
  // (src<<32|dst) records grouped by 4096-vertex dst tile so gathers stay
  // L2-resident. pr_ntiles==0 after build => fell back (memory gate).
  DeviceBuffer<unsigned long long> pr_tiles;
  DeviceBuffer<uint64_t> pr_tile_off;
  uint32_t pr_ntiles = 0;
  uint64_t pr_own_lo = 0, pr_own_hi = 0, pr_total = 0;
  bool pr_tiles_built = false;
  uint32_t owned() const { return v_end - v_begin; }
  // cached hipGraph of one PageRank iteration (single-GPU fixed-iter
  // path) + the working set it references: the capture bakes device
  // pointers, so these buffers must live as long as the graph
  void* pr_graph_exec = nullptr;
  double pr_graph_damping = 0.0;
  int pr_calls = 0;  // capture lazily: one-shot runs skip the ~10ms
                     // instantiate; repeated runs (bench warmup) get it
  DeviceBuffer<double> pr_rank, pr_acc, pr_dangling;
  DeviceBuffer<float> pr_contrib;
  ~DeviceGraph();
};

struct GpuRunResult {
  std::vector<int64_t> i64;   // BFS depth / WCC & CDLP labels (owned range)
  std::vector<double> f64;    // SSSP dist / PR rank / LCC coeff
  int rounds = 0;
  double seconds = 0;         // max over ranks, kernel-side barrier-bracketed
  uint64_t traversed_edges = 0;  // for TEPS accounting (global)
  // this rank's data-plane sends during the run (comm-volume evidence:
  // p2p halo/mirror/row-fetch vs collective traffic)
  uint64_t bytes_p2p = 0, bytes_coll = 0;
};

class GpuContext {
 public:
  GpuContext(TcpComm* comm, int rank, int world);
  ~GpuContext();

  // Build a device graph from a host fragment (identity vertex map only).
  std::unique_ptr<DeviceGraph> upload(const Fragment& frag);

  // Generate an LDBC-datagen-shaped synthetic graph directly in HBM3E
  // (RMAT skew, random [1,100) weights), one owned slice per rank.
  std::unique_ptr<DeviceGraph> gen_synthetic(uint64_t nv, uint64_t ne,
                                             uint64_t seed, bool directed,
                                             bool weighted, bool build_in_csr,
                                             double rmat_a, double rmat_b,
                                             double rmat_c);

  // fetch=false skips the result D2H/convert (bench times algorithm only,
  // like the reference's "run algorithm" timer phase vs the Output phase).
  GpuRunResult bfs(DeviceGraph& g, int64_t source, bool fetch = true);
  GpuRunResult sssp(DeviceGraph& g, int64_t source, float delta,
                    bool fetch = true);
  GpuRunResult pagerank(DeviceGraph& g, double damping, int iters,
                        double tol = 0.0, bool fetch = true);
  GpuRunResult wcc(DeviceGraph& g, bool fetch = true);
  GpuRunResult cdlp(DeviceGraph& g, int iters, bool fetch = true);
  GpuRunResult lcc(DeviceGraph& g, bool fetch = true);
  GpuRunResult lcc_directed(DeviceGraph& g, bool fetch = true);

  void device_sync();
  // test hook: exclusive scan of host u32 data on the device
  std::vector<uint64_t> debug_scan(const std::vector<uint32_t>& in);
  int device_id() const { return dev_; }
  TcpComm* comm() { return comm_; }
  int rank() const { return rank_; }
  int world() const { return world_; }

  struct Impl;  // internal (kernels + scratch); public for free helpers

 private:
  std::unique_ptr<Impl> impl_;
  TcpComm* comm_;
  int rank_, world_, dev_ = 0;
};

}  // namespace grapehip
