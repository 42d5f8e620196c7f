// grapehip — vertex-cut fragment + gather-scatter PageRank (CPU).
// Reference parity: grape/fragment/immutable_vertexcut_fragment.h (edges
// hash-partitioned across fragments, bucketed into bucket_num^2 src×dst
// tiles for cache locality; dense uint64 oids only),
// grape/parallel/gather_scatter_message_manager.h (GatherMasterVertices =
// per-segment partial reduction at the master, ScatterMasterVertices =
// broadcast back) and examples/analytical_apps/pagerank/pagerank_vc.h.
// grapehip keeps the same structure over the TCP control plane: partial
// per-vertex arrays are reduced segment-wise at masters (reduce-scatter
// shape) and replicated back (allgather shape).
#pragma once

#include <cmath>
#include <cstring>
#include <memory>
#include <vector>

#include "net.hpp"
#include "parallel.hpp"
#include "types.hpp"

namespace grapehip {

class VertexcutFragment {
 public:
  // Collective. Each rank passes an arbitrary slice of the global edge
  // list; edges are shuffled to their hash owner (hash of the edge, not of
  // a vertex — the defining property of a vertex cut).
  static std::unique_ptr<VertexcutFragment> Build(
      TcpComm* comm, fid_t fid, int fnum, uint64_t nv,
      const std::vector<uint32_t>& src_in,
      const std::vector<uint32_t>& dst_in, int bucket_num = 8) {
    auto out = std::make_unique<VertexcutFragment>();
    VertexcutFragment& F = *out;
    F.fid_ = fid;
    F.fnum_ = fnum;
    F.nv_ = nv;
    F.bucket_num_ = bucket_num;
    F.seg_.resize(fnum + 1);
    uint64_t slice = (nv + fnum - 1) / fnum;
    for (int f = 0; f <= fnum; ++f)
      F.seg_[f] = std::min<uint64_t>(static_cast<uint64_t>(f) * slice, nv);

    // shuffle edges by edge hash
    std::vector<std::vector<uint64_t>> bins(fnum);
    for (size_t i = 0; i < src_in.size(); ++i) {
      uint64_t e = (static_cast<uint64_t>(src_in[i]) << 32) | dst_in[i];
      uint64_t h = e * 0x9e3779b97f4a7c15ULL;
      bins[(h >> 33) % fnum].push_back(e);
    }
    std::vector<std::string> send(fnum);
    for (int f = 0; f < fnum; ++f)
      send[f].assign(reinterpret_cast<const char*>(bins[f].data()),
                     bins[f].size() * 8);
    std::vector<std::string> recv =
        (comm && fnum > 1) ? comm->exchange_all(send) : std::move(send);

    size_t n = 0;
    for (auto& blob : recv) n += blob.size() / 8;
    // bucket into bucket_num^2 src×dst tiles (cache-friendly gathers)
    int B = bucket_num;
    uint64_t bslice = (nv + B - 1) / B;
    std::vector<std::vector<uint64_t>> tiles(B * B);
    for (auto& blob : recv) {
      const uint64_t* p = reinterpret_cast<const uint64_t*>(blob.data());
      size_t m = blob.size() / 8;
      for (size_t i = 0; i < m; ++i) {
        uint32_t s32 = static_cast<uint32_t>(p[i] >> 32);
        uint32_t d32 = static_cast<uint32_t>(p[i]);
        tiles[(s32 / bslice) * B + (d32 / bslice)].push_back(p[i]);
      }
    }
    F.src_.reserve(n);
    F.dst_.reserve(n);
    F.tile_off_.resize(B * B + 1, 0);
    for (int t = 0; t < B * B; ++t) {
      F.tile_off_[t] = F.src_.size();
      for (uint64_t e : tiles[t]) {
        F.src_.push_back(static_cast<uint32_t>(e >> 32));
        F.dst_.push_back(static_cast<uint32_t>(e));
      }
    }
    F.tile_off_[B * B] = F.src_.size();
    uint64_t local = F.src_.size();
    F.total_edges_ = comm ? comm->allreduce_sum(local) : local;
    return out;
  }

  fid_t fid() const { return fid_; }
  int fnum() const { return fnum_; }
  uint64_t num_vertices() const { return nv_; }
  uint64_t total_edges() const { return total_edges_; }
  uint64_t local_edges() const { return src_.size(); }
  const std::vector<uint64_t>& segments() const { return seg_; }
  const std::vector<uint32_t>& srcs() const { return src_; }
  const std::vector<uint32_t>& dsts() const { return dst_; }
  std::pair<size_t, size_t> tile(int t) const {
    return {tile_off_[t], tile_off_[t + 1]};
  }
  int bucket_num() const { return bucket_num_; }

  // Gather: elementwise-sum partial dense arrays at segment masters, then
  // scatter the reduced segments back to every rank (in place on `data`,
  // which is a full nv-sized array on every rank).
  void gather_scatter_sum(TcpComm* comm, double* data) const {
    if (!comm || fnum_ == 1) return;
    // reduce-scatter shape: send my partial slice of segment f to rank f
    std::vector<std::string> send(fnum_);
    for (int f = 0; f < fnum_; ++f) {
      if (f == static_cast<int>(fid_)) continue;
      send[f].assign(reinterpret_cast<const char*>(data + seg_[f]),
                     (seg_[f + 1] - seg_[f]) * 8);
    }
    auto recv = comm->exchange_all(send);
    uint64_t b = seg_[fid_], e = seg_[fid_ + 1];
    for (int f = 0; f < fnum_; ++f) {
      if (f == static_cast<int>(fid_)) continue;
      const double* p = reinterpret_cast<const double*>(recv[f].data());
      size_t m = recv[f].size() / 8;
      parallel_for(0, m, [&](size_t i) { data[b + i] += p[i]; }, 8192);
      (void)e;
    }
    // allgather shape: broadcast my reduced segment to every rank
    std::string mine(reinterpret_cast<const char*>(data + b), (e - b) * 8);
    std::vector<std::string> send2(fnum_, mine);
    send2[fid_].clear();
    auto recv2 = comm->exchange_all(send2);
    for (int f = 0; f < fnum_; ++f) {
      if (f == static_cast<int>(fid_)) continue;
      std::memcpy(data + seg_[f], recv2[f].data(), recv2[f].size());
    }
  }

 private:
  fid_t fid_ = 0;
  int fnum_ = 1;
  uint64_t nv_ = 0, total_edges_ = 0;
  int bucket_num_ = 8;
  std::vector<uint64_t> seg_;
  std::vector<uint32_t> src_, dst_;
  std::vector<size_t> tile_off_;
};

// PageRank over a vertex-cut fragment (pagerank_vc.h semantics: directed
// edges, masters apply the damping update, dangling mass redistributed).
inline std::vector<double> pagerank_vc(const VertexcutFragment& F,
                                       TcpComm* comm, double damping,
                                       int iters) {
  const uint64_t nv = F.num_vertices();
  std::vector<double> deg(nv, 0.0), rank(nv), partial(nv);
  // out-degrees: local counts, gather-scatter sum
  for (uint32_t s : F.srcs()) deg[s] += 1.0;
  F.gather_scatter_sum(comm, deg.data());
  double r0 = 1.0 / static_cast<double>(nv);
  std::fill(rank.begin(), rank.end(), r0);

  const int B = F.bucket_num();
  for (int it = 0; it < iters; ++it) {
    std::fill(partial.begin(), partial.end(), 0.0);
    // tiles sharing a dst range never race when processed per dst-column
    parallel_for(0, static_cast<size_t>(B), [&](size_t dcol) {
      for (int srow = 0; srow < B; ++srow) {
        auto [b, e] = F.tile(srow * B + dcol);
        const auto& src = F.srcs();
        const auto& dst = F.dsts();
        for (size_t i = b; i < e; ++i)
          partial[dst[i]] += rank[src[i]] / deg[src[i]];
      }
    }, 1);
    F.gather_scatter_sum(comm, partial.data());
    // dangling mass (identical on every rank after the scatter)
    double dangling = 0;
    uint64_t bseg = F.segments()[F.fid()], eseg = F.segments()[F.fid() + 1];
    for (uint64_t v = bseg; v < eseg; ++v)
      if (deg[v] == 0.0) dangling += rank[v];
    if (comm && F.fnum() > 1) {
      std::vector<double> all(F.fnum());
      comm->allgather(&dangling, 8, all.data());
      dangling = 0;
      for (double x : all) dangling += x;
    }
    double base = (1.0 - damping) / nv + damping * dangling / nv;
    parallel_for(0, nv, [&](size_t v) {
      rank[v] = base + damping * partial[v];
    }, 8192);
  }
  return rank;
}

}  // namespace grapehip
