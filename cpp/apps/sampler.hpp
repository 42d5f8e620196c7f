// grapehip — GNN random-walk sampler, CPU path.
// Reference parity: examples/gnn_sampler/sampler.h:35-238 (multi-hop walks
// from query vertices, strategies random / edge_weight / top_k via
// per-vertex index structures, fragment_indices.h) as a ParallelApp over
// the BSP engine: each round advances every live walk one hop; walks whose
// current vertex is remote travel to its owner as (walk, hop, gid)
// messages, and every sampled step is reported back to the walk's origin
// fragment the same way. Kafka streaming in/out (kafka_consumer.h) is not
// available in this environment; the Python driver replays edge streams
// through mutate_graph instead.
#pragma once

#include <algorithm>
#include <cstdint>
#include <mutex>
#include <random>
#include <vector>

#include "../core/fragment.hpp"
#include "../core/message_manager.hpp"

namespace grapehip {

enum class SampleStrategy : int { kRandom = 0, kEdgeWeight = 1, kTopK = 2 };

// Per-vertex alias tables for O(1) weighted draws (reference
// fragment_indices.h builds the same structure per vertex). Flat over the
// fragment's edge-id space; built once per query in parallel.
struct AliasTables {
  std::vector<float> prob;     // [local eid]
  std::vector<uint32_t> alias; // [local eid] (intra-row index)

  void build(const Fragment& frag) {
    size_t ne = frag.local_edges();
    prob.assign(ne, 1.0f);
    alias.assign(ne, 0);
    parallel_for(0, frag.ivnum(), [&](size_t vs) {
      vid_t v = static_cast<vid_t>(vs);
      auto adj = frag.out_edges(v);
      if (adj.n == 0) return;
      size_t base = adj.dst - frag.oe_dsts().data();
      // Vose's method over the row's weights
      double total = 0;
      for (size_t i = 0; i < adj.n; ++i)
        total += adj.w ? adj.w[i] : 1.0;
      std::vector<double> scaled(adj.n);
      std::vector<uint32_t> small, large;
      for (size_t i = 0; i < adj.n; ++i) {
        scaled[i] = (adj.w ? adj.w[i] : 1.0) * adj.n / total;
        (scaled[i] < 1.0 ? small : large).push_back(
            static_cast<uint32_t>(i));
      }
      while (!small.empty() && !large.empty()) {
        uint32_t s = small.back();
        small.pop_back();
        uint32_t l = large.back();
        large.pop_back();
        prob[base + s] = static_cast<float>(scaled[s]);
        alias[base + s] = l;
        scaled[l] = scaled[l] + scaled[s] - 1.0;
        (scaled[l] < 1.0 ? small : large).push_back(l);
      }
      for (uint32_t i : large) prob[base + i] = 1.0f;
      for (uint32_t i : small) prob[base + i] = 1.0f;
    }, 256);
  }
};

struct SamplerContext {
  SampleStrategy strategy = SampleStrategy::kRandom;
  int hops = 2;
  int top_k = 4;
  uint64_t seed = 7;
  AliasTables alias;  // built when strategy == kEdgeWeight
  // walks this fragment OWNS (origin here); path[w][h], gid space,
  // kInvalidVid where the walk died (dead end)
  std::vector<uint64_t> walk_ids;            // global walk index
  std::vector<std::vector<vid_t>> paths;     // [local walk][hops+1]
  // walks currently AT this fragment: (walk id, hop just reached, vertex)
  struct Live {
    uint64_t walk;
    int hop;
    vid_t gid;
  };
  std::vector<Live> live;

  void init(const Fragment& frag, const std::vector<oid_t>& starts,
            int hops_, SampleStrategy st, int k, uint64_t seed_) {
    strategy = st;
    hops = hops_;
    top_k = k;
    seed = seed_;
    walk_ids.clear();
    paths.clear();
    live.clear();
    for (size_t i = 0; i < starts.size(); ++i) {
      vid_t lid;
      if (frag.oid2lid(starts[i], &lid) && frag.inner(lid)) {
        vid_t g = frag.lid2gid(lid);
        walk_ids.push_back(i);
        std::vector<vid_t> p(hops + 1, kInvalidVid);
        p[0] = g;
        paths.push_back(std::move(p));
        live.push_back({i, 0, g});
      }
    }
  }
};

class SamplerApp {
 public:
  struct Msg {
    uint64_t walk;
    int32_t hop;     // hop index of `gid` in the walk
    int32_t record;  // 1 = report to origin's path, 0 = advance here
    vid_t gid;
  };

  void PEval(const Fragment& frag, SamplerContext& ctx, MessageManager& mm) {
    if (ctx.strategy == SampleStrategy::kEdgeWeight)
      ctx.alias.build(frag);
    advance(frag, ctx, mm);
  }

  void IncEval(const Fragment& frag, SamplerContext& ctx,
               MessageManager& mm) {
    // collect arrivals + records
    std::vector<SamplerContext::Live> arrivals;
    mm.process_raw<Msg>([&](int, vid_t, Msg m) {
      if (m.record) {
        rec_mutex_.lock();
        records_.push_back(m);
        rec_mutex_.unlock();
      } else {
        arr_mutex_.lock();
        arrivals.push_back({m.walk, m.hop, m.gid});
        arr_mutex_.unlock();
      }
    });
    // apply records to owned paths (walk origin == this fragment)
    for (const Msg& m : records_) {
      auto it = std::lower_bound(ctx.walk_ids.begin(), ctx.walk_ids.end(),
                                 m.walk);
      if (it != ctx.walk_ids.end() && *it == m.walk)
        ctx.paths[it - ctx.walk_ids.begin()][m.hop] = m.gid;
    }
    records_.clear();
    // arrivals JOIN the locally-advancing walks (replacing them dropped
    // every local walk after its first hop)
    ctx.live.insert(ctx.live.end(), arrivals.begin(), arrivals.end());
    advance(frag, ctx, mm);
  }

 private:
  std::mutex rec_mutex_, arr_mutex_;
  std::vector<Msg> records_;

  template <typename RNG>
  vid_t sample_neighbor(const Fragment& frag, const SamplerContext& ctx,
                        vid_t lid, RNG& rng) const {
    auto adj = frag.out_edges(lid);
    if (adj.n == 0) return kInvalidVid;
    size_t pick = 0;
    switch (ctx.strategy) {
      case SampleStrategy::kRandom:
        pick = rng() % adj.n;
        break;
      case SampleStrategy::kEdgeWeight: {
        // O(1) alias draw (Vose): pick a slot, then flip against its prob
        size_t base = adj.dst - frag.oe_dsts().data();
        size_t slot = rng() % adj.n;
        double coin = static_cast<double>(rng() & 0xFFFFFFFFFFFFull) /
                      double(0x1000000000000ull);
        pick = coin < ctx.alias.prob[base + slot]
                   ? slot
                   : ctx.alias.alias[base + slot];
        break;
      }
      case SampleStrategy::kTopK: {
        // uniform among the k heaviest edges (ties by dst gid for
        // determinism, like the reference's sorted index)
        size_t k = std::min<size_t>(ctx.top_k, adj.n);
        std::vector<size_t> idx(adj.n);
        for (size_t i = 0; i < adj.n; ++i) idx[i] = i;
        std::partial_sort(idx.begin(), idx.begin() + k, idx.end(),
                          [&](size_t a, size_t b) {
                            float wa = adj.w ? adj.w[a] : 1.0f;
                            float wb = adj.w ? adj.w[b] : 1.0f;
                            if (wa != wb) return wa > wb;
                            return frag.lid2gid(adj.dst[a]) <
                                   frag.lid2gid(adj.dst[b]);
                          });
        pick = idx[rng() % k];
        break;
      }
    }
    return frag.lid2gid(adj.dst[pick]);
  }

  void advance(const Fragment& frag, SamplerContext& ctx,
               MessageManager& mm) {
    bool any_live = false;
    for (const auto& lv : ctx.live) {
      if (lv.hop >= ctx.hops) continue;
      vid_t lid = frag.gid2lid(lv.gid);
      if (lid == kInvalidVid || !frag.inner(lid)) continue;
      // per-(walk, hop) deterministic RNG stream
      std::mt19937_64 rng(ctx.seed * 0x9e3779b97f4a7c15ULL + lv.walk * 1000003ULL +
                          static_cast<uint64_t>(lv.hop));
      vid_t nxt = sample_neighbor(frag, ctx, lid, rng);
      if (nxt == kInvalidVid) continue;  // dead end: walk stops
      int nhop = lv.hop + 1;
      // record the step at the walk's origin
      fid_t origin = origin_of(frag, ctx, lv.walk);
      if (origin == frag.fid()) {
        auto it = std::lower_bound(ctx.walk_ids.begin(), ctx.walk_ids.end(),
                                   lv.walk);
        ctx.paths[it - ctx.walk_ids.begin()][nhop] = nxt;
      } else {
        mm.send_to_fragment(0, origin, /*routing gid (unused)*/ 0,
                            Msg{lv.walk, nhop, 1, nxt});
      }
      // advance the walk
      if (nhop < ctx.hops) {
        fid_t owner = frag.parser().fid(nxt);
        if (owner == frag.fid()) {
          next_live_.push_back({lv.walk, nhop, nxt});
          any_live = true;
        } else {
          mm.send_to_fragment(0, owner, nxt, Msg{lv.walk, nhop, 0, nxt});
          any_live = true;
        }
      }
    }
    ctx.live = std::move(next_live_);
    next_live_.clear();
    if (any_live || !ctx.live.empty()) mm.force_continue();
  }

  fid_t origin_of(const Fragment& frag, const SamplerContext& ctx,
                  uint64_t walk) const {
    // walks are issued per-origin; origin = owner of the start vertex.
    // The start oid isn't carried, so the origin is resolved from the walk
    // table replicated at init: ranks only look up walks they own, and
    // remote records carry the origin in origin_map_.
    return origin_map_.empty() ? frag.fid()
                               : origin_map_[walk];
  }

 public:
  // set by the driver before Query: owner fragment of each walk's start
  std::vector<fid_t> origin_map_;

 private:
  std::vector<SamplerContext::Live> next_live_;
};

}  // namespace grapehip
