// grapehip — SSSP (parallel Bellman-Ford / frontier relaxation), CPU path.
// Reference parity: examples/analytical_apps/sssp/sssp.h (atomic_min relax,
// DenseVertexSet frontier, MessageStrategy kSyncOnOuterVertex, aggregate-min
// on receive). Output: double distance, unreachable = DBL_MAX (reference
// prints std::numeric_limits<double>::max() as-is; we match).
#pragma once

#include <limits>
#include <vector>

#include "../core/bitset.hpp"
#include "../core/fragment.hpp"
#include "../core/message_manager.hpp"

namespace grapehip {

struct SSSPContext {
  oid_t source;
  std::vector<std::atomic<double>> dist;  // over tvnum
  std::vector<double> sent;               // last dist sent per outer vertex
  DenseVertexSet curr, next;

  void init(const Fragment& frag, oid_t src) {
    source = src;
    dist = std::vector<std::atomic<double>>(frag.tvnum());
    for (auto& d : dist)
      d.store(std::numeric_limits<double>::max(), std::memory_order_relaxed);
    sent.assign(frag.ovnum(), std::numeric_limits<double>::max());
    curr.init(frag.ivnum());
    next.init(frag.ivnum());
  }
};

class SSSPApp {
 public:
  void PEval(const Fragment& frag, SSSPContext& ctx, MessageManager& mm) {
    vid_t src_lid;
    if (frag.oid2lid(ctx.source, &src_lid) && frag.inner(src_lid)) {
      ctx.dist[src_lid].store(0.0, std::memory_order_relaxed);
      ctx.curr.insert(src_lid);
    }
    relax(frag, ctx, mm);
  }

  void IncEval(const Fragment& frag, SSSPContext& ctx, MessageManager& mm) {
    mm.process<double>([&](int, vid_t lid, double d) {
      if (atomic_min_update(ctx.dist[lid], d)) ctx.curr.insert(lid);
    });
    relax(frag, ctx, mm);
  }

 private:
  void relax(const Fragment& frag, SSSPContext& ctx, MessageManager& mm) {
    ctx.curr.parallel_iterate([&](vid_t v) {
      double dv = ctx.dist[v].load(std::memory_order_relaxed);
      auto adj = frag.out_edges(v);
      for (size_t i = 0; i < adj.n; ++i) {
        vid_t u = adj.dst[i];
        double nd = dv + (adj.w ? static_cast<double>(adj.w[i]) : 1.0);
        if (atomic_min_update(ctx.dist[u], nd)) {
          if (frag.inner(u)) ctx.next.insert(u);
          // outer improvements are flushed after the sweep (one aggregated
          // min message per improved outer vertex per round).
        }
      }
    });
    const vid_t iv = frag.ivnum(), tv = frag.tvnum();
    if (tv > iv) {
      parallel_for_tid(iv, tv, [&](int tid, size_t u) {
        double d = ctx.dist[u].load(std::memory_order_relaxed);
        if (d < ctx.sent[u - iv]) {
          ctx.sent[u - iv] = d;
          mm.sync_state_on_outer_vertex(tid, static_cast<vid_t>(u), d);
        }
      }, 2048);
    }
    ctx.curr.clear();
    ctx.curr.swap(ctx.next);
    if (ctx.curr.count() > 0) mm.force_continue();
  }
};

}  // namespace grapehip
