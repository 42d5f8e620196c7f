// grapehip — betweenness centrality (single-source Brandes), CPU path.
// Reference parity: examples/analytical_apps/bc/{staged_bc,staged_bc_bfs}.h
// (two staged phases sharing one context): forward level-synchronous BFS
// accumulating shortest-path counts sigma, then a backward dependency sweep
// one depth level per BSP round. Output is the source's dependency
// delta(v) = sigma(v) * sum_{w in succ(v)} (1 + delta(w)) / sigma(w)
// (staged_bc.h:197-207 — centrality_value after the path_num multiply).
#pragma once

#include <cstdint>
#include <limits>
#include <vector>

#include "../core/bitset.hpp"
#include "../core/fragment.hpp"
#include "../core/message_manager.hpp"

namespace grapehip {

inline void atomic_add_double(std::atomic<double>& slot, double v) {
  double cur = slot.load(std::memory_order_relaxed);
  while (!slot.compare_exchange_weak(cur, cur + v,
                                     std::memory_order_relaxed)) {
  }
}

struct BCContext {
  oid_t source;
  std::vector<std::atomic<int64_t>> depth;   // tvnum
  std::vector<std::atomic<double>> sigma;    // tvnum (outer = send-side acc)
  std::vector<std::atomic<double>> delta;    // ivnum (owned dependencies)
  DenseVertexSet curr, next;
  int64_t level = 0;
  int phase = 0;  // 0 = forward BFS, 1 = mirror depth refresh, 2 = backward
  int64_t back_depth = 0;

  void init(const Fragment& frag, oid_t src) {
    source = src;
    depth = std::vector<std::atomic<int64_t>>(frag.tvnum());
    sigma = std::vector<std::atomic<double>>(frag.tvnum());
    delta = std::vector<std::atomic<double>>(frag.ivnum());
    for (auto& d : depth)
      d.store(std::numeric_limits<int64_t>::max(), std::memory_order_relaxed);
    for (auto& s : sigma) s.store(0.0, std::memory_order_relaxed);
    for (auto& d : delta) d.store(0.0, std::memory_order_relaxed);
    curr.init(frag.ivnum());
    next.init(frag.ivnum());
    level = 0;
    phase = 0;
    back_depth = 0;
  }
};

class BCApp {
 public:
  struct FwdMsg {
    int64_t d;
    double sig;
  };

  void PEval(const Fragment& frag, BCContext& ctx, MessageManager& mm) {
    vid_t src_lid;
    if (frag.oid2lid(ctx.source, &src_lid) && frag.inner(src_lid)) {
      ctx.depth[src_lid].store(0, std::memory_order_relaxed);
      ctx.sigma[src_lid].store(1.0, std::memory_order_relaxed);
      ctx.curr.insert(src_lid);
    }
    forward(frag, ctx, mm);
    mm.force_continue();  // even an empty frontier must reach the barrier
  }

  void IncEval(const Fragment& frag, BCContext& ctx, MessageManager& mm) {
    if (ctx.phase == 0) {
      mm.process<FwdMsg>([&](int, vid_t lid, FwdMsg m) {
        if (atomic_min_update(ctx.depth[lid], m.d)) ctx.curr.insert(lid);
        if (ctx.depth[lid].load(std::memory_order_relaxed) == m.d)
          atomic_add_double(ctx.sigma[lid], m.sig);
      });
      bool local_empty = ctx.curr.empty();
      bool global_done =
          mm.comm() ? !mm.comm()->allreduce_or(!local_empty) : local_empty;
      if (!global_done) {
        forward(frag, ctx, mm);
        mm.force_continue();
        return;
      }
      // forward converged: publish final depths of my border vertices to
      // the fragments mirroring them (local outer copies only hold the
      // depth at which THIS rank first relaxed them — possibly stale)
      if (mm.comm()) {
        for (fid_t f = 0; f < static_cast<fid_t>(frag.fnum()); ++f) {
          if (f == frag.fid()) continue;
          for (vid_t lid : frag.mirrors(f))
            mm.send_to_fragment(
                0, f, frag.lid2gid(lid),
                ctx.depth[lid].load(std::memory_order_relaxed));
        }
      }
      ctx.phase = 1;
      int64_t local_max = 0;
      for (vid_t v = 0; v < frag.ivnum(); ++v) {
        int64_t d = ctx.depth[v].load(std::memory_order_relaxed);
        if (d != std::numeric_limits<int64_t>::max() && d > local_max)
          local_max = d;
      }
      ctx.back_depth =
          mm.comm() ? static_cast<int64_t>(mm.comm()->allreduce_max_double(
                          static_cast<double>(local_max)))
                    : local_max;
      mm.force_continue();  // run the refresh/backward rounds even if 0
      return;
    }
    if (ctx.phase == 1) {
      // apply mirror depth refresh (targets OUR outer copies)
      mm.process_any<int64_t>([&](int, vid_t lid, int64_t d) {
        ctx.depth[lid].store(d, std::memory_order_relaxed);
      });
      ctx.phase = 2;
    } else {
      // backward: one depth level per round, deepest first
      mm.process<double>([&](int, vid_t lid, double accum) {
        // delta(u) += sigma(u) * accum (owner-side multiply: outer copies
        // of sigma are incomplete, so the wire carries accum only)
        atomic_add_double(ctx.delta[lid],
                          ctx.sigma[lid].load(std::memory_order_relaxed) *
                              accum);
      });
    }
    const int64_t d = ctx.back_depth;
    if (d < 1) return;
    parallel_for_tid(0, frag.ivnum(), [&](int tid, size_t vs) {
      vid_t v = static_cast<vid_t>(vs);
      if (ctx.depth[v].load(std::memory_order_relaxed) != d) return;
      double accum =
          (1.0 + ctx.delta[v].load(std::memory_order_relaxed)) /
          ctx.sigma[v].load(std::memory_order_relaxed);
      auto adj = frag.out_edges(v);
      for (size_t i = 0; i < adj.n; ++i) {
        vid_t u = adj.dst[i];
        if (ctx.depth[u].load(std::memory_order_relaxed) != d - 1) continue;
        if (frag.inner(u)) {
          atomic_add_double(ctx.delta[u],
                            ctx.sigma[u].load(std::memory_order_relaxed) *
                                accum);
        } else {
          mm.sync_state_on_outer_vertex(tid, u, accum);
        }
      }
    }, 256);
    --ctx.back_depth;
    // back_depth was allreduced, so every rank counts down in lockstep;
    // the final round's in-flight messages still get delivered (termination
    // requires a silent round) and are applied by the process() above.
    if (ctx.back_depth >= 1) mm.force_continue();
  }

 private:
  void forward(const Fragment& frag, BCContext& ctx, MessageManager& mm) {
    const int64_t nd = ctx.level + 1;
    ctx.curr.parallel_iterate_tid([&](int tid, vid_t v) {
      double sig = ctx.sigma[v].load(std::memory_order_relaxed);
      auto adj = frag.out_edges(v);
      for (size_t i = 0; i < adj.n; ++i) {
        vid_t u = adj.dst[i];
        if (frag.inner(u)) {
          if (atomic_min_update(ctx.depth[u], nd)) ctx.next.insert(u);
          if (ctx.depth[u].load(std::memory_order_relaxed) == nd)
            atomic_add_double(ctx.sigma[u], sig);
        } else {
          // owner merges: depth min + sigma sum
          if (ctx.depth[u].load(std::memory_order_relaxed) > nd)
            ctx.depth[u].store(nd, std::memory_order_relaxed);
          mm.sync_state_on_outer_vertex(tid, u, FwdMsg{nd, sig});
        }
      }
    });
    ctx.curr.clear();
    ctx.curr.swap(ctx.next);
    ctx.level = nd;
  }
};

}  // namespace grapehip
