// grapehip — BFS (level-synchronous), CPU path.
// Reference parity: examples/analytical_apps/bfs/bfs.h (ParallelAppBase,
// message = new depth on outer vertices, frontier DenseVertexSet,
// ForceContinue while a local frontier remains).
// Output: int64 depth from source, unreachable = INT64_MAX (bfs_context.h).
#pragma once

#include <limits>
#include <vector>

#include "../core/bitset.hpp"
#include "../core/fragment.hpp"
#include "../core/message_manager.hpp"

namespace grapehip {

struct BFSContext {
  oid_t source;
  std::vector<std::atomic<int64_t>> depth;  // over tvnum
  DenseVertexSet curr, next;
  int64_t level = 0;  // depth of the current frontier

  void init(const Fragment& frag, oid_t src) {
    source = src;
    depth = std::vector<std::atomic<int64_t>>(frag.tvnum());
    for (auto& d : depth)
      d.store(std::numeric_limits<int64_t>::max(), std::memory_order_relaxed);
    curr.init(frag.ivnum());
    next.init(frag.ivnum());
    level = 0;
  }
};

class BFSApp {
 public:
  void PEval(const Fragment& frag, BFSContext& ctx, MessageManager& mm) {
    vid_t src_lid;
    if (frag.oid2lid(ctx.source, &src_lid) && frag.inner(src_lid)) {
      ctx.depth[src_lid].store(0, std::memory_order_relaxed);
      ctx.curr.insert(src_lid);
    }
    expand(frag, ctx, mm);
  }

  void IncEval(const Fragment& frag, BFSContext& ctx, MessageManager& mm) {
    mm.process<int64_t>([&](int, vid_t lid, int64_t d) {
      if (atomic_min_update(ctx.depth[lid], d)) ctx.curr.insert(lid);
    });
    expand(frag, ctx, mm);
  }

 private:
  // Level-synchronous: every BSP round advances exactly one level; the
  // frontier (local inserts + incoming messages) is uniformly at ctx.level.
  void expand(const Fragment& frag, BFSContext& ctx, MessageManager& mm) {
    const int64_t next_depth = ctx.level + 1;
    ctx.curr.parallel_iterate_tid([&](int tid, vid_t v) {
      auto adj = frag.out_edges(v);
      for (size_t i = 0; i < adj.n; ++i) {
        vid_t u = adj.dst[i];
        if (atomic_min_update(ctx.depth[u], next_depth)) {
          if (frag.inner(u)) {
            ctx.next.insert(u);
          } else {
            mm.sync_state_on_outer_vertex(tid, u, next_depth);
          }
        }
      }
    });
    ctx.curr.clear();
    ctx.curr.swap(ctx.next);
    ctx.level = next_depth;
    if (ctx.curr.count() > 0) mm.force_continue();
  }
};

}  // namespace grapehip
