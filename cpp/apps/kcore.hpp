// grapehip — k-core + core decomposition, CPU path.
// Reference parity: examples/analytical_apps/kcore/kcore.h (iterative
// peeling of vertices with residual degree < k; result = membership) and
// examples/analytical_apps/core_decomposition/core_decomposition.h
// (coreness per vertex). Degrees count stored multiplicity, self-loops
// excluded. Coreness uses the h-index fixpoint (est[v] <- H(est of
// neighbors)), which converges to the peeling coreness.
#pragma once

#include <algorithm>
#include <cstdint>
#include <limits>
#include <vector>

#include "../core/bitset.hpp"
#include "../core/fragment.hpp"
#include "../core/message_manager.hpp"

namespace grapehip {

struct KCoreContext {
  int k = 0;
  std::vector<std::atomic<int64_t>> deg;  // residual degree, ivnum
  std::vector<uint8_t> removed;           // ivnum
  DenseVertexSet to_remove;
  bool first = true;

  void init(const Fragment& frag, int k_) {
    k = k_;
    deg = std::vector<std::atomic<int64_t>>(frag.ivnum());
    removed.assign(frag.ivnum(), 0);
    to_remove.init(frag.ivnum());
    first = true;
    parallel_for(0, frag.ivnum(), [&](size_t vs) {
      vid_t v = static_cast<vid_t>(vs);
      auto adj = frag.out_edges(v);
      int64_t d = 0;
      for (size_t i = 0; i < adj.n; ++i)
        if (adj.dst[i] != v) ++d;
      ctx_store(v, d);
    }, 2048);
  }
  void ctx_store(vid_t v, int64_t d) {
    deg[v].store(d, std::memory_order_relaxed);
  }
};

// message = decrement count for an owned vertex
class KCoreApp {
 public:
  void PEval(const Fragment& frag, KCoreContext& ctx, MessageManager& mm) {
    sweep(frag, ctx, mm);
  }

  void IncEval(const Fragment& frag, KCoreContext& ctx, MessageManager& mm) {
    mm.process<int64_t>([&](int, vid_t lid, int64_t dec) {
      ctx.deg[lid].fetch_sub(dec, std::memory_order_relaxed);
    });
    sweep(frag, ctx, mm);
  }

 private:
  void sweep(const Fragment& frag, KCoreContext& ctx, MessageManager& mm) {
    // collect vertices that fell below k
    std::atomic<size_t> n_removed{0};
    parallel_for(0, frag.ivnum(), [&](size_t vs) {
      vid_t v = static_cast<vid_t>(vs);
      if (!ctx.removed[v] &&
          ctx.deg[v].load(std::memory_order_relaxed) < ctx.k) {
        ctx.removed[v] = 1;
        ctx.to_remove.insert(v);
        n_removed.fetch_add(1, std::memory_order_relaxed);
      }
    }, 2048);
    // per-peer decrement aggregation happens naturally via messages; local
    // neighbors decremented directly
    ctx.to_remove.parallel_iterate_tid([&](int tid, vid_t v) {
      auto adj = frag.out_edges(v);
      for (size_t i = 0; i < adj.n; ++i) {
        vid_t u = adj.dst[i];
        if (u == v) continue;
        if (frag.inner(u)) {
          ctx.deg[u].fetch_sub(1, std::memory_order_relaxed);
        } else {
          mm.sync_state_on_outer_vertex(tid, u, static_cast<int64_t>(1));
        }
      }
    });
    ctx.to_remove.clear();
    if (n_removed.load() > 0) mm.force_continue();
  }
};

// -- core decomposition (coreness via h-index fixpoint) ---------------------

struct CoreDecompContext {
  std::vector<std::atomic<int64_t>> est;  // over tvnum (outer mirrored)
  DenseVertexSet changed;

  void init(const Fragment& frag) {
    est = std::vector<std::atomic<int64_t>>(frag.tvnum());
    changed.init(frag.ivnum());
    parallel_for(0, frag.ivnum(), [&](size_t vs) {
      vid_t v = static_cast<vid_t>(vs);
      auto adj = frag.out_edges(v);
      int64_t d = 0;
      for (size_t i = 0; i < adj.n; ++i)
        if (adj.dst[i] != v) ++d;
      est[v].store(d, std::memory_order_relaxed);
    }, 2048);
    for (vid_t v = frag.ivnum(); v < frag.tvnum(); ++v)
      est[v].store(std::numeric_limits<int64_t>::max(),
                   std::memory_order_relaxed);
  }
};

class CoreDecompApp {
 public:
  void PEval(const Fragment& frag, CoreDecompContext& ctx,
             MessageManager& mm) {
    // publish initial estimates of border vertices, then iterate
    broadcast_borders(frag, ctx, mm, true);
    mm.force_continue();
  }

  void IncEval(const Fragment& frag, CoreDecompContext& ctx,
               MessageManager& mm) {
    // mirror refresh: owners pushed their inner estimates to fragments
    // holding them as outer copies
    mm.process_any<int64_t>([&](int, vid_t lid, int64_t e) {
      ctx.est[lid].store(e, std::memory_order_relaxed);
    });
    // h-index sweep over all owned vertices
    int nt = mm.n_threads();
    std::vector<std::vector<int64_t>> scratch(nt);
    std::atomic<size_t> n_changed{0};
    parallel_for_tid(0, frag.ivnum(), [&](int tid, size_t vs) {
      vid_t v = static_cast<vid_t>(vs);
      auto& vals = scratch[tid];
      vals.clear();
      auto adj = frag.out_edges(v);
      for (size_t i = 0; i < adj.n; ++i) {
        vid_t u = adj.dst[i];
        if (u == v) continue;
        vals.push_back(ctx.est[u].load(std::memory_order_relaxed));
      }
      // h-index: largest h with >= h values >= h
      std::sort(vals.begin(), vals.end(), std::greater<int64_t>());
      int64_t h = 0;
      for (size_t i = 0; i < vals.size(); ++i)
        if (vals[i] >= static_cast<int64_t>(i + 1))
          h = static_cast<int64_t>(i + 1);
        else
          break;
      if (h < ctx.est[v].load(std::memory_order_relaxed)) {
        ctx.est[v].store(h, std::memory_order_relaxed);
        ctx.changed.insert(v);
        n_changed.fetch_add(1, std::memory_order_relaxed);
      }
    }, 512);
    // push changed border estimates to mirrors
    broadcast_changed(frag, ctx, mm);
    ctx.changed.clear();
    bool any = n_changed.load() > 0;
    bool global_any = mm.comm() ? mm.comm()->allreduce_or(any) : any;
    if (global_any) mm.force_continue();
  }

 private:
  void broadcast_borders(const Fragment& frag, CoreDecompContext& ctx,
                         MessageManager& mm, bool all) {
    (void)all;
    if (!mm.comm()) return;
    for (fid_t f = 0; f < static_cast<fid_t>(frag.fnum()); ++f) {
      if (f == frag.fid()) continue;
      for (vid_t lid : frag.mirrors(f))
        mm.send_to_fragment(0, f, frag.lid2gid(lid),
                            ctx.est[lid].load(std::memory_order_relaxed));
    }
  }
  void broadcast_changed(const Fragment& frag, CoreDecompContext& ctx,
                         MessageManager& mm) {
    if (!mm.comm()) return;
    for (fid_t f = 0; f < static_cast<fid_t>(frag.fnum()); ++f) {
      if (f == frag.fid()) continue;
      for (vid_t lid : frag.mirrors(f))
        if (ctx.changed.exist(lid))
          mm.send_to_fragment(0, f, frag.lid2gid(lid),
                              ctx.est[lid].load(std::memory_order_relaxed));
    }
  }
};

}  // namespace grapehip
