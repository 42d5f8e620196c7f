// grapehip — WCC (weakly connected components, min-label propagation).
// Reference parity: examples/analytical_apps/wcc/wcc.h (parallel label-min
// spread, messages = improved labels on outer vertices). Labels are oids so
// output is partitioner-independent; LDBC validates the partition up to
// isomorphism (misc/wcc_check.cc). Directed inputs propagate over both edge
// directions (weak connectivity) — requires the in-CSR.
#pragma once

#include <limits>
#include <vector>

#include "../core/bitset.hpp"
#include "../core/fragment.hpp"
#include "../core/message_manager.hpp"

namespace grapehip {

struct WCCContext {
  std::vector<std::atomic<int64_t>> label;  // over tvnum (oid labels)
  std::vector<int64_t> sent;                // per outer: last label sent
  DenseVertexSet curr, next;

  void init(const Fragment& frag) {
    label = std::vector<std::atomic<int64_t>>(frag.tvnum());
    parallel_for(0, frag.tvnum(), [&](size_t v) {
      label[v].store(frag.lid2oid(static_cast<vid_t>(v)),
                     std::memory_order_relaxed);
    }, 4096);
    sent.assign(frag.ovnum(), std::numeric_limits<int64_t>::max());
    curr.init(frag.ivnum());
    next.init(frag.ivnum());
  }
};

class WCCApp {
 public:
  void PEval(const Fragment& frag, WCCContext& ctx, MessageManager& mm) {
    parallel_for(0, frag.ivnum(), [&](size_t v) {
      ctx.curr.insert(static_cast<vid_t>(v));
    }, 4096);
    spread(frag, ctx, mm);
  }

  void IncEval(const Fragment& frag, WCCContext& ctx, MessageManager& mm) {
    mm.process<int64_t>([&](int, vid_t lid, int64_t lab) {
      if (atomic_min_update(ctx.label[lid], lab)) ctx.curr.insert(lid);
    });
    spread(frag, ctx, mm);
  }

 private:
  void spread(const Fragment& frag, WCCContext& ctx, MessageManager& mm) {
    bool use_in = frag.directed() && frag.has_in_csr();
    ctx.curr.parallel_iterate([&](vid_t v) {
      int64_t lv = ctx.label[v].load(std::memory_order_relaxed);
      auto push = [&](const Fragment::AdjRange& adj) {
        for (size_t i = 0; i < adj.n; ++i) {
          vid_t u = adj.dst[i];
          if (atomic_min_update(ctx.label[u], lv)) {
            if (frag.inner(u)) ctx.next.insert(u);
          }
        }
      };
      push(frag.out_edges(v));
      if (use_in) push(frag.in_edges(v));
    });
    // flush improved outer labels once per round
    const vid_t iv = frag.ivnum(), tv = frag.tvnum();
    if (tv > iv) {
      parallel_for_tid(iv, tv, [&](int tid, size_t u) {
        int64_t l = ctx.label[u].load(std::memory_order_relaxed);
        if (l < ctx.sent[u - iv]) {
          ctx.sent[u - iv] = l;
          mm.sync_state_on_outer_vertex(tid, static_cast<vid_t>(u), l);
        }
      }, 2048);
    }
    ctx.curr.clear();
    ctx.curr.swap(ctx.next);
    if (ctx.curr.count() > 0) mm.force_continue();
  }
};

}  // namespace grapehip
