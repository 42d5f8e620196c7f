// grapehip — CDLP (community detection by synchronous label propagation).
// Reference parity: examples/analytical_apps/cdlp/cdlp.h (+cdlp_utils.h):
// fixed cdlp_mr rounds; new label = most frequent label among neighbors
// (directed: in+out multiset — an edge in both directions counts twice),
// tie-break = smallest label; labels are vertex oids (int64). Outer labels
// are refreshed each round with a dense mirror exchange (the reference
// sends along edges; dense batch sync is equivalent for full refresh).
#pragma once

#include <algorithm>
#include <vector>

#include "../core/fragment.hpp"
#include "../core/message_manager.hpp"

namespace grapehip {

struct CDLPContext {
  int max_iters;
  int iter = 0;
  std::vector<int64_t> label;       // over tvnum
  std::vector<int64_t> next_label;  // over ivnum

  void init(const Fragment& frag, int mr) {
    max_iters = mr;
    iter = 0;
    label.resize(frag.tvnum());
    parallel_for(0, frag.tvnum(), [&](size_t v) {
      label[v] = frag.lid2oid(static_cast<vid_t>(v));
    }, 4096);
    next_label.assign(frag.ivnum(), 0);
  }
};

class CDLPApp {
 public:
  void PEval(const Fragment& frag, CDLPContext& ctx, MessageManager& mm) {
    if (ctx.max_iters > 0) mm.force_continue();
  }

  void IncEval(const Fragment& frag, CDLPContext& ctx, MessageManager& mm) {
    const bool use_in = frag.directed() && frag.has_in_csr();
    // scratch per thread to avoid per-vertex allocation
    int nt = mm.n_threads();
    std::vector<std::vector<int64_t>> scratch(nt);

    parallel_for_tid(0, frag.ivnum(), [&](int tid, size_t vs) {
      vid_t v = static_cast<vid_t>(vs);
      auto& labs = scratch[tid];
      labs.clear();
      auto gather = [&](const Fragment::AdjRange& adj) {
        for (size_t i = 0; i < adj.n; ++i) labs.push_back(ctx.label[adj.dst[i]]);
      };
      gather(frag.out_edges(v));
      if (use_in) gather(frag.in_edges(v));
      if (labs.empty()) {
        ctx.next_label[v] = ctx.label[v];
        return;
      }
      std::sort(labs.begin(), labs.end());
      // mode with min-label tie-break: scan runs; first max-length run wins
      int64_t best = labs[0], cur = labs[0];
      size_t best_n = 0, cur_n = 0;
      for (size_t i = 0; i < labs.size(); ++i) {
        if (labs[i] == cur) {
          ++cur_n;
        } else {
          if (cur_n > best_n) { best_n = cur_n; best = cur; }
          cur = labs[i];
          cur_n = 1;
        }
      }
      if (cur_n > best_n) { best_n = cur_n; best = cur; }
      ctx.next_label[v] = best;
    }, 512);

    parallel_for(0, frag.ivnum(), [&](size_t v) {
      ctx.label[v] = ctx.next_label[v];
    }, 8192);

    // dense refresh of outer labels
    if (frag.fnum() > 1 && mm.comm()) {
      int fnum = frag.fnum();
      std::vector<std::string> send(fnum);
      for (int f = 0; f < fnum; ++f) {
        const auto& mir = frag.mirrors(f);
        std::vector<int64_t> vals(mir.size());
        for (size_t i = 0; i < mir.size(); ++i) vals[i] = ctx.label[mir[i]];
        send[f].assign(reinterpret_cast<const char*>(vals.data()),
                       vals.size() * sizeof(int64_t));
      }
      auto recv = mm.comm()->exchange_all(send);
      for (int f = 0; f < fnum; ++f) {
        auto [b, e] = frag.outer_range(f);
        const int64_t* vals = reinterpret_cast<const int64_t*>(recv[f].data());
        for (vid_t u = b; u < e; ++u) ctx.label[u] = vals[u - b];
      }
    }

    if (++ctx.iter < ctx.max_iters) mm.force_continue();
  }
};

}  // namespace grapehip
