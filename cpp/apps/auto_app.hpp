// grapehip — auto-app support (reference parity:
// grape/parallel/auto_parallel_message_manager.h + grape/app/auto_app_base.h
// + grape/parallel/sync_buffer.h). An auto-app registers a per-vertex
// buffer and an aggregate op; after every round the engine automatically
// ships updated border values to the fragments mirroring them and applies
// the aggregate on arrival — PEval/IncEval never touch the message API.
// Shipped example: SSSPAuto (reference sssp_auto.h, aggregate-min).
#pragma once

#include <limits>
#include <vector>

#include "../core/bitset.hpp"
#include "../core/fragment.hpp"
#include "../core/message_manager.hpp"

namespace grapehip {

enum class SyncOp { kMin, kMax, kSum };

// Dense per-vertex sync buffer over tvnum; tracks which OUTER vertices this
// rank updated (their owners must aggregate) and which INNER border
// vertices changed (their mirrors must refresh).
template <typename T>
class AutoSyncBuffer {
 public:
  void init(const Fragment& frag, T init_val, SyncOp op) {
    frag_ = &frag;
    op_ = op;
    data_.assign(frag.tvnum(), init_val);
    updated_.init(frag.tvnum());
  }
  T& operator[](vid_t v) { return data_[v]; }
  const T& operator[](vid_t v) const { return data_[v]; }
  // mark v's value as changed this round (any lid)
  void mark(vid_t v) { updated_.insert(v); }

  // Round boundary: ship updates. Outer updates -> owner (aggregate);
  // inner border updates -> mirrors (overwrite). Termination falls out of
  // the message manager's silent-round rule.
  void flush(MessageManager& mm) {
    const Fragment& frag = *frag_;
    // outer -> owner
    for (vid_t v = frag.ivnum(); v < frag.tvnum(); ++v)
      if (updated_.exist(v))
        mm.sync_state_on_outer_vertex(0, v, data_[v]);
    // inner border -> mirrors
    for (fid_t f = 0; f < static_cast<fid_t>(frag.fnum()); ++f) {
      if (f == frag.fid()) continue;
      for (vid_t lid : frag.mirrors(f))
        if (updated_.exist(lid))
          mm.send_to_fragment(0, f, frag.lid2gid(lid), data_[lid]);
    }
    updated_.clear();
  }

  // apply received values; marks owned vertices whose aggregate improved
  // so the app can treat them as its next frontier
  template <typename OnChange>
  void absorb(MessageManager& mm, OnChange&& on_change) {
    mm.process_any<T>([&](int, vid_t lid, T val) {
      T& slot = data_[lid];
      bool ch = false;
      switch (op_) {
        case SyncOp::kMin:
          if (val < slot) {
            slot = val;
            ch = true;
          }
          break;
        case SyncOp::kMax:
          if (val > slot) {
            slot = val;
            ch = true;
          }
          break;
        case SyncOp::kSum:
          slot += val;
          ch = true;
          break;
      }
      if (ch) on_change(lid);
    });
  }

 private:
  const Fragment* frag_ = nullptr;
  SyncOp op_ = SyncOp::kMin;
  std::vector<T> data_;
  DenseVertexSet updated_;
};

// SSSP as an auto-app (reference sssp_auto.h): the app only relaxes local
// edges and marks changes; the sync buffer moves values between fragments.
struct SSSPAutoContext {
  oid_t source;
  AutoSyncBuffer<double> dist;
  std::vector<vid_t> frontier;

  void init(const Fragment& frag, oid_t src) {
    source = src;
    dist.init(frag, std::numeric_limits<double>::max(), SyncOp::kMin);
    frontier.clear();
  }
};

class SSSPAutoApp {
 public:
  void PEval(const Fragment& frag, SSSPAutoContext& ctx,
             MessageManager& mm) {
    vid_t src_lid;
    if (frag.oid2lid(ctx.source, &src_lid) && frag.inner(src_lid)) {
      ctx.dist[src_lid] = 0.0;
      ctx.frontier.push_back(src_lid);
    }
    relax(frag, ctx);
    ctx.dist.flush(mm);
  }

  void IncEval(const Fragment& frag, SSSPAutoContext& ctx,
               MessageManager& mm) {
    ctx.dist.absorb(mm, [&](vid_t lid) {
      if (frag.inner(lid)) ctx.frontier.push_back(lid);
    });
    relax(frag, ctx);
    ctx.dist.flush(mm);
  }

 private:
  // serial local fixpoint — auto-apps trade speed for the simplest
  // possible app body; the Parallel SSSP app is the fast path
  void relax(const Fragment& frag, SSSPAutoContext& ctx) {
    std::vector<vid_t> next;
    while (!ctx.frontier.empty()) {
      for (vid_t v : ctx.frontier) {
        double dv = ctx.dist[v];
        auto adj = frag.out_edges(v);
        for (size_t i = 0; i < adj.n; ++i) {
          vid_t u = adj.dst[i];
          double nd = dv + (adj.w ? adj.w[i] : 1.0f);
          if (nd < ctx.dist[u]) {
            ctx.dist[u] = nd;
            ctx.dist.mark(u);
            if (frag.inner(u)) next.push_back(u);
          }
        }
      }
      ctx.frontier.swap(next);
      next.clear();
    }
  }
};

}  // namespace grapehip
