// grapehip — k-clique counting, CPU path.
// Reference parity: examples/analytical_apps/kclique/kclique.h (+utils):
// global count of k-cliques via recursive expansion over a gid-ordered
// oriented simple adjacency (every clique is enumerated exactly once from
// its smallest-gid vertex). Multi-fragment runs replicate the oriented
// adjacency (it halves the edge set) with one all-to-all — the CPU
// analogue of the GPU LCC's oriented-CSR allgather.
#pragma once

#include <algorithm>
#include <cstring>
#include <vector>

#include "../core/fragment.hpp"
#include "../core/message_manager.hpp"

namespace grapehip {

struct KCliqueContext {
  int k = 3;
  uint64_t clique_num = 0;  // global (after IncEval)
  // oriented adjacency over ALL vertices, indexed by gid via parser
  std::vector<std::vector<std::vector<vid_t>>> adj;  // [fid][lid] sorted gids

  void init(const Fragment& frag, int k_) {
    k = k_;
    clique_num = 0;
    adj.assign(frag.fnum(), {});
  }
};

class KCliqueApp {
 public:
  void PEval(const Fragment& frag, KCliqueContext& ctx, MessageManager& mm) {
    // oriented rows for owned vertices: {u : gid(u) > gid(v)}, dedup'd
    auto& mine = ctx.adj[frag.fid()];
    mine.assign(frag.ivnum(), {});
    parallel_for(0, frag.ivnum(), [&](size_t vs) {
      vid_t v = static_cast<vid_t>(vs);
      vid_t vg = frag.lid2gid(v);
      auto& row = mine[vs];
      auto adj = frag.out_edges(v);
      for (size_t i = 0; i < adj.n; ++i) {
        vid_t g = frag.lid2gid(adj.dst[i]);
        if (g > vg) row.push_back(g);
      }
      std::sort(row.begin(), row.end());
      row.erase(std::unique(row.begin(), row.end()), row.end());
    }, 512);
    mm.force_continue();  // count next round, after replication
  }

  void IncEval(const Fragment& frag, KCliqueContext& ctx,
               MessageManager& mm) {
    if (done_) return;
    done_ = true;
    replicate(frag, ctx, mm);
    count(frag, ctx, mm);
  }

 private:
  bool done_ = false;

  void replicate(const Fragment& frag, KCliqueContext& ctx,
                 MessageManager& mm) {
    if (!mm.comm() || frag.fnum() == 1) return;
    int fnum = frag.fnum();
    // serialize my rows once, broadcast-style exchange
    InArchive ar;
    const auto& mine = ctx.adj[frag.fid()];
    ar.add(static_cast<uint64_t>(mine.size()));
    for (const auto& row : mine) {
      ar.add(static_cast<uint32_t>(row.size()));
      ar.add_bytes(row.data(), row.size() * sizeof(vid_t));
    }
    std::string blob = ar.release();
    std::vector<std::string> send(fnum, blob);
    send[frag.fid()].clear();
    auto recv = mm.comm()->exchange_all(send);
    for (int f = 0; f < fnum; ++f) {
      if (f == static_cast<int>(frag.fid())) continue;
      const char* p = recv[f].data();
      uint64_t n;
      std::memcpy(&n, p, 8);
      p += 8;
      auto& rows = ctx.adj[f];
      rows.assign(n, {});
      for (uint64_t i = 0; i < n; ++i) {
        uint32_t len;
        std::memcpy(&len, p, 4);
        p += 4;
        rows[i].resize(len);
        std::memcpy(rows[i].data(), p, static_cast<size_t>(len) * 4);
        p += static_cast<size_t>(len) * 4;
      }
    }
  }

  const std::vector<vid_t>& row_of(const Fragment& frag,
                                   const KCliqueContext& ctx,
                                   vid_t gid) const {
    return ctx.adj[frag.parser().fid(gid)][frag.parser().lid(gid)];
  }

  void count(const Fragment& frag, KCliqueContext& ctx, MessageManager& mm) {
    const int k = ctx.k;
    std::atomic<uint64_t> total{0};
    parallel_for_tid(0, frag.ivnum(), [&](int tid, size_t vs) {
      (void)tid;
      const auto& row = ctx.adj[frag.fid()][vs];
      if (row.empty()) return;
      uint64_t local = 0;
      if (k == 2) {
        local = row.size();
      } else {
        // depth counts fixed vertices; v itself is the first
        recurse(frag, ctx, row, 1, k, &local);
      }
      total.fetch_add(local, std::memory_order_relaxed);
    }, 64);
    uint64_t mine = total.load();
    ctx.clique_num = mm.comm() ? mm.comm()->allreduce_sum(mine) : mine;
  }

  void recurse(const Fragment& frag, const KCliqueContext& ctx,
               const std::vector<vid_t>& cand, int depth, int k,
               uint64_t* out) const {
    if (depth == k - 1) {
      // depth vertices are fixed and mutually adjacent; any candidate
      // completes a k-clique (candidates all follow in gid order)
      *out += cand.size();
      return;
    }
    std::vector<vid_t> next;
    for (vid_t g : cand) {
      const auto& r = row_of(frag, ctx, g);
      next.clear();
      std::set_intersection(cand.begin(), cand.end(), r.begin(), r.end(),
                            std::back_inserter(next));
      if (!next.empty()) recurse(frag, ctx, next, depth + 1, k, out);
    }
  }
};

}  // namespace grapehip
