// grapehip — LCC (local clustering coefficient), CPU path.
// Reference parity: examples/analytical_apps/lcc/lcc.h (3-stage: degree/
// neighborhood exchange -> sorted-adjacency intersection count).
// LDBC semantics: N(v) = distinct in∪out neighbors (excl. self);
//   lcc(v) = |{(u,w) ∈ N(v)×N(v) : u→w ∈ E}| / (|N(v)|·(|N(v)|−1)),
// computed as Σ_{u∈N(v)} |N(v) ∩ Nout(u)| — for undirected symmetric
// storage this counts each triangle edge twice, matching the reference
// golden outputs (dataset/p2p-31-LCC). lcc(v) = 0 when |N(v)| < 2.
#pragma once

#include <algorithm>
#include <vector>

#include "../core/fragment.hpp"
#include "../core/message_manager.hpp"

namespace grapehip {

struct LCCContext {
  int stage = 0;
  // sorted gid adjacency for inner vertices
  std::vector<std::vector<vid_t>> nbr_union;  // N(v), inner only
  std::vector<std::vector<vid_t>> nbr_out;    // Nout(v), inner only (directed)
  // received Nout for outer vertices: flat CSR over outer lids
  std::vector<size_t> outer_off;
  std::vector<vid_t> outer_adj;
  std::vector<double> lcc;  // result over ivnum

  void init(const Fragment& frag) {
    stage = 0;
    nbr_union.assign(frag.ivnum(), {});
    nbr_out.clear();
    outer_off.assign(frag.ovnum() + 1, 0);
    outer_adj.clear();
    lcc.assign(frag.ivnum(), 0.0);
  }
};

class LCCApp {
 public:
  void PEval(const Fragment& frag, LCCContext& ctx, MessageManager& mm) {
    const bool directed = frag.directed();
    // Build sorted union-neighborhood (and out-neighborhood if directed).
    if (directed) ctx.nbr_out.assign(frag.ivnum(), {});
    parallel_for(0, frag.ivnum(), [&](size_t vs) {
      vid_t v = static_cast<vid_t>(vs);
      vid_t vgid = frag.lid2gid(v);
      auto& un = ctx.nbr_union[vs];
      auto push = [&](const Fragment::AdjRange& adj, std::vector<vid_t>* out) {
        for (size_t i = 0; i < adj.n; ++i) {
          vid_t g = frag.lid2gid(adj.dst[i]);
          if (g == vgid) continue;  // self loop
          un.push_back(g);
          if (out) out->push_back(g);
        }
      };
      push(frag.out_edges(v), directed ? &ctx.nbr_out[vs] : nullptr);
      if (directed && frag.has_in_csr()) push(frag.in_edges(v), nullptr);
      std::sort(un.begin(), un.end());
      un.erase(std::unique(un.begin(), un.end()), un.end());
      if (directed) {
        auto& no = ctx.nbr_out[vs];
        std::sort(no.begin(), no.end());
        no.erase(std::unique(no.begin(), no.end()), no.end());
      }
    }, 256);
    mm.force_continue();  // stage 1 runs next round (after mirror exchange)
  }

  void IncEval(const Fragment& frag, LCCContext& ctx, MessageManager& mm) {
    if (ctx.stage == 0) {
      ctx.stage = 1;
      exchange_neighborhoods(frag, ctx, mm);
      compute(frag, ctx);
      // done — no force_continue
    }
  }

 private:
  const std::vector<vid_t>& out_nbrs_of_inner(const Fragment& frag,
                                              const LCCContext& ctx,
                                              vid_t lid) const {
    return frag.directed() ? ctx.nbr_out[lid] : ctx.nbr_union[lid];
  }

  void exchange_neighborhoods(const Fragment& frag, LCCContext& ctx,
                              MessageManager& mm) {
    if (frag.fnum() == 1 || !mm.comm()) return;
    int fnum = frag.fnum();
    // send, per peer f: for each mirror lid (in order): [u32 len, gids...]
    std::vector<std::string> send(fnum);
    for (int f = 0; f < fnum; ++f) {
      InArchive ar;
      for (vid_t lid : frag.mirrors(f)) {
        const auto& no = out_nbrs_of_inner(frag, ctx, lid);
        ar.add(static_cast<uint32_t>(no.size()));
        ar.add_bytes(no.data(), no.size() * sizeof(vid_t));
      }
      send[f] = ar.release();
    }
    auto recv = mm.comm()->exchange_all(send);
    // fill outer CSR in outer-lid order (per-owner slices are contiguous):
    // pass 1 records lengths, pass 2 copies payloads after the prefix sum.
    const vid_t iv = frag.ivnum();
    size_t total = 0;
    for (int f = 0; f < fnum; ++f) {
      const char* p = recv[f].data();
      auto [b, e] = frag.outer_range(f);
      for (vid_t u = b; u < e; ++u) {
        uint32_t len;
        std::memcpy(&len, p, sizeof(len));
        p += sizeof(len) + static_cast<size_t>(len) * sizeof(vid_t);
        ctx.outer_off[u - iv + 1] = len;
        total += len;
      }
    }
    for (size_t i = 1; i < ctx.outer_off.size(); ++i)
      ctx.outer_off[i] += ctx.outer_off[i - 1];
    ctx.outer_adj.resize(total);
    for (int f = 0; f < fnum; ++f) {
      const char* p = recv[f].data();
      auto [b, e] = frag.outer_range(f);
      for (vid_t u = b; u < e; ++u) {
        uint32_t len;
        std::memcpy(&len, p, sizeof(len));
        p += sizeof(len);
        std::memcpy(ctx.outer_adj.data() + ctx.outer_off[u - iv], p,
                    static_cast<size_t>(len) * sizeof(vid_t));
        p += static_cast<size_t>(len) * sizeof(vid_t);
      }
    }
  }

  void compute(const Fragment& frag, LCCContext& ctx) {
    const vid_t iv = frag.ivnum();
    auto out_nbrs = [&](vid_t gid_lid /*lid*/) -> std::pair<const vid_t*, size_t> {
      if (frag.inner(gid_lid)) {
        const auto& no = out_nbrs_of_inner(frag, ctx, gid_lid);
        return {no.data(), no.size()};
      }
      size_t b = ctx.outer_off[gid_lid - iv], e = ctx.outer_off[gid_lid - iv + 1];
      return {ctx.outer_adj.data() + b, e - b};
    };
    parallel_for(0, iv, [&](size_t vs) {
      const auto& un = ctx.nbr_union[vs];
      size_t d = un.size();
      if (d < 2) {
        ctx.lcc[vs] = 0.0;
        return;
      }
      uint64_t cnt = 0;
      for (vid_t ugid : un) {
        vid_t ulid = frag.gid2lid(ugid);
        if (ulid == kInvalidVid) continue;
        auto [p, n] = out_nbrs(ulid);
        // sorted intersection |un ∩ p|
        size_t i = 0, j = 0;
        while (i < un.size() && j < n) {
          if (un[i] < p[j]) ++i;
          else if (un[i] > p[j]) ++j;
          else { ++cnt; ++i; ++j; }
        }
      }
      ctx.lcc[vs] = static_cast<double>(cnt) /
                    (static_cast<double>(d) * static_cast<double>(d - 1));
    }, 64);
  }
};

}  // namespace grapehip
