// grapehip — edge-cut CSR fragment + distributed builder.
//
// Reference parity: grape/fragment/immutable_edgecut_fragment.h (storage,
// inner/outer lid spaces, outer_vertices_of_frag ranges, mirror info),
// grape/fragment/csr_edgecut_fragment_base.h (CSR build), and the loader
// pipeline of grape/fragment/ev_fragment_loader.h + basic_fragment_loader.h
// (partition -> shuffle -> outer-vid collection -> CSR fill).
//
// MI355X-first design choices vs the reference:
//   * CSR offsets are 64-bit INDICES into flat dst/weight arrays, not
//     nbr_t* pointers — device-relocatable, uploaded once into HBM3E.
//   * lids are 32-bit; inner lids [0, ivnum), outer lids [ivnum, tvnum).
//   * outer gids are sorted, so per-owner outer lid ranges are contiguous
//     slices — RCCL halo sends are contiguous buffers, no gather kernel for
//     the dense (batch-shuffle) path.
#pragma once

#include <algorithm>
#include <cassert>
#include <cstring>
#include <memory>
#include <type_traits>
#include <unordered_map>
#include <utility>
#include <vector>

#include "archive.hpp"
#include "net.hpp"
#include "parallel.hpp"
#include "types.hpp"
#include "vertex_map.hpp"

namespace grapehip {

struct EdgeTriple {
  oid_t src;
  oid_t dst;
  weight_t w;
};

class Fragment {
 public:
  fid_t fid() const { return fid_; }
  int fnum() const { return fnum_; }
  bool directed() const { return directed_; }
  bool has_in_csr() const { return !ie_off_.empty(); }
  bool has_weights() const { return !oe_w_.empty(); }
  vid_t ivnum() const { return ivnum_; }
  vid_t ovnum() const { return ovnum_; }
  vid_t tvnum() const { return ivnum_ + ovnum_; }
  uint64_t total_vertices() const { return total_vertices_; }
  uint64_t total_edges() const { return total_edges_; }   // global, stored dir
  uint64_t input_edges() const { return input_edges_; }   // global, as input
  uint64_t local_edges() const { return oe_dst_.size(); }
  const VertexMap& vm() const { return *vm_; }
  const IdParser& parser() const { return parser_; }

  // --- id conversions ---------------------------------------------------
  vid_t lid2gid(vid_t lid) const {
    return lid < ivnum_ ? parser_.gid(fid_, lid) : ovgid_[lid - ivnum_];
  }
  oid_t lid2oid(vid_t lid) const { return vm_->get_oid(lid2gid(lid)); }
  fid_t lid2fid(vid_t lid) const {
    return lid < ivnum_ ? fid_ : parser_.fid(ovgid_[lid - ivnum_]);
  }
  bool inner(vid_t lid) const { return lid < ivnum_; }
  // gid -> lid; returns kInvalidVid if not present in this fragment.
  vid_t gid2lid(vid_t gid) const {
    if (parser_.fid(gid) == fid_) return parser_.lid(gid);
    auto it = ovg2l_.find(gid);
    return it == ovg2l_.end() ? kInvalidVid : it->second;
  }
  vid_t inner_gid2lid(vid_t gid) const { return parser_.lid(gid); }
  bool oid2lid(oid_t oid, vid_t* lid) const {
    vid_t gid;
    if (!vm_->get_gid(oid, &gid)) return false;
    vid_t l = gid2lid(gid);
    if (l == kInvalidVid) return false;
    *lid = l;
    return true;
  }

  // --- adjacency --------------------------------------------------------
  // Mutable mode (after an in-place Mutate): rows live at explicit
  // (start, len) with capacity slack, relocated to an end arena when they
  // outgrow it (reference DeMutableCSR semantics, de_mutable_csr.h).
  // compact() squeezes back to canonical monotonic-offset form.
  struct AdjRange {
    const vid_t* dst;
    const weight_t* w;  // nullptr if unweighted
    size_t n;
  };
  AdjRange out_edges(vid_t u) const {
    eid_t b = mutable_mode_ ? oe_start_[u] : oe_off_[u];
    size_t n = mutable_mode_ ? oe_len_[u]
                             : static_cast<size_t>(oe_off_[u + 1] - b);
    return {oe_dst_.data() + b, oe_w_.empty() ? nullptr : oe_w_.data() + b,
            n};
  }
  AdjRange in_edges(vid_t u) const {
    eid_t b = mutable_mode_ ? ie_start_[u] : ie_off_[u];
    size_t n = mutable_mode_ ? ie_len_[u]
                             : static_cast<size_t>(ie_off_[u + 1] - b);
    return {ie_dst_.data() + b, ie_w_.empty() ? nullptr : ie_w_.data() + b,
            n};
  }
  size_t out_degree(vid_t u) const {
    return mutable_mode_ ? oe_len_[u] : oe_off_[u + 1] - oe_off_[u];
  }
  size_t in_degree(vid_t u) const {
    return mutable_mode_ ? ie_len_[u] : ie_off_[u + 1] - ie_off_[u];
  }
  bool mutable_mode() const { return mutable_mode_; }

  // Raw arrays (used by the GPU uploader and serializer).
  const std::vector<eid_t>& oe_offsets() const { return oe_off_; }
  const std::vector<vid_t>& oe_dsts() const { return oe_dst_; }
  const std::vector<weight_t>& oe_weights() const { return oe_w_; }
  const std::vector<eid_t>& ie_offsets() const { return ie_off_; }
  const std::vector<vid_t>& ie_dsts() const { return ie_dst_; }
  const std::vector<weight_t>& ie_weights() const { return ie_w_; }
  const std::vector<vid_t>& outer_gids() const { return ovgid_; }

  // Outer lid range [first, second) owned by fragment f.
  std::pair<vid_t, vid_t> outer_range(fid_t f) const { return outer_range_[f]; }
  // Inner lids of this fragment that fragment f holds as outer vertices.
  const std::vector<vid_t>& mirrors(fid_t f) const { return mirrors_[f]; }

  // --- builder ----------------------------------------------------------
  // Collective. `edges` is this rank's arbitrary slice of the input edge
  // list (oids). For undirected inputs both directions are materialized in
  // the out-CSR. For directed inputs, `build_in_csr` additionally builds the
  // incoming CSR (needed by pull PageRank / direction-optimizing BFS).
  static std::unique_ptr<Fragment> Build(std::shared_ptr<VertexMap> vm,
                                         TcpComm* comm, fid_t fid, int fnum,
                                         std::vector<EdgeTriple> edges,
                                         bool directed, bool weighted,
                                         bool build_in_csr,
                                         uint64_t input_edge_count_local) {
    auto frag = std::make_unique<Fragment>();
    Fragment& F = *frag;
    F.vm_ = std::move(vm);
    F.fid_ = fid;
    F.fnum_ = fnum;
    F.parser_ = F.vm_->parser();
    F.directed_ = directed;
    F.total_vertices_ = F.vm_->total_vertices();
    F.ivnum_ = F.vm_->frag_vnum(fid);

    const VertexMap& V = *F.vm_;
    const IdParser& P = F.parser_;
    bool want_in = directed && build_in_csr;

    // 1) route edges: out stream -> owner(src); in stream -> owner(dst).
    //    Undirected: emit both orientations into the out stream.
    struct GidEdge {
      vid_t src, dst;
      weight_t w;
    };
    std::vector<std::string> out_send(fnum), in_send(fnum);
    {
      std::vector<std::vector<GidEdge>> out_bins(fnum), in_bins(fnum);
      for (const EdgeTriple& e : edges) {
        vid_t sg, dg;
        if (!V.get_gid(e.src, &sg) || !V.get_gid(e.dst, &dg)) continue;
        out_bins[P.fid(sg)].push_back({sg, dg, e.w});
        if (!directed) {
          // reference parity: undirected storage holds BOTH orientations,
          // including self loops twice (csr_edgecut_fragment_base.h
          // parse_iter_out_undirected increments both endpoints)
          out_bins[P.fid(dg)].push_back({dg, sg, e.w});
        } else if (want_in) {
          in_bins[P.fid(dg)].push_back({dg, sg, e.w});
        }
      }
      edges.clear();
      edges.shrink_to_fit();
      for (int f = 0; f < fnum; ++f) {
        out_send[f].assign(
            reinterpret_cast<const char*>(out_bins[f].data()),
            out_bins[f].size() * sizeof(GidEdge));
        in_send[f].assign(reinterpret_cast<const char*>(in_bins[f].data()),
                          in_bins[f].size() * sizeof(GidEdge));
      }
    }
    std::vector<std::string> out_recv, in_recv;
    if (comm && fnum > 1) {
      out_recv = comm->exchange_all(out_send);
      in_recv = want_in ? comm->exchange_all(in_send)
                        : std::vector<std::string>(fnum);
    } else {
      out_recv = std::move(out_send);
      in_recv = std::move(in_send);
    }

    auto as_edges = [](const std::string& blob) {
      return std::make_pair(
          reinterpret_cast<const GidEdge*>(blob.data()),
          blob.size() / sizeof(GidEdge));
    };

    // 2) collect outer gids (dst of out edges + "dst" of in edges that are
    //    remote sources).
    std::vector<vid_t> outer;
    for (int f = 0; f < fnum; ++f) {
      for (auto* blob : {&out_recv[f], &in_recv[f]}) {
        auto [p, n] = as_edges(*blob);
        for (size_t i = 0; i < n; ++i)
          if (P.fid(p[i].dst) != fid) outer.push_back(p[i].dst);
      }
    }
    std::sort(outer.begin(), outer.end());
    outer.erase(std::unique(outer.begin(), outer.end()), outer.end());
    F.ovgid_ = std::move(outer);
    F.ovnum_ = static_cast<vid_t>(F.ovgid_.size());
    F.ovg2l_.reserve(F.ovnum_ * 2);
    for (vid_t i = 0; i < F.ovnum_; ++i)
      F.ovg2l_.emplace(F.ovgid_[i], F.ivnum_ + i);

    // contiguous per-owner outer ranges (gids sorted => grouped by fid
    // because fid occupies the high bits of the gid).
    F.outer_range_.assign(fnum, {0, 0});
    {
      vid_t i = 0;
      for (int f = 0; f < fnum; ++f) {
        vid_t b = i;
        while (i < F.ovnum_ &&
               P.fid(F.ovgid_[i]) == static_cast<fid_t>(f))
          ++i;
        F.outer_range_[f] = {F.ivnum_ + b, F.ivnum_ + i};
      }
    }

    // 3) CSR fill (count -> exclusive scan -> scatter), parallel.
    auto build_csr = [&](const std::vector<std::string>& blobs,
                         std::vector<eid_t>& off, std::vector<vid_t>& dst,
                         std::vector<weight_t>& wts) {
      std::vector<std::atomic<eid_t>> deg(F.ivnum_ + 1);
      for (auto& d : deg) d.store(0, std::memory_order_relaxed);
      for (auto& blob : blobs) {
        auto pe = as_edges(blob);
        const GidEdge* p = pe.first;
        size_t n = pe.second;
        parallel_for(0, n, [&](size_t i) {
          deg[P.lid(p[i].src)].fetch_add(1, std::memory_order_relaxed);
        }, 8192);
      }
      off.resize(F.ivnum_ + 1);
      eid_t run = 0;
      for (vid_t v = 0; v < F.ivnum_; ++v) {
        off[v] = run;
        run += deg[v].load(std::memory_order_relaxed);
        deg[v].store(off[v], std::memory_order_relaxed);  // reuse as cursor
      }
      off[F.ivnum_] = run;
      dst.resize(run);
      if (weighted) wts.resize(run);
      for (auto& blob : blobs) {
        auto pe = as_edges(blob);
        const GidEdge* p = pe.first;
        size_t n = pe.second;
        parallel_for(0, n, [&](size_t i) {
          vid_t u = P.lid(p[i].src);
          eid_t slot = deg[u].fetch_add(1, std::memory_order_relaxed);
          vid_t d = p[i].dst;
          vid_t dl = P.fid(d) == fid ? P.lid(d) : F.ovg2l_.at(d);
          dst[slot] = dl;
          if (weighted) wts[slot] = p[i].w;
        }, 8192);
      }
    };
    build_csr(out_recv, F.oe_off_, F.oe_dst_, F.oe_w_);
    if (want_in) build_csr(in_recv, F.ie_off_, F.ie_dst_, F.ie_w_);
    else if (!directed) {
      // undirected symmetric storage: in == out
    }

    // 4) mirror info: peer f's slice of MY vertices = the outer gids peer f
    //    collected that I own. One exchange of gid lists.
    F.mirrors_.assign(fnum, {});
    if (comm && fnum > 1) {
      std::vector<std::string> send(fnum);
      for (int f = 0; f < fnum; ++f) {
        auto [b, e] = F.outer_range_[f];
        send[f].assign(
            reinterpret_cast<const char*>(F.ovgid_.data() + (b - F.ivnum_)),
            (e - b) * sizeof(vid_t));
      }
      auto recv = comm->exchange_all(send);
      for (int f = 0; f < fnum; ++f) {
        size_t n = recv[f].size() / sizeof(vid_t);
        F.mirrors_[f].resize(n);
        const vid_t* g = reinterpret_cast<const vid_t*>(recv[f].data());
        for (size_t i = 0; i < n; ++i) F.mirrors_[f][i] = P.lid(g[i]);
      }
    }

    uint64_t local_out = F.oe_dst_.size();
    F.total_edges_ = comm ? comm->allreduce_sum(local_out) : local_out;
    F.input_edges_ = comm ? comm->allreduce_sum(input_edge_count_local)
                          : input_edge_count_local;
    return frag;
  }

  // Decode this fragment's stored edges back to oid-space triples, one per
  // ORIGINAL input edge (undirected storage holds both orientations; only
  // the src<=dst copy is emitted so a rebuild reproduces the multiset).
  // Used by the mutation path (reference mutable_edgecut_fragment.h:289-399
  // applies deltas in place; grapehip mutates by functional rebuild — the
  // parallel builder is fast enough that slack-tracking CSRs don't pay).
  std::vector<EdgeTriple> to_triples() const {
    std::vector<EdgeTriple> out;
    out.reserve(oe_dst_.size());
    for (vid_t v = 0; v < ivnum_; ++v) {
      oid_t so = lid2oid(v);
      auto adj = out_edges(v);
      bool drop_self = false;  // self loops stored twice: emit every other
      for (size_t i = 0; i < adj.n; ++i) {
        oid_t dg = lid2oid(adj.dst[i]);
        if (!directed_) {
          vid_t sg = lid2gid(v), dgid = lid2gid(adj.dst[i]);
          if (sg > dgid) continue;  // one orientation per input edge
          if (sg == dgid) {
            drop_self = !drop_self;
            if (drop_self) continue;
          }
        }
        out.push_back({so, dg, adj.w ? adj.w[i] : 1.0f});
      }
    }
    return out;
  }

  // Rebuild from a serialized checkpoint (see core/serialize.hpp). R is any
  // reader with pod<T>() / vec(std::vector<T>&).
  template <typename R>
  static std::unique_ptr<Fragment> FromParts(std::shared_ptr<VertexMap> vm,
                                             R& r, uint32_t fnum) {
    auto frag = std::make_unique<Fragment>();
    Fragment& F = *frag;
    F.vm_ = std::move(vm);
    F.fnum_ = static_cast<int>(fnum);
    F.parser_ = F.vm_->parser();
    F.fid_ = static_cast<fid_t>(r.template pod<uint32_t>());
    F.directed_ = r.template pod<uint8_t>() != 0;
    F.total_edges_ = r.template pod<uint64_t>();
    F.input_edges_ = r.template pod<uint64_t>();
    F.total_vertices_ = F.vm_->total_vertices();
    F.ivnum_ = F.vm_->frag_vnum(F.fid_);
    r.vec(F.oe_off_);
    r.vec(F.oe_dst_);
    r.vec(F.oe_w_);
    r.vec(F.ie_off_);
    r.vec(F.ie_dst_);
    r.vec(F.ie_w_);
    r.vec(F.ovgid_);
    F.ovnum_ = static_cast<vid_t>(F.ovgid_.size());
    F.ovg2l_.clear();
    F.ovg2l_.reserve(F.ovnum_ * 2);
    for (vid_t i = 0; i < F.ovnum_; ++i)
      F.ovg2l_.emplace(F.ovgid_[i], F.ivnum_ + i);
    F.outer_range_.resize(fnum);
    F.mirrors_.assign(fnum, {});
    for (uint32_t f = 0; f < fnum; ++f) {
      vid_t b = r.template pod<vid_t>();
      vid_t e = r.template pod<vid_t>();
      F.outer_range_[f] = {b, e};
      r.vec(F.mirrors_[f]);
    }
    return frag;
  }

  // --- in-place delta mutation -------------------------------------------
  // Reference parity: MutableEdgecutFragment::Mutate
  // (mutable_edgecut_fragment.h:289-399) — cost proportional to the delta
  // (plus, only when NEW outer vertices appear, one local remap sweep to
  // keep the sorted-outer invariant the dense paths rely on). Collective.
  // adds/removes are this rank's arbitrary slice of the delta, in oids.
  // Unknown oids are skipped like the builder does.
  void MutateDelta(TcpComm* comm,
                   const std::vector<EdgeTriple>& adds,
                   const std::vector<std::pair<oid_t, oid_t>>& removes) {
    const VertexMap& V = *vm_;
    const IdParser& P = parser_;
    const bool weighted = has_weights();
    const bool want_in = has_in_csr();
    struct GidEdge {
      vid_t src, dst;
      weight_t w;
    };
    // 1) route deltas to owners (same shape as Build): tag removals with
    //    a NaN-free sentinel channel — two separate streams.
    std::vector<std::vector<GidEdge>> add_out(fnum_), add_in(fnum_);
    std::vector<std::vector<std::pair<vid_t, vid_t>>> rm_out(fnum_),
        rm_in(fnum_);
    for (const EdgeTriple& e : adds) {
      vid_t sg, dg;
      if (!V.get_gid(e.src, &sg) || !V.get_gid(e.dst, &dg)) continue;
      add_out[P.fid(sg)].push_back({sg, dg, e.w});
      if (!directed_) add_out[P.fid(dg)].push_back({dg, sg, e.w});
      else if (want_in) add_in[P.fid(dg)].push_back({dg, sg, e.w});
    }
    for (const auto& r : removes) {
      vid_t sg, dg;
      if (!V.get_gid(r.first, &sg) || !V.get_gid(r.second, &dg)) continue;
      rm_out[P.fid(sg)].push_back({sg, dg});
      if (!directed_) rm_out[P.fid(dg)].push_back({dg, sg});
      else if (want_in) rm_in[P.fid(dg)].push_back({dg, sg});
    }
    auto exchange_vec = [&](auto& bins) {
      using ElemT =
          typename std::decay_t<decltype(bins[0])>::value_type;
      std::vector<std::string> send(fnum_);
      for (int f = 0; f < fnum_; ++f)
        send[f].assign(reinterpret_cast<const char*>(bins[f].data()),
                       bins[f].size() * sizeof(ElemT));
      std::vector<std::string> recv =
          (comm && fnum_ > 1) ? comm->exchange_all(send) : std::move(send);
      std::vector<ElemT> all;
      for (auto& blob : recv) {
        size_t n = blob.size() / sizeof(ElemT);
        const ElemT* p = reinterpret_cast<const ElemT*>(blob.data());
        all.insert(all.end(), p, p + n);
      }
      return all;
    };
    std::vector<GidEdge> my_add_out = exchange_vec(add_out);
    std::vector<GidEdge> my_add_in = exchange_vec(add_in);
    std::vector<std::pair<vid_t, vid_t>> my_rm_out = exchange_vec(rm_out);
    std::vector<std::pair<vid_t, vid_t>> my_rm_in = exchange_vec(rm_in);

    // 2) new outer gids referenced by the adds
    std::vector<vid_t> fresh;
    for (auto* lst : {&my_add_out, &my_add_in})
      for (const GidEdge& e : *lst)
        if (P.fid(e.dst) != fid_ && !ovg2l_.count(e.dst))
          fresh.push_back(e.dst);
    std::sort(fresh.begin(), fresh.end());
    fresh.erase(std::unique(fresh.begin(), fresh.end()), fresh.end());
    uint64_t fresh_any = fresh.size();
    if (comm && fnum_ > 1) fresh_any = comm->allreduce_sum(fresh_any);
    if (fresh_any) grow_outer(comm, fresh);

    // 3) enter mutable mode and apply
    enter_mutable_mode();
    auto apply = [&](std::vector<eid_t>& start, std::vector<vid_t>& len,
                     std::vector<vid_t>& cap, std::vector<vid_t>& dst,
                     std::vector<weight_t>& wts,
                     const std::vector<std::pair<vid_t, vid_t>>& rms,
                     const std::vector<GidEdge>& ads) {
      for (const auto& r : rms) {
        vid_t u = P.lid(r.first);
        vid_t dl = P.fid(r.second) == fid_ ? P.lid(r.second)
                                           : lookup_outer(r.second);
        if (dl == kInvalidVid) continue;
        eid_t b = start[u];
        for (vid_t k = 0; k < len[u];) {
          if (dst[b + k] == dl) {
            dst[b + k] = dst[b + len[u] - 1];
            if (weighted) wts[b + k] = wts[b + len[u] - 1];
            --len[u];
          } else {
            ++k;
          }
        }
      }
      for (const GidEdge& e : ads) {
        vid_t u = P.lid(e.src);
        vid_t dl = P.fid(e.dst) == fid_ ? P.lid(e.dst)
                                        : lookup_outer(e.dst);
        if (dl == kInvalidVid) continue;
        if (len[u] == cap[u]) {
          // relocate to the end arena with doubled capacity
          vid_t ncap = cap[u] ? cap[u] * 2 : 4;
          eid_t nb = static_cast<eid_t>(dst.size());
          dst.resize(nb + ncap);
          if (weighted) wts.resize(nb + ncap);
          std::memcpy(dst.data() + nb, dst.data() + start[u],
                      len[u] * sizeof(vid_t));
          if (weighted)
            std::memcpy(wts.data() + nb, wts.data() + start[u],
                        len[u] * sizeof(weight_t));
          start[u] = nb;
          cap[u] = ncap;
        }
        dst[start[u] + len[u]] = dl;
        if (weighted) wts[start[u] + len[u]] = e.w;
        ++len[u];
      }
    };
    apply(oe_start_, oe_len_, oe_cap_, oe_dst_, oe_w_, my_rm_out,
          my_add_out);
    if (want_in)
      apply(ie_start_, ie_len_, ie_cap_, ie_dst_, ie_w_, my_rm_in,
            my_add_in);

    // 4) refresh global counts
    uint64_t local = 0;
    for (vid_t v = 0; v < ivnum_; ++v) local += oe_len_[v];
    total_edges_ = (comm && fnum_ > 1) ? comm->allreduce_sum(local) : local;
    input_edges_ = directed_ ? total_edges_ : total_edges_ / 2;
  }

  // Squeeze mutable-mode slack back into canonical monotonic offsets
  // (called before checkpointing or a GPU upload).
  void compact() {
    if (!mutable_mode_) return;
    auto squeeze = [&](std::vector<eid_t>& off, std::vector<vid_t>& dst,
                       std::vector<weight_t>& wts, std::vector<eid_t>& start,
                       std::vector<vid_t>& len) {
      std::vector<eid_t> noff(ivnum_ + 1);
      eid_t run = 0;
      for (vid_t v = 0; v < ivnum_; ++v) {
        noff[v] = run;
        run += len[v];
      }
      noff[ivnum_] = run;
      std::vector<vid_t> ndst(run);
      std::vector<weight_t> nw(wts.empty() ? 0 : run);
      parallel_for(0, static_cast<size_t>(ivnum_), [&](size_t v) {
        std::memcpy(ndst.data() + noff[v], dst.data() + start[v],
                    len[v] * sizeof(vid_t));
        if (!wts.empty())
          std::memcpy(nw.data() + noff[v], wts.data() + start[v],
                      len[v] * sizeof(weight_t));
      }, 256);
      off = std::move(noff);
      dst = std::move(ndst);
      wts = std::move(nw);
    };
    squeeze(oe_off_, oe_dst_, oe_w_, oe_start_, oe_len_);
    if (has_in_csr()) squeeze(ie_off_, ie_dst_, ie_w_, ie_start_, ie_len_);
    oe_start_.clear();
    oe_len_.clear();
    oe_cap_.clear();
    ie_start_.clear();
    ie_len_.clear();
    ie_cap_.clear();
    mutable_mode_ = false;
  }

 private:
  vid_t lookup_outer(vid_t gid) const {
    auto it = ovg2l_.find(gid);
    return it == ovg2l_.end() ? kInvalidVid : it->second;
  }

  void enter_mutable_mode() {
    if (mutable_mode_) return;
    auto init = [&](const std::vector<eid_t>& off, std::vector<eid_t>& start,
                    std::vector<vid_t>& len, std::vector<vid_t>& cap) {
      start.resize(ivnum_);
      len.resize(ivnum_);
      cap.resize(ivnum_);
      for (vid_t v = 0; v < ivnum_; ++v) {
        start[v] = off[v];
        len[v] = static_cast<vid_t>(off[v + 1] - off[v]);
        cap[v] = len[v];
      }
    };
    init(oe_off_, oe_start_, oe_len_, oe_cap_);
    if (has_in_csr()) init(ie_off_, ie_start_, ie_len_, ie_cap_);
    mutable_mode_ = true;
  }

  // Insert new outer gids while keeping ovgid_ sorted (the contiguity
  // invariant every dense path relies on): one local remap sweep of the
  // stored dst lids plus a mirror-info refresh — no global rebuild.
  void grow_outer(TcpComm* comm, const std::vector<vid_t>& fresh) {
    std::vector<vid_t> merged(ovgid_.size() + fresh.size());
    std::merge(ovgid_.begin(), ovgid_.end(), fresh.begin(), fresh.end(),
               merged.begin());
    // old outer lid -> new outer lid
    std::vector<vid_t> remap(ovnum_);
    {
      size_t j = 0;
      for (size_t i = 0; i < ovgid_.size(); ++i) {
        while (merged[j] != ovgid_[i]) ++j;
        remap[i] = ivnum_ + static_cast<vid_t>(j);
      }
    }
    auto sweep = [&](std::vector<vid_t>& dst, const std::vector<eid_t>& start,
                     const std::vector<vid_t>& len) {
      if (mutable_mode_) {
        parallel_for(0, static_cast<size_t>(ivnum_), [&](size_t v) {
          eid_t b = start[v];
          for (vid_t k = 0; k < len[v]; ++k)
            if (dst[b + k] >= ivnum_) dst[b + k] = remap[dst[b + k] - ivnum_];
        }, 256);
      } else {
        parallel_for(0, dst.size(), [&](size_t i) {
          if (dst[i] >= ivnum_) dst[i] = remap[dst[i] - ivnum_];
        }, 8192);
      }
    };
    sweep(oe_dst_, oe_start_, oe_len_);
    if (has_in_csr()) sweep(ie_dst_, ie_start_, ie_len_);
    ovgid_ = std::move(merged);
    ovnum_ = static_cast<vid_t>(ovgid_.size());
    ovg2l_.clear();
    ovg2l_.reserve(ovnum_ * 2);
    for (vid_t i = 0; i < ovnum_; ++i)
      ovg2l_.emplace(ovgid_[i], ivnum_ + i);
    outer_range_.assign(fnum_, {0, 0});
    {
      vid_t i = 0;
      for (int f = 0; f < fnum_; ++f) {
        vid_t b = i;
        while (i < ovnum_ &&
               parser_.fid(ovgid_[i]) == static_cast<fid_t>(f))
          ++i;
        outer_range_[f] = {ivnum_ + b, ivnum_ + i};
      }
    }
    // mirror info depends on every peer's outer order — refresh it
    mirrors_.assign(fnum_, {});
    if (comm && fnum_ > 1) {
      std::vector<std::string> send(fnum_);
      for (int f = 0; f < fnum_; ++f) {
        auto [b, e] = outer_range_[f];
        send[f].assign(
            reinterpret_cast<const char*>(ovgid_.data() + (b - ivnum_)),
            (e - b) * sizeof(vid_t));
      }
      auto recv = comm->exchange_all(send);
      for (int f = 0; f < fnum_; ++f) {
        size_t n = recv[f].size() / sizeof(vid_t);
        mirrors_[f].resize(n);
        const vid_t* g = reinterpret_cast<const vid_t*>(recv[f].data());
        for (size_t i = 0; i < n; ++i) mirrors_[f][i] = parser_.lid(g[i]);
      }
    }
  }

  fid_t fid_ = 0;
  int fnum_ = 1;
  IdParser parser_;
  std::shared_ptr<const VertexMap> vm_;
  bool directed_ = false;
  vid_t ivnum_ = 0, ovnum_ = 0;
  uint64_t total_vertices_ = 0, total_edges_ = 0, input_edges_ = 0;
  std::vector<eid_t> oe_off_, ie_off_;
  std::vector<vid_t> oe_dst_, ie_dst_;
  std::vector<weight_t> oe_w_, ie_w_;
  std::vector<vid_t> ovgid_;
  std::unordered_map<vid_t, vid_t> ovg2l_;
  std::vector<std::pair<vid_t, vid_t>> outer_range_;
  std::vector<std::vector<vid_t>> mirrors_;
  // mutable-mode row bookkeeping (empty until the first MutateDelta)
  bool mutable_mode_ = false;
  std::vector<eid_t> oe_start_, ie_start_;
  std::vector<vid_t> oe_len_, oe_cap_, ie_len_, ie_cap_;
};

}  // namespace grapehip
