// grapehip — global oid <-> gid service + partitioners + id parser.
//
// Reference parity: grape/fragment/id_parser.h (gid = fid high bits | lid),
// grape/vertex_map/{vertex_map,partitioner}.h and the idxer strategies
// (idxers/). Two idxer modes are kept:
//   * identity — oids are dense 0..V-1 and the partitioner is segmented or
//     modulo; oid<->lid is arithmetic, zero memory. This is the fast path for
//     synthetic LDBC datagen graphs and the GPU bench.
//   * hashmap — arbitrary int64 oids; per-fragment open-addressing index,
//     replicated on every rank via allgather (the reference's default
//     HashMapIdxer, vertex_map.h:312).
#pragma once

#include <algorithm>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <unordered_map>
#include <vector>

#include "net.hpp"
#include "parallel.hpp"
#include "mph.hpp"
#include "types.hpp"

namespace grapehip {

class IdParser {
 public:
  void init(int fnum) {
    fid_bits_ = 0;
    while ((1u << fid_bits_) < static_cast<uint32_t>(fnum)) ++fid_bits_;
    offset_ = 32 - fid_bits_;
    lid_mask_ = offset_ == 32 ? ~0u : ((1u << offset_) - 1);
  }
  fid_t fid(vid_t gid) const {
    return fid_bits_ == 0 ? 0 : (gid >> offset_);
  }
  vid_t lid(vid_t gid) const { return gid & lid_mask_; }
  vid_t gid(fid_t f, vid_t l) const {
    return fid_bits_ == 0 ? l : ((static_cast<vid_t>(f) << offset_) | l);
  }
  vid_t max_lid() const { return lid_mask_; }
  int fid_bits() const { return fid_bits_; }
  int lid_offset() const { return offset_; }

 private:
  uint32_t fid_bits_ = 0;
  uint32_t offset_ = 32;
  vid_t lid_mask_ = ~0u;
};

enum class IdxerKind : uint8_t { kIdentity = 0, kHashmap = 1, kMph = 2 };

class VertexMap {
 public:
  // ---- construction ----------------------------------------------------
  // Identity mode: dense oids [0, nv), segmented ownership.
  void init_identity(int fnum, uint64_t nv) {
    fnum_ = fnum;
    parser_.init(fnum);
    idxer_ = IdxerKind::kIdentity;
    pkind_ = PartitionerKind::kSegmented;
    total_vertices_ = nv;
    // uniform ceil-slices: owner(v) == min(v / slice, fnum-1) — O(1) on the
    // GPU and reduce-scatter/allgather slices are equal-sized (last padded).
    seg_.resize(fnum + 1);
    uint64_t slice = (nv + fnum - 1) / fnum;
    for (int f = 0; f <= fnum; ++f)
      seg_[f] = std::min<uint64_t>(static_cast<uint64_t>(f) * slice, nv);
  }

  // Hashmap mode: every rank supplies the oids it OWNS (dedup'd); lids are
  // assigned in the given order; oid lists replicated via allgather.
  void init_hashmap(int fnum, PartitionerKind pk, TcpComm* comm,
                    std::vector<oid_t> owned_oids) {
    fnum_ = fnum;
    parser_.init(fnum);
    idxer_ = IdxerKind::kHashmap;
    pkind_ = pk;
    l2o_.assign(fnum, {});
    o2g_.clear();
    // Replicate: exchange per-fragment oid lists.
    std::vector<std::string> send(fnum);
    std::string mine(reinterpret_cast<const char*>(owned_oids.data()),
                     owned_oids.size() * sizeof(oid_t));
    for (int f = 0; f < fnum; ++f) send[f] = mine;
    std::vector<std::string> recv =
        comm ? comm->exchange_all(send) : std::vector<std::string>{mine};
    total_vertices_ = 0;
    for (int f = 0; f < fnum; ++f) {
      const std::string& blob = recv[f];
      size_t n = blob.size() / sizeof(oid_t);
      l2o_[f].resize(n);
      std::memcpy(l2o_[f].data(), blob.data(), blob.size());
      total_vertices_ += n;
    }
    o2g_.reserve(total_vertices_);
    for (int f = 0; f < fnum; ++f) {
      for (size_t l = 0; l < l2o_[f].size(); ++l) {
        o2g_.emplace(l2o_[f][l],
                     parser_.gid(f, static_cast<vid_t>(l)));
      }
    }
  }

  // ---- queries ---------------------------------------------------------
  fid_t owner(oid_t oid) const {
    if (pkind_ == PartitionerKind::kSegmented) {
      uint64_t o = static_cast<uint64_t>(oid);
      // branchless-ish upper_bound over <=8 entries
      fid_t f = 0;
      while (f + 1 < static_cast<fid_t>(fnum_) && o >= seg_[f + 1]) ++f;
      return f;
    }
    if (pkind_ == PartitionerKind::kMap)
      throw std::runtime_error(
          "owner() undefined for map partitioning: ownership is whatever "
          "rank supplied the oid — route through get_gid instead");
    return static_cast<fid_t>(hash_oid(oid) % fnum_);
  }

  bool get_gid(oid_t oid, vid_t* gid) const {
    if (idxer_ == IdxerKind::kIdentity) {
      if (static_cast<uint64_t>(oid) >= total_vertices_ || oid < 0)
        return false;
      fid_t f = owner(oid);
      *gid = parser_.gid(f, static_cast<vid_t>(oid - seg_[f]));
      return true;
    }
    if (idxer_ == IdxerKind::kMph) {
      fid_t f = static_cast<fid_t>(hash_oid(oid) % fnum_);
      uint64_t lid = mph_[f].lookup(oid);
      if (lid >= l2o_[f].size() || l2o_[f][lid] != oid) return false;
      *gid = parser_.gid(f, static_cast<vid_t>(lid));
      return true;
    }
    auto it = o2g_.find(oid);
    if (it == o2g_.end()) return false;
    *gid = it->second;
    return true;
  }

  oid_t get_oid(vid_t gid) const {
    fid_t f = parser_.fid(gid);
    vid_t l = parser_.lid(gid);
    if (idxer_ == IdxerKind::kIdentity)
      return static_cast<oid_t>(seg_[f] + l);
    return l2o_[f][l];  // hashmap and mph both keep lid->oid dense
  }

  vid_t frag_vnum(fid_t f) const {
    if (idxer_ == IdxerKind::kIdentity)
      return static_cast<vid_t>(seg_[f + 1] - seg_[f]);
    return static_cast<vid_t>(l2o_[f].size());
  }

  // MPH mode (reference PTHashIdxer parity): hash-partitioned ownership +
  // per-fragment minimal perfect hash for oid->lid, ~5 bits/key of index
  // structure instead of an open-addressing map. Deterministic build, so
  // every rank derives identical lids from the replicated oid lists.
  void init_mph(int fnum, TcpComm* comm, std::vector<oid_t> owned_oids) {
    fnum_ = fnum;
    parser_.init(fnum);
    idxer_ = IdxerKind::kMph;
    pkind_ = PartitionerKind::kHash;
    l2o_.assign(fnum, {});
    // replicate per-fragment oid lists (same exchange as hashmap mode)
    std::vector<std::string> send(fnum);
    std::string mine(reinterpret_cast<const char*>(owned_oids.data()),
                     owned_oids.size() * sizeof(oid_t));
    for (int f = 0; f < fnum; ++f) send[f] = mine;
    std::vector<std::string> recv =
        comm ? comm->exchange_all(send) : std::vector<std::string>{mine};
    total_vertices_ = 0;
    mph_.assign(fnum, {});
    for (int f = 0; f < fnum; ++f) {
      const std::string& blob = recv[f];
      size_t n = blob.size() / sizeof(oid_t);
      std::vector<oid_t> keys(n);
      std::memcpy(keys.data(), blob.data(), blob.size());
      mph_[f].build(keys);
      // place keys at their MPH index: lid = mph(oid)
      l2o_[f].assign(n, 0);
      for (oid_t k : keys) l2o_[f][mph_[f].lookup(k)] = k;
      total_vertices_ += n;
    }
  }
  void init_mph_local(int fnum, std::vector<std::vector<oid_t>> l2o) {
    // deserialization path: rebuild the (deterministic) MPH per fragment
    fnum_ = fnum;
    parser_.init(fnum);
    idxer_ = IdxerKind::kMph;
    pkind_ = PartitionerKind::kHash;
    l2o_.assign(fnum, {});
    mph_.assign(fnum, {});
    total_vertices_ = 0;
    for (int f = 0; f < fnum; ++f) {
      mph_[f].build(l2o[f]);
      l2o_[f].assign(l2o[f].size(), 0);
      for (oid_t k : l2o[f]) l2o_[f][mph_[f].lookup(k)] = k;
      total_vertices_ += l2o[f].size();
    }
  }

  // Rebuild hashmap mode from already-replicated per-fragment oid lists
  // (deserialization path — no communication).
  void init_hashmap_local(int fnum, PartitionerKind pk,
                          std::vector<std::vector<oid_t>> l2o) {
    fnum_ = fnum;
    parser_.init(fnum);
    idxer_ = IdxerKind::kHashmap;
    pkind_ = pk;
    l2o_ = std::move(l2o);
    o2g_.clear();
    total_vertices_ = 0;
    for (int f = 0; f < fnum; ++f) total_vertices_ += l2o_[f].size();
    o2g_.reserve(total_vertices_);
    for (int f = 0; f < fnum; ++f)
      for (size_t l = 0; l < l2o_[f].size(); ++l)
        o2g_.emplace(l2o_[f][l], parser_.gid(f, static_cast<vid_t>(l)));
  }

  const std::vector<oid_t>& frag_oids(fid_t f) const { return l2o_[f]; }

  uint64_t total_vertices() const { return total_vertices_; }
  const IdParser& parser() const { return parser_; }
  int fnum() const { return fnum_; }
  IdxerKind idxer() const { return idxer_; }
  PartitionerKind partitioner() const { return pkind_; }
  const std::vector<uint64_t>& segments() const { return seg_; }

 private:
  int fnum_ = 1;
  IdParser parser_;
  IdxerKind idxer_ = IdxerKind::kIdentity;
  PartitionerKind pkind_ = PartitionerKind::kSegmented;
  uint64_t total_vertices_ = 0;
  std::vector<uint64_t> seg_;                   // identity/segmented
  std::vector<std::vector<oid_t>> l2o_;         // hashmap/mph: lid->oid
  std::unordered_map<oid_t, vid_t> o2g_;        // hashmap: oid->gid
  std::vector<MinimalPerfectHash> mph_;         // mph: per-fid oid->lid
};

}  // namespace grapehip
