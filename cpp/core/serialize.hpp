// grapehip — fragment + vertex-map checkpoint (serialize/deserialize).
// Reference parity: grape/fragment/immutable_edgecut_fragment.h:508-584
// (Serialize/Deserialize to "%s/frag_%d.s", config.h:67) and the vertex-map
// serialization it depends on; driven by --serialize/--deserialize in the
// loader (ev_fragment_loader.h:75-93). One binary file per fragment holds
// the vertex map + CSR arrays so reloads skip the partition/shuffle/build
// pipeline entirely.
#pragma once

#include <cstdio>
#include <memory>
#include <stdexcept>
#include <string>
#include <vector>

#include "fragment.hpp"
#include "vertex_map.hpp"

namespace grapehip {

namespace ser {

constexpr uint64_t kMagic = 0x47524150454849ULL;  // "GRAPEHI"
constexpr uint32_t kVersion = 1;

struct Writer {
  FILE* f;
  explicit Writer(const std::string& path) : f(fopen(path.c_str(), "wb")) {
    if (!f) throw std::runtime_error("serialize: cannot open " + path);
  }
  ~Writer() {
    if (f) fclose(f);
  }
  void raw(const void* p, size_t n) {
    if (fwrite(p, 1, n, f) != n)
      throw std::runtime_error("serialize: short write");
  }
  template <typename T>
  void pod(const T& v) {
    raw(&v, sizeof(T));
  }
  template <typename T>
  void vec(const std::vector<T>& v) {
    uint64_t n = v.size();
    pod(n);
    if (n) raw(v.data(), n * sizeof(T));
  }
};

struct Reader {
  FILE* f;
  explicit Reader(const std::string& path) : f(fopen(path.c_str(), "rb")) {
    if (!f) throw std::runtime_error("deserialize: cannot open " + path);
  }
  ~Reader() {
    if (f) fclose(f);
  }
  void raw(void* p, size_t n) {
    if (fread(p, 1, n, f) != n)
      throw std::runtime_error("deserialize: short read");
  }
  template <typename T>
  T pod() {
    T v;
    raw(&v, sizeof(T));
    return v;
  }
  template <typename T>
  void vec(std::vector<T>& v) {
    uint64_t n = pod<uint64_t>();
    v.resize(n);
    if (n) raw(v.data(), n * sizeof(T));
  }
};

inline std::string frag_path(const std::string& prefix, fid_t fid) {
  return prefix + "/frag_" + std::to_string(fid) + ".s";
}

}  // namespace ser

inline void serialize_graph(Fragment& frag, const std::string& prefix) {
  frag.compact();  // mutable-mode slack is not checkpointed
  ser::Writer w(ser::frag_path(prefix, frag.fid()));
  w.pod(ser::kMagic);
  w.pod(ser::kVersion);
  // vertex map
  const VertexMap& vm = frag.vm();
  w.pod(static_cast<uint32_t>(vm.fnum()));
  w.pod(static_cast<uint8_t>(vm.idxer()));
  w.pod(static_cast<uint8_t>(vm.partitioner()));
  w.pod(vm.total_vertices());
  w.vec(vm.segments());
  if (vm.idxer() == IdxerKind::kHashmap || vm.idxer() == IdxerKind::kMph) {
    for (int f = 0; f < vm.fnum(); ++f) w.vec(vm.frag_oids(f));
  }
  // fragment
  w.pod(static_cast<uint32_t>(frag.fid()));
  w.pod(static_cast<uint8_t>(frag.directed() ? 1 : 0));
  w.pod(frag.total_edges());
  w.pod(frag.input_edges());
  w.vec(frag.oe_offsets());
  w.vec(frag.oe_dsts());
  w.vec(frag.oe_weights());
  w.vec(frag.ie_offsets());
  w.vec(frag.ie_dsts());
  w.vec(frag.ie_weights());
  w.vec(frag.outer_gids());
  uint32_t fnum = frag.fnum();
  for (fid_t f = 0; f < static_cast<fid_t>(fnum); ++f) {
    auto [b, e] = frag.outer_range(f);
    w.pod(b);
    w.pod(e);
    w.vec(frag.mirrors(f));
  }
}

// expected_fnum: pass the running world size; a checkpoint written at a
// different world would silently route gids to fragments no rank serves
// (reference semantics: one frag file per rank, frag_%d.s, config.h:67).
// 0 skips the check (offline inspection tools).
inline std::pair<std::shared_ptr<VertexMap>, std::unique_ptr<Fragment>>
deserialize_graph(const std::string& prefix, fid_t fid,
                  uint32_t expected_fnum = 0) {
  ser::Reader r(ser::frag_path(prefix, fid));
  if (r.pod<uint64_t>() != ser::kMagic)
    throw std::runtime_error("deserialize: bad magic");
  if (r.pod<uint32_t>() != ser::kVersion)
    throw std::runtime_error("deserialize: version mismatch");
  auto vm = std::make_shared<VertexMap>();
  uint32_t fnum = r.pod<uint32_t>();
  if (expected_fnum && fnum != expected_fnum)
    throw std::runtime_error(
        "deserialize: checkpoint written at world " + std::to_string(fnum) +
        " but engine runs world " + std::to_string(expected_fnum));
  IdxerKind idx = static_cast<IdxerKind>(r.pod<uint8_t>());
  PartitionerKind pk = static_cast<PartitionerKind>(r.pod<uint8_t>());
  uint64_t nv = r.pod<uint64_t>();
  std::vector<uint64_t> seg;
  r.vec(seg);
  if (idx == IdxerKind::kIdentity) {
    vm->init_identity(fnum, nv);
    if (vm->segments() != seg)
      throw std::runtime_error(
          "deserialize: identity segments in checkpoint do not match the "
          "uniform ceil-slices init_identity derives — non-uniform "
          "segmented maps are not round-trippable");
  } else {
    std::vector<std::vector<oid_t>> oids(fnum);
    for (uint32_t f = 0; f < fnum; ++f) r.vec(oids[f]);
    if (idx == IdxerKind::kMph)
      vm->init_mph_local(fnum, std::move(oids));
    else
      vm->init_hashmap_local(fnum, pk, std::move(oids));
  }
  auto frag = Fragment::FromParts(vm, r, fnum);
  if (frag->fid() != fid)
    throw std::runtime_error(
        "deserialize: file " + ser::frag_path(prefix, fid) +
        " holds fragment " + std::to_string(frag->fid()));
  return {vm, std::move(frag)};
}

}  // namespace grapehip
