// grapehip — Python bindings (pybind11).
//
// Engine = one rank of the distributed runtime (TCP control plane; RCCL data
// plane on GPU). Graph = one edge-cut fragment (host CSR and/or device
// graph). App entry points return (oid array, value array) for this rank's
// owned vertices plus timing metadata.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <chrono>
#include <fstream>
#include <limits>
#include <memory>
#include <optional>
#include <unordered_set>

#include "apps/bc.hpp"
#include "apps/auto_app.hpp"
#include "apps/bfs.hpp"
#include "apps/cdlp.hpp"
#include "apps/kclique.hpp"
#include "apps/kcore.hpp"
#include "apps/lcc.hpp"
#include "apps/pagerank.hpp"
#include "apps/sampler.hpp"
#include "apps/sssp.hpp"
#include "apps/wcc.hpp"
#include "core/fragment.hpp"
#include "core/serialize.hpp"
#include "core/vertexcut.hpp"
#include "core/message_manager.hpp"
#include "core/net.hpp"

#ifdef GRAPEHIP_WITH_HIP
#include "hip/gpu_engine.hpp"
#endif

namespace py = pybind11;
using namespace grapehip;

namespace {

double now_s() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

struct PyGraph {
  std::shared_ptr<VertexMap> vm;
  std::unique_ptr<Fragment> frag;  // host CSR (may be null for device-only)
#ifdef GRAPEHIP_WITH_HIP
  std::unique_ptr<DeviceGraph> dev;
#endif
  uint64_t nv() const {
#ifdef GRAPEHIP_WITH_HIP
    if (dev) return dev->nv_global;
#endif
    return frag->total_vertices();
  }
  uint64_t ne() const {
#ifdef GRAPEHIP_WITH_HIP
    if (dev) return dev->total_edges;
#endif
    return frag->total_edges();
  }
  uint64_t input_ne() const {
#ifdef GRAPEHIP_WITH_HIP
    if (dev) return dev->input_edges;
#endif
    return frag->input_edges();
  }
};

struct PyEngine {
  TcpComm comm;
  int rank = 0, world = 1, n_threads = 0;
  bool use_gpu = false;
#ifdef GRAPEHIP_WITH_HIP
  std::unique_ptr<GpuContext> gpu;
#endif

  PyEngine(int rank_, int world_, const std::string& addr, int port,
           int threads, bool gpu_) {
    rank = rank_;
    world = world_;
    use_gpu = gpu_;
    if (threads > 0) ThreadPool::Get().set_num_threads(threads);
    n_threads = ThreadPool::Get().num_threads();
    comm.init(rank, world, addr, port);
#ifdef GRAPEHIP_WITH_HIP
    if (use_gpu) gpu = std::make_unique<GpuContext>(&comm, rank, world);
#else
    if (use_gpu)
      throw std::runtime_error(
          "grapehip was built without HIP support but gpu=True was "
          "requested — rebuild with GRAPEHIP_WITH_HIP");
#endif
  }

  TcpComm* c() { return world > 1 ? &comm : nullptr; }
};

using arr_i64 = py::array_t<int64_t, py::array::c_style | py::array::forcecast>;
using arr_f32 = py::array_t<float, py::array::c_style | py::array::forcecast>;

std::shared_ptr<PyGraph> load_edges(
    PyEngine& eng, arr_i64 src, arr_i64 dst, std::optional<arr_f32> weights,
    bool directed, int64_t num_vertices, std::optional<arr_i64> vertex_oids,
    bool build_in_csr, const std::string& partitioner,
    const std::string& idxer) {
  size_t n = src.size();
  if (static_cast<size_t>(dst.size()) != n)
    throw std::runtime_error("src/dst size mismatch");
  bool weighted = weights.has_value();
  if (weighted && static_cast<size_t>(weights->size()) != n)
    throw std::runtime_error("weights size mismatch (expected one per edge)");
  std::vector<EdgeTriple> edges(n);
  {
    auto s = src.unchecked<1>();
    auto d = dst.unchecked<1>();
    const float* w = weighted ? weights->data() : nullptr;
    for (size_t i = 0; i < n; ++i) edges[i] = {s(i), d(i), w ? w[i] : 1.0f};
  }

  auto g = std::make_shared<PyGraph>();
  g->vm = std::make_shared<VertexMap>();
  py::gil_scoped_release rel;

  if (vertex_oids.has_value()) {
    const int64_t* po = vertex_oids->data();
    size_t nv = vertex_oids->size();
    std::vector<oid_t> owned;
    if (idxer == "mph" && partitioner == "map")
      throw std::runtime_error(
          "idxer=mph requires hash ownership (partitioner=hash)");
    if (partitioner == "map") {
      // MapPartitioner semantics (reference partitioner.h:103): ownership
      // = the rank that supplied the oid (e.g. a rebalanced range)
      owned.assign(po, po + nv);
      std::sort(owned.begin(), owned.end());
      owned.erase(std::unique(owned.begin(), owned.end()), owned.end());
    } else {
      // HashPartitioner: shuffle oids to hash owners
      std::vector<std::vector<oid_t>> bins(eng.world);
      for (size_t i = 0; i < nv; ++i)
        bins[hash_oid(po[i]) % eng.world].push_back(po[i]);
      std::vector<std::string> send(eng.world);
      for (int f = 0; f < eng.world; ++f)
        send[f].assign(reinterpret_cast<const char*>(bins[f].data()),
                       bins[f].size() * sizeof(oid_t));
      std::vector<std::string> recv =
          eng.c() ? eng.c()->exchange_all(send) : std::move(send);
      for (auto& blob : recv) {
        size_t m = blob.size() / sizeof(oid_t);
        const oid_t* p = reinterpret_cast<const oid_t*>(blob.data());
        owned.insert(owned.end(), p, p + m);
      }
      std::sort(owned.begin(), owned.end());
      owned.erase(std::unique(owned.begin(), owned.end()), owned.end());
    }
    if (idxer == "mph")
      g->vm->init_mph(eng.world, eng.c(), std::move(owned));
    else
      g->vm->init_hashmap(eng.world,
                          partitioner == "map" ? PartitionerKind::kMap
                                               : PartitionerKind::kHash,
                          eng.c(), std::move(owned));
  } else {
    if (num_vertices <= 0)
      throw std::runtime_error("num_vertices required for identity mapping");
    g->vm->init_identity(eng.world, static_cast<uint64_t>(num_vertices));
  }

  g->frag = Fragment::Build(g->vm, eng.c(), eng.rank, eng.world,
                            std::move(edges), directed, weighted,
                            build_in_csr, n);
#ifdef GRAPEHIP_WITH_HIP
  if (eng.use_gpu) g->dev = eng.gpu->upload(*g->frag);
#endif
  return g;
}

template <typename T>
py::array_t<T> to_np(const std::vector<T>& v) {
  py::array_t<T> a(v.size());
  std::memcpy(a.mutable_data(), v.data(), v.size() * sizeof(T));
  return a;
}

py::array_t<int64_t> inner_oids(const Fragment& f) {
  py::array_t<int64_t> a(f.ivnum());
  auto* p = a.mutable_data();
  for (vid_t v = 0; v < f.ivnum(); ++v) p[v] = f.lid2oid(v);
  return a;
}

template <typename RunFn>
py::dict run_timed(PyEngine& eng, RunFn&& run) {
  py::dict out;
  double t0, t1;
  int rounds;
  {
    py::gil_scoped_release rel;
    if (eng.c()) eng.c()->barrier();
    t0 = now_s();
    rounds = run();
    if (eng.c()) eng.c()->barrier();
    t1 = now_s();
  }
  double secs = t1 - t0;
  if (eng.c()) secs = eng.c()->allreduce_max_double(secs);
  out["rounds"] = rounds;
  out["seconds"] = secs;
  return out;
}

#ifdef GRAPEHIP_WITH_HIP
// device id <-> oid translation: identity maps use oids as device ids;
// hashmap maps are densely renumbered on upload (dev = fid*slice + lid)
struct DevIdMap {
  const Fragment* frag = nullptr;  // null => identity (dev id == oid)
  uint32_t slice = 0;
  int64_t to_oid(int64_t dev) const {
    if (!frag) return dev;
    const IdParser& P = frag->parser();
    fid_t f = static_cast<fid_t>(dev / slice);
    vid_t lid = static_cast<vid_t>(dev % slice);
    return frag->vm().get_oid(P.gid(f, lid));
  }
  uint32_t to_dev(int64_t oid) const {
    if (!frag) return static_cast<uint32_t>(oid);
    vid_t gid;
    if (!frag->vm().get_gid(oid, &gid))
      throw std::runtime_error("unknown vertex id " + std::to_string(oid));
    const IdParser& P = frag->parser();
    return static_cast<uint32_t>(
        static_cast<uint64_t>(P.fid(gid)) * slice + P.lid(gid));
  }
};

DevIdMap dev_id_map(const PyGraph& g) {
  DevIdMap m;
  if (g.frag && g.frag->vm().idxer() != IdxerKind::kIdentity) {
    m.frag = g.frag.get();
    m.slice = g.dev->seg_host.size() > 1
                  ? g.dev->seg_host[1] - g.dev->seg_host[0]
                  : g.dev->nv_global;
  }
  return m;
}

py::dict gpu_dict(const GpuRunResult& r, const PyGraph& pg, bool is_i64,
                  bool with_values, bool labels_are_ids = false,
                  const std::vector<int64_t>* label_lut = nullptr) {
  const DeviceGraph& g = *pg.dev;
  py::dict out;
  out["rounds"] = r.rounds;
  out["seconds"] = r.seconds;
  out["traversed_edges"] = r.traversed_edges;
  out["bytes_p2p"] = r.bytes_p2p;
  out["bytes_coll"] = r.bytes_coll;
  if (with_values) {
    DevIdMap m = dev_id_map(pg);
    // hashmap uploads pad the owned range; emit only real vertices
    uint32_t n = m.frag ? m.frag->ivnum() : g.owned();
    py::array_t<int64_t> oids(n);
    auto* p = oids.mutable_data();
    if (!m.frag && g.permuted) {
      // hub renumbering: row i holds old id inv[v_begin+i] (cached)
      DeviceGraph& gm = const_cast<DeviceGraph&>(g);
      if (gm.inv_host.size() != n) {
        gm.inv_host.resize(n);
        if (hipMemcpy(gm.inv_host.data(), gm.inv.data() + gm.v_begin,
                      static_cast<size_t>(n) * 4,
                      hipMemcpyDeviceToHost) != hipSuccess)
          throw std::runtime_error("inv fetch failed");
      }
      for (uint32_t i = 0; i < n; ++i) p[i] = gm.inv_host[i];
    } else {
      for (uint32_t i = 0; i < n; ++i)
        p[i] = m.frag ? m.frag->lid2oid(i)
                      : static_cast<int64_t>(g.v_begin) + i;
    }
    out["oids"] = oids;
    if (is_i64) {
      py::array_t<int64_t> vals(n);
      auto* q = vals.mutable_data();
      for (uint32_t i = 0; i < n; ++i)
        q[i] = (label_lut && r.i64[i] >= 0 &&
                static_cast<size_t>(r.i64[i]) < label_lut->size())
                   ? (*label_lut)[r.i64[i]]
                   : ((labels_are_ids && m.frag &&
                       r.i64[i] != std::numeric_limits<int64_t>::max())
                          ? m.to_oid(r.i64[i])
                          : r.i64[i]);
      out["values"] = vals;
    } else {
      py::array_t<double> vals(n);
      std::memcpy(vals.mutable_data(), r.f64.data(), n * 8);
      out["values"] = vals;
    }
  }
  return out;
}
#endif

}  // namespace

PYBIND11_MODULE(_core, m) {
  m.doc() = "grapehip core engine";

  py::class_<VertexcutFragment, std::shared_ptr<VertexcutFragment>>(
      m, "VertexcutGraph")
      .def_property_readonly("num_vertices",
                             &VertexcutFragment::num_vertices)
      .def_property_readonly("num_edges", &VertexcutFragment::total_edges)
      .def_property_readonly("local_edges", &VertexcutFragment::local_edges);

  py::class_<PyGraph, std::shared_ptr<PyGraph>>(m, "Graph")
      .def_property_readonly("num_vertices",
                             [](const PyGraph& g) { return g.nv(); })
      .def_property_readonly("num_edges",
                             [](const PyGraph& g) { return g.ne(); })
      .def_property_readonly("input_edges",
                             [](const PyGraph& g) { return g.input_ne(); })
      .def_property_readonly(
          "ivnum",
          [](const PyGraph& g) {
#ifdef GRAPEHIP_WITH_HIP
            if (g.dev) return static_cast<uint32_t>(g.dev->owned());
#endif
            return g.frag->ivnum();
          })
      .def_property_readonly("ovnum", [](const PyGraph& g) {
        return g.frag ? g.frag->ovnum() : 0;
      });

  py::class_<PyEngine>(m, "Engine")
      .def(py::init<int, int, const std::string&, int, int, bool>(),
           py::arg("rank") = 0, py::arg("world") = 1,
           py::arg("master_addr") = "127.0.0.1",
           py::arg("master_port") = 29517, py::arg("n_threads") = 0,
           py::arg("gpu") = false)
      .def_readonly("rank", &PyEngine::rank)
      .def_readonly("world", &PyEngine::world)
      .def_property_readonly("gpu",
                             [](const PyEngine& e) { return e.use_gpu; })
      .def("barrier",
           [](PyEngine& e) {
             py::gil_scoped_release rel;
             if (e.c()) e.c()->barrier();
           })
      .def("_debug_scan",
           [](PyEngine& e, std::vector<uint32_t> in) {
#ifdef GRAPEHIP_WITH_HIP
             return e.gpu->debug_scan(in);
#else
             throw std::runtime_error("no hip");
#endif
           })
      .def("allreduce_max",
           [](PyEngine& e, double v) {
             py::gil_scoped_release rel;
             return e.c() ? e.c()->allreduce_max_double(v) : v;
           })
      .def("device_sync",
           [](PyEngine& e) {
#ifdef GRAPEHIP_WITH_HIP
             if (e.gpu) e.gpu->device_sync();
#endif
           })
      .def("memory_info",
           [](PyEngine& eng) {
             // reference GetMemoryUsage (util.h:51) + TRACKING_MEMORY
             py::dict out;
             std::ifstream st("/proc/self/status");
             std::string line;
             while (std::getline(st, line)) {
               if (line.rfind("VmHWM", 0) == 0 ||
                   line.rfind("VmRSS", 0) == 0) {
                 auto c = line.find(':');
                 std::string key = line.substr(0, c);
                 long kb = atol(line.c_str() + c + 1);
                 out[key.c_str()] = kb * 1024L;
               }
             }
#ifdef GRAPEHIP_WITH_HIP
             if (eng.use_gpu) {
               size_t free_b = 0, total_b = 0;
               if (hipMemGetInfo(&free_b, &total_b) == hipSuccess) {
                 out["hip_free"] = free_b;
                 out["hip_total"] = total_b;
                 out["hip_used"] = total_b - free_b;
               }
               // engine-tracked DeviceBuffer bytes (reference
               // utils/memory_tracker parity): current + high-water
               out["hip_alloc_current"] = hip_alloc_current().load();
               out["hip_alloc_peak"] = hip_alloc_peak().load();
             }
#endif
             return out;
           })
      .def("_exchange_all",
           [](PyEngine& eng, std::vector<py::bytes> blobs) {
             if (static_cast<int>(blobs.size()) != eng.world)
               throw std::runtime_error(
                   "_exchange_all needs one blob per rank");
             std::vector<std::string> send(eng.world);
             for (int f = 0; f < eng.world; ++f)
               send[f] = std::string(blobs[f]);
             std::vector<std::string> recv;
             {
               py::gil_scoped_release rel;
               recv = eng.c() ? eng.c()->exchange_all(send)
                              : std::move(send);
             }
             std::vector<py::bytes> out;
             out.reserve(recv.size());
             for (auto& r : recv) out.emplace_back(r);
             return out;
           },
           py::arg("blobs"))
      // test hook: validates the cooperative-abort path (reference
      // ForceTerminate/TerminateInfo, default_message_manager.h:156-166) —
      // one rank aborts in PEval, EVERY rank must raise with its info
      .def("_test_force_terminate",
           [](PyEngine& eng, std::shared_ptr<PyGraph> g, int fail_rank) {
             struct AbortApp {
               int fail_rank, rank;
               void PEval(const Fragment&, int&, MessageManager& mm) {
                 if (rank == fail_rank)
                   mm.force_terminate("boom from rank " +
                                      std::to_string(rank));
                 else
                   mm.force_continue();
               }
               void IncEval(const Fragment&, int&, MessageManager&) {}
             };
             AbortApp app{fail_rank, eng.rank};
             int ctx = 0;
             MessageManager mm;
             mm.init(eng.c(), g->frag.get(), eng.n_threads);
             py::gil_scoped_release rel;
             return RunWorker(app, ctx, *g->frag, mm);
           },
           py::arg("graph"), py::arg("fail_rank"))
      .def("load_edges", &load_edges, py::arg("src"), py::arg("dst"),
           py::arg("weights") = std::nullopt, py::arg("directed") = false,
           py::arg("num_vertices") = -1, py::arg("vertex_oids") = std::nullopt,
           py::arg("build_in_csr") = false,
           py::arg("partitioner") = "segmented",
           py::arg("idxer") = "hashmap")
      .def("mutate_graph",
           [](PyEngine& eng, std::shared_ptr<PyGraph> gp, arr_i64 add_src,
              arr_i64 add_dst, std::optional<arr_f32> add_w, arr_i64 rm_src,
              arr_i64 rm_dst, arr_i64 rm_vertices) {
             // Reference parity: LoadGraphAndMutate / Mutation{add/remove
             // edges,vertices} (loader.h:55-68, basic_fragment_mutator.h,
             // mutable_edgecut_fragment.h:289-399). Edge-only deltas apply
             // IN PLACE with cost proportional to the delta (slack-tracked
             // rows, end-arena relocation); vertex removals and
             // weight-introducing deltas fall back to a functional rebuild.
             PyGraph& g = *gp;
             if (!g.frag)
               throw std::runtime_error("mutate_graph needs a host fragment");
             size_t na = add_src.size();
             if (static_cast<size_t>(add_dst.size()) != na)
               throw std::runtime_error("add src/dst size mismatch");
             size_t nr = rm_src.size();
             if (static_cast<size_t>(rm_dst.size()) != nr)
               throw std::runtime_error("remove src/dst size mismatch");
             const bool delta_ok =
                 rm_vertices.size() == 0 &&
                 (!add_w.has_value() || g.frag->has_weights());
             if (delta_ok) {
               std::vector<EdgeTriple> adds(na);
               {
                 auto sp = add_src.unchecked<1>();
                 auto dp = add_dst.unchecked<1>();
                 const float* wp = add_w ? add_w->data() : nullptr;
                 for (size_t i = 0; i < na; ++i)
                   adds[i] = {sp(i), dp(i), wp ? wp[i] : 1.0f};
               }
               std::vector<std::pair<oid_t, oid_t>> rms(nr);
               {
                 auto sp = rm_src.unchecked<1>();
                 auto dp = rm_dst.unchecked<1>();
                 for (size_t i = 0; i < nr; ++i)
                   rms[i] = {sp(i), dp(i)};
               }
               py::gil_scoped_release rel;
               g.frag->MutateDelta(eng.c(), adds, rms);
#ifdef GRAPEHIP_WITH_HIP
               if (eng.use_gpu) {
                 g.frag->compact();
                 g.dev = eng.gpu->upload(*g.frag);
               }
#endif
               return gp;
             }
             auto out = std::make_shared<PyGraph>();
             out->vm = g.vm;
             bool weighted =
                 g.frag->has_weights() || add_w.has_value();
             py::gil_scoped_release rel;

             // replicate removal lists to every rank
             auto replicate = [&](const int64_t* p, size_t n) {
               std::vector<oid_t> all(p, p + n);
               if (eng.c()) {
                 std::string mine(reinterpret_cast<const char*>(p),
                                  n * sizeof(int64_t));
                 std::vector<std::string> send(eng.world, mine);
                 send[eng.rank].clear();
                 auto recv = eng.c()->exchange_all(send);
                 for (int f = 0; f < eng.world; ++f) {
                   if (f == eng.rank) continue;
                   size_t m = recv[f].size() / sizeof(int64_t);
                   const int64_t* q =
                       reinterpret_cast<const int64_t*>(recv[f].data());
                   all.insert(all.end(), q, q + m);
                 }
               }
               return all;
             };
             std::vector<oid_t> rms = replicate(rm_src.data(), nr);
             std::vector<oid_t> rmd = replicate(rm_dst.data(), nr);
             std::vector<oid_t> rmv =
                 replicate(rm_vertices.data(), rm_vertices.size());

             struct PairHash {
               size_t operator()(const std::pair<oid_t, oid_t>& p) const {
                 return std::hash<uint64_t>()(
                     static_cast<uint64_t>(p.first) * 0x9e3779b97f4a7c15ULL ^
                     static_cast<uint64_t>(p.second));
               }
             };
             std::unordered_set<std::pair<oid_t, oid_t>, PairHash> rm_edges;
             for (size_t i = 0; i < rms.size(); ++i) {
               rm_edges.emplace(rms[i], rmd[i]);
               if (!g.frag->directed()) rm_edges.emplace(rmd[i], rms[i]);
             }
             std::unordered_set<oid_t> rm_verts(rmv.begin(), rmv.end());

             std::vector<EdgeTriple> triples = g.frag->to_triples();
             uint64_t kept = 0;
             for (auto& e : triples) {
               if (rm_edges.count({e.src, e.dst})) continue;
               if (rm_verts.count(e.src) || rm_verts.count(e.dst)) continue;
               triples[kept++] = e;
             }
             triples.resize(kept);
             {
               auto sp = add_src.unchecked<1>();
               auto dp = add_dst.unchecked<1>();
               const float* wp = add_w ? add_w->data() : nullptr;
               for (size_t i = 0; i < na; ++i)
                 triples.push_back({sp(i), dp(i), wp ? wp[i] : 1.0f});
             }
             uint64_t n_input = triples.size();
             out->frag = Fragment::Build(
                 out->vm, eng.c(), eng.rank, eng.world, std::move(triples),
                 g.frag->directed(), weighted, g.frag->has_in_csr(),
                 n_input);
#ifdef GRAPEHIP_WITH_HIP
             if (eng.use_gpu) out->dev = eng.gpu->upload(*out->frag);
#endif
             return out;
           },
           py::arg("graph"), py::arg("add_src"), py::arg("add_dst"),
           py::arg("add_weights") = std::nullopt, py::arg("remove_src"),
           py::arg("remove_dst"), py::arg("remove_vertices"))
      .def("save_graph",
           [](PyEngine& eng, PyGraph& g, const std::string& prefix) {
             if (!g.frag)
               throw std::runtime_error(
                   "save_graph needs a host fragment (device-only synthetic "
                   "graphs are regenerated, not checkpointed)");
             py::gil_scoped_release rel;
             serialize_graph(*g.frag, prefix);
             if (eng.c()) eng.c()->barrier();
           },
           py::arg("graph"), py::arg("prefix"))
      .def("load_serialized",
           [](PyEngine& eng, const std::string& prefix) {
             auto g = std::make_shared<PyGraph>();
             py::gil_scoped_release rel;
             auto [vm, frag] = deserialize_graph(
                 prefix, eng.rank, static_cast<uint32_t>(eng.world));
             g->vm = vm;
             g->frag = std::move(frag);
#ifdef GRAPEHIP_WITH_HIP
             if (eng.use_gpu) g->dev = eng.gpu->upload(*g->frag);
#endif
             if (eng.c()) eng.c()->barrier();
             return g;
           },
           py::arg("prefix"))
      .def("load_synthetic",
           [](PyEngine& eng, uint64_t nv, uint64_t ne, uint64_t seed,
              bool directed, bool weighted, bool build_in_csr, double a,
              double b, double c) {
#ifdef GRAPEHIP_WITH_HIP
             if (!eng.use_gpu)
               throw std::runtime_error("load_synthetic requires gpu=True");
             auto g = std::make_shared<PyGraph>();
             py::gil_scoped_release rel;
             g->dev = eng.gpu->gen_synthetic(nv, ne, seed, directed, weighted,
                                             build_in_csr, a, b, c);
             return g;
#else
             throw std::runtime_error("built without HIP");
#endif
           },
           py::arg("num_vertices"), py::arg("num_edges"), py::arg("seed") = 42,
           py::arg("directed") = false, py::arg("weighted") = false,
           py::arg("build_in_csr") = false,
           py::arg("a") = 0.57, py::arg("b") = 0.19, py::arg("c") = 0.19)
      .def("bfs",
           [](PyEngine& eng, PyGraph& g, int64_t source, bool values) {
#ifdef GRAPEHIP_WITH_HIP
             if (eng.use_gpu) {
               GpuRunResult r;
               {
                 py::gil_scoped_release rel;
                 r = eng.gpu->bfs(*g.dev, dev_id_map(g).to_dev(source),
                                  values);
               }
               return gpu_dict(r, g, true, values);
             }
#endif
             BFSApp app;
             BFSContext ctx;
             MessageManager mm;
             mm.init(eng.c(), g.frag.get(), eng.n_threads);
             ctx.init(*g.frag, source);
             py::dict meta = run_timed(
                 eng, [&] { return RunWorker(app, ctx, *g.frag, mm); });
             std::vector<int64_t> vals(g.frag->ivnum());
             for (vid_t v = 0; v < g.frag->ivnum(); ++v)
               vals[v] = ctx.depth[v].load(std::memory_order_relaxed);
             meta["oids"] = inner_oids(*g.frag);
             meta["values"] = to_np(vals);
             return meta;
           },
           py::arg("graph"), py::arg("source") = 0,
           py::arg("values") = true)
      .def("sssp",
           [](PyEngine& eng, PyGraph& g, int64_t source, float delta,
              bool values) {
#ifdef GRAPEHIP_WITH_HIP
             if (eng.use_gpu) {
               GpuRunResult r;
               {
                 py::gil_scoped_release rel;
                 r = eng.gpu->sssp(*g.dev, dev_id_map(g).to_dev(source),
                                   delta, values);
               }
               return gpu_dict(r, g, false, values);
             }
#endif
             SSSPApp app;
             SSSPContext ctx;
             MessageManager mm;
             mm.init(eng.c(), g.frag.get(), eng.n_threads);
             ctx.init(*g.frag, source);
             py::dict meta = run_timed(
                 eng, [&] { return RunWorker(app, ctx, *g.frag, mm); });
             std::vector<double> vals(g.frag->ivnum());
             for (vid_t v = 0; v < g.frag->ivnum(); ++v)
               vals[v] = ctx.dist[v].load(std::memory_order_relaxed);
             meta["oids"] = inner_oids(*g.frag);
             meta["values"] = to_np(vals);
             return meta;
           },
           py::arg("graph"), py::arg("source") = 0, py::arg("delta") = -1.0f,
           py::arg("values") = true)
      .def("pagerank",
           [](PyEngine& eng, PyGraph& g, double damping, int iters,
              double tol, bool values) {
#ifdef GRAPEHIP_WITH_HIP
             if (eng.use_gpu) {
               GpuRunResult r;
               {
                 py::gil_scoped_release rel;
                 r = eng.gpu->pagerank(*g.dev, damping, iters, tol, values);
               }
               return gpu_dict(r, g, false, values);
             }
#endif
             PageRankApp app;
             PageRankContext ctx;
             MessageManager mm;
             mm.init(eng.c(), g.frag.get(), eng.n_threads);
             ctx.init(*g.frag, damping, iters, tol);
             py::dict meta = run_timed(
                 eng, [&] { return RunWorker(app, ctx, *g.frag, mm); });
             meta["oids"] = inner_oids(*g.frag);
             meta["values"] = to_np(ctx.rank);
             return meta;
           },
           py::arg("graph"), py::arg("damping") = 0.85, py::arg("iters") = 10,
           py::arg("tol") = 0.0, py::arg("values") = true)
      .def("wcc",
           [](PyEngine& eng, PyGraph& g, bool values) {
#ifdef GRAPEHIP_WITH_HIP
             if (eng.use_gpu) {
               GpuRunResult r;
               {
                 py::gil_scoped_release rel;
                 r = eng.gpu->wcc(*g.dev, values);
               }
               return gpu_dict(r, g, true, values, /*labels_are_ids=*/true);
             }
#endif
             WCCApp app;
             WCCContext ctx;
             MessageManager mm;
             mm.init(eng.c(), g.frag.get(), eng.n_threads);
             ctx.init(*g.frag);
             py::dict meta = run_timed(
                 eng, [&] { return RunWorker(app, ctx, *g.frag, mm); });
             std::vector<int64_t> vals(g.frag->ivnum());
             for (vid_t v = 0; v < g.frag->ivnum(); ++v)
               vals[v] = ctx.label[v].load(std::memory_order_relaxed);
             meta["oids"] = inner_oids(*g.frag);
             meta["values"] = to_np(vals);
             return meta;
           },
           py::arg("graph"), py::arg("values") = true)
      .def("cdlp",
           [](PyEngine& eng, PyGraph& g, int iters, bool values) {
#ifdef GRAPEHIP_WITH_HIP
             if (eng.use_gpu) {
               GpuRunResult r;
               {
                 py::gil_scoped_release rel;
                 r = eng.gpu->cdlp(*g.dev, iters, values);
               }
               // non-identity maps run CDLP in global sorted-OID order
               // space (reference tie-break semantics); translate back
               std::vector<int64_t> lut;
               if (g.frag &&
                   g.frag->vm().idxer() != IdxerKind::kIdentity) {
                 const VertexMap& vm = g.frag->vm();
                 const IdParser& P = g.frag->parser();
                 lut.reserve(vm.total_vertices());
                 for (int f = 0; f < vm.fnum(); ++f) {
                   vid_t nn = vm.frag_vnum(static_cast<fid_t>(f));
                   for (vid_t l = 0; l < nn; ++l)
                     lut.push_back(
                         vm.get_oid(P.gid(static_cast<fid_t>(f), l)));
                 }
                 std::sort(lut.begin(), lut.end());
               }
               return gpu_dict(r, g, true, values, /*labels_are_ids=*/true,
                               lut.empty() ? nullptr : &lut);
             }
#endif
             CDLPApp app;
             CDLPContext ctx;
             MessageManager mm;
             mm.init(eng.c(), g.frag.get(), eng.n_threads);
             ctx.init(*g.frag, iters);
             py::dict meta = run_timed(
                 eng, [&] { return RunWorker(app, ctx, *g.frag, mm); });
             meta["oids"] = inner_oids(*g.frag);
             std::vector<int64_t> vals(ctx.label.begin(),
                                       ctx.label.begin() + g.frag->ivnum());
             meta["values"] = to_np(vals);
             return meta;
           },
           py::arg("graph"), py::arg("iters") = 10,
           py::arg("values") = true)
      .def("lcc",
           [](PyEngine& eng, PyGraph& g, bool values) {
#ifdef GRAPEHIP_WITH_HIP
             if (eng.use_gpu) {
               GpuRunResult r;
               {
                 py::gil_scoped_release rel;
                 r = eng.gpu->lcc(*g.dev, values);
               }
               return gpu_dict(r, g, false, values);
             }
#endif
             LCCApp app;
             LCCContext ctx;
             MessageManager mm;
             mm.init(eng.c(), g.frag.get(), eng.n_threads);
             ctx.init(*g.frag);
             py::dict meta = run_timed(
                 eng, [&] { return RunWorker(app, ctx, *g.frag, mm); });
             meta["oids"] = inner_oids(*g.frag);
             meta["values"] = to_np(ctx.lcc);
             return meta;
           },
           py::arg("graph"), py::arg("values") = true)
      .def("bc",
           [](PyEngine& eng, PyGraph& g, int64_t source) {
             if (!g.frag)
               throw std::runtime_error(
                   "bc runs on the CPU engine and needs a host fragment "
                   "(load with load_edges)");
             BCApp app;
             BCContext ctx;
             MessageManager mm;
             mm.init(eng.c(), g.frag.get(), eng.n_threads);
             ctx.init(*g.frag, source);
             py::dict meta = run_timed(
                 eng, [&] { return RunWorker(app, ctx, *g.frag, mm); });
             std::vector<double> vals(g.frag->ivnum());
             std::vector<double> sig(g.frag->ivnum());
             std::vector<int64_t> dep(g.frag->ivnum());
             for (vid_t v = 0; v < g.frag->ivnum(); ++v) {
               vals[v] = ctx.delta[v].load(std::memory_order_relaxed);
               sig[v] = ctx.sigma[v].load(std::memory_order_relaxed);
               dep[v] = ctx.depth[v].load(std::memory_order_relaxed);
             }
             meta["oids"] = inner_oids(*g.frag);
             meta["values"] = to_np(vals);
             meta["path_num"] = to_np(sig);
             meta["depth"] = to_np(dep);
             return meta;
           },
           py::arg("graph"), py::arg("source") = 0)
      .def("kcore",
           [](PyEngine& eng, PyGraph& g, int k) {
             if (!g.frag)
               throw std::runtime_error("kcore needs a host fragment");
             KCoreApp app;
             KCoreContext ctx;
             MessageManager mm;
             mm.init(eng.c(), g.frag.get(), eng.n_threads);
             ctx.init(*g.frag, k);
             py::dict meta = run_timed(
                 eng, [&] { return RunWorker(app, ctx, *g.frag, mm); });
             std::vector<int64_t> vals(g.frag->ivnum());
             for (vid_t v = 0; v < g.frag->ivnum(); ++v)
               vals[v] = ctx.removed[v] ? 0 : 1;
             meta["oids"] = inner_oids(*g.frag);
             meta["values"] = to_np(vals);
             return meta;
           },
           py::arg("graph"), py::arg("k") = 3)
      .def("core_decomposition",
           [](PyEngine& eng, PyGraph& g) {
             if (!g.frag)
               throw std::runtime_error(
                   "core_decomposition needs a host fragment");
             CoreDecompApp app;
             CoreDecompContext ctx;
             MessageManager mm;
             mm.init(eng.c(), g.frag.get(), eng.n_threads);
             ctx.init(*g.frag);
             py::dict meta = run_timed(
                 eng, [&] { return RunWorker(app, ctx, *g.frag, mm); });
             std::vector<int64_t> vals(g.frag->ivnum());
             for (vid_t v = 0; v < g.frag->ivnum(); ++v)
               vals[v] = ctx.est[v].load(std::memory_order_relaxed);
             meta["oids"] = inner_oids(*g.frag);
             meta["values"] = to_np(vals);
             return meta;
           },
           py::arg("graph"))
      .def("load_vertexcut",
           [](PyEngine& eng, arr_i64 src, arr_i64 dst, int64_t num_vertices,
              int bucket_num) {
             size_t n = src.size();
             std::vector<uint32_t> s32(n), d32(n);
             auto sp = src.unchecked<1>();
             auto dp = dst.unchecked<1>();
             for (size_t i = 0; i < n; ++i) {
               s32[i] = static_cast<uint32_t>(sp(i));
               d32[i] = static_cast<uint32_t>(dp(i));
             }
             py::gil_scoped_release rel;
             return std::shared_ptr<VertexcutFragment>(
                 VertexcutFragment::Build(eng.c(), eng.rank, eng.world,
                                          static_cast<uint64_t>(num_vertices),
                                          s32, d32, bucket_num));
           },
           py::arg("src"), py::arg("dst"), py::arg("num_vertices"),
           py::arg("bucket_num") = 8)
      .def("pagerank_vc",
           [](PyEngine& eng, VertexcutFragment& g, double damping,
              int iters) {
             std::vector<double> r;
             double t0, t1;
             {
               py::gil_scoped_release rel;
               if (eng.c()) eng.c()->barrier();
               t0 = now_s();
               r = pagerank_vc(g, eng.c(), damping, iters);
               if (eng.c()) eng.c()->barrier();
               t1 = now_s();
             }
             // result is fully replicated; report this rank's segment
             uint64_t b = g.segments()[eng.rank];
             uint64_t e = g.segments()[eng.rank + 1];
             py::dict out;
             out["seconds"] = t1 - t0;
             out["rounds"] = iters;
             py::array_t<int64_t> oids(e - b);
             py::array_t<double> vals(e - b);
             for (uint64_t v = b; v < e; ++v) {
               oids.mutable_data()[v - b] = static_cast<int64_t>(v);
               vals.mutable_data()[v - b] = r[v];
             }
             out["oids"] = oids;
             out["values"] = vals;
             return out;
           },
           py::arg("graph"), py::arg("damping") = 0.85,
           py::arg("iters") = 10)
      .def("sample",
           [](PyEngine& eng, PyGraph& g, arr_i64 starts, int hops,
              const std::string& strategy, int top_k, uint64_t seed) {
             if (!g.frag)
               throw std::runtime_error("sample needs a host fragment");
             SampleStrategy st;
             if (strategy == "random")
               st = SampleStrategy::kRandom;
             else if (strategy == "edge_weight")
               st = SampleStrategy::kEdgeWeight;
             else if (strategy == "top_k")
               st = SampleStrategy::kTopK;
             else
               throw std::runtime_error(
                   "strategy must be random|edge_weight|top_k");
             std::vector<oid_t> st_oids(starts.data(),
                                        starts.data() + starts.size());
             SamplerApp app;
             SamplerContext ctx;
             MessageManager mm;
             py::dict meta;
             {
               py::gil_scoped_release rel;
               mm.init(eng.c(), g.frag.get(), eng.n_threads);
               ctx.init(*g.frag, st_oids, hops, st, top_k, seed);
               app.origin_map_.resize(st_oids.size());
               for (size_t i = 0; i < st_oids.size(); ++i) {
                 vid_t gid;
                 app.origin_map_[i] =
                     g.frag->vm().get_gid(st_oids[i], &gid)
                         ? g.frag->parser().fid(gid)
                         : 0;
               }
               RunWorker(app, ctx, *g.frag, mm);
             }
             size_t n = ctx.paths.size();
             py::array_t<int64_t> walks(n);
             py::array_t<int64_t> paths({n, static_cast<size_t>(hops + 1)});
             auto* wp = walks.mutable_data();
             auto* pp = paths.mutable_data();
             for (size_t i = 0; i < n; ++i) {
               wp[i] = static_cast<int64_t>(ctx.walk_ids[i]);
               for (int h = 0; h <= hops; ++h) {
                 vid_t gv = ctx.paths[i][h];
                 pp[i * (hops + 1) + h] =
                     gv == kInvalidVid ? -1 : g.frag->vm().get_oid(gv);
               }
             }
             py::dict out;
             out["walk_ids"] = walks;
             out["paths"] = paths;
             return out;
           },
           py::arg("graph"), py::arg("starts"), py::arg("hops") = 2,
           py::arg("strategy") = "random", py::arg("top_k") = 4,
           py::arg("seed") = 7)
      .def("sssp_auto",
           [](PyEngine& eng, PyGraph& g, int64_t source) {
             if (!g.frag)
               throw std::runtime_error("sssp_auto needs a host fragment");
             SSSPAutoApp app;
             SSSPAutoContext ctx;
             MessageManager mm;
             mm.init(eng.c(), g.frag.get(), eng.n_threads);
             ctx.init(*g.frag, source);
             py::dict meta = run_timed(
                 eng, [&] { return RunWorker(app, ctx, *g.frag, mm); });
             std::vector<double> vals(g.frag->ivnum());
             for (vid_t v = 0; v < g.frag->ivnum(); ++v)
               vals[v] = ctx.dist[v];
             meta["oids"] = inner_oids(*g.frag);
             meta["values"] = to_np(vals);
             return meta;
           },
           py::arg("graph"), py::arg("source") = 0)
      .def("kclique",
           [](PyEngine& eng, PyGraph& g, int k) {
             if (!g.frag)
               throw std::runtime_error("kclique needs a host fragment");
             if (k < 2) throw std::runtime_error("kclique: k must be >= 2");
             KCliqueApp app;
             KCliqueContext ctx;
             MessageManager mm;
             mm.init(eng.c(), g.frag.get(), eng.n_threads);
             ctx.init(*g.frag, k);
             py::dict meta = run_timed(
                 eng, [&] { return RunWorker(app, ctx, *g.frag, mm); });
             meta["clique_count"] = ctx.clique_num;
             return meta;
           },
           py::arg("graph"), py::arg("k") = 3);

  m.attr("WITH_HIP") =
#ifdef GRAPEHIP_WITH_HIP
      true;
#else
      false;
#endif
}
