// grapehip — PageRank (LDBC Graphalytics semantics), CPU path.
// Reference parity: examples/analytical_apps/pagerank/pagerank.h
// (BatchShuffleApp: fixed pr_mr iterations, dangling-sum via
// Communicator::Sum, dense per-round sync of boundary ranks). We push
// d*rank/outdeg along out-edges with atomic adds, then flush per-peer dense
// partial-sum slices (contiguous outer ranges) — the batch-shuffle exchange.
// r'(v) = (1-d)/N + d*(sum_in contrib + dangling/N); unreachable semantics
// N/A. Output: double rank per vertex.
#pragma once

#include <cmath>
#include <vector>

#include "../core/fragment.hpp"
#include "../core/message_manager.hpp"

namespace grapehip {

struct PageRankContext {
  double damping;
  int max_iters;
  double tol = 0.0;  // >0: competitor-equivalent convergence (reference
                     // pagerank_local.h) — stop at global L1 delta < tol
  int iter = 0;
  std::vector<double> rank;               // over ivnum
  std::vector<std::atomic<double>> acc;   // over tvnum: pushed contributions

  void init(const Fragment& frag, double d, int mr, double tol_ = 0.0) {
    damping = d;
    max_iters = mr;
    tol = tol_;
    iter = 0;
    double r0 = 1.0 / static_cast<double>(frag.total_vertices());
    rank.assign(frag.ivnum(), r0);
    acc = std::vector<std::atomic<double>>(frag.tvnum());
    for (auto& a : acc) a.store(0.0, std::memory_order_relaxed);
  }
};

class PageRankApp {
 public:
  void PEval(const Fragment& frag, PageRankContext& ctx, MessageManager& mm) {
    if (ctx.max_iters > 0) mm.force_continue();
  }

  void IncEval(const Fragment& frag, PageRankContext& ctx,
               MessageManager& mm) {
    const vid_t iv = frag.ivnum();
    const double N = static_cast<double>(frag.total_vertices());
    const double d = ctx.damping;

    // dangling sum (vertices with no outgoing edges)
    double local_dangling = 0;
    {
      std::vector<double> per_thread(mm.n_threads(), 0.0);
      parallel_for_tid(0, iv, [&](int tid, size_t v) {
        if (frag.out_degree(static_cast<vid_t>(v)) == 0)
          per_thread[tid] += ctx.rank[v];
      }, 4096);
      for (double x : per_thread) local_dangling += x;
    }
    double dangling = mm.sum_double(local_dangling);

    // push contributions
    parallel_for(0, iv, [&](size_t vs) {
      vid_t v = static_cast<vid_t>(vs);
      auto adj = frag.out_edges(v);
      if (adj.n == 0) return;
      double c = ctx.rank[v] / static_cast<double>(adj.n);
      for (size_t i = 0; i < adj.n; ++i) {
        auto& slot = ctx.acc[adj.dst[i]];
        double cur = slot.load(std::memory_order_relaxed);
        while (!slot.compare_exchange_weak(cur, cur + c,
                                           std::memory_order_relaxed)) {
        }
      }
    }, 512);

    // batch-shuffle flush of outer partial sums to owners (dense slices)
    if (frag.fnum() > 1 && mm.comm()) {
      int fnum = frag.fnum();
      std::vector<std::string> send(fnum);
      for (int f = 0; f < fnum; ++f) {
        auto [b, e] = frag.outer_range(f);
        std::vector<double> part(e - b);
        for (vid_t u = b; u < e; ++u) {
          part[u - b] = ctx.acc[u].load(std::memory_order_relaxed);
          ctx.acc[u].store(0.0, std::memory_order_relaxed);
        }
        send[f].assign(reinterpret_cast<const char*>(part.data()),
                       part.size() * sizeof(double));
      }
      auto recv = mm.comm()->exchange_all(send);
      for (int f = 0; f < fnum; ++f) {
        const auto& mir = frag.mirrors(f);
        const double* part = reinterpret_cast<const double*>(recv[f].data());
        parallel_for(0, mir.size(), [&](size_t i) {
          auto& slot = ctx.acc[mir[i]];
          double cur = slot.load(std::memory_order_relaxed);
          while (!slot.compare_exchange_weak(cur, cur + part[i],
                                             std::memory_order_relaxed)) {
          }
        }, 4096);
      }
    }

    // apply (+ optional L1 delta for convergence mode)
    const double base = (1.0 - d) / N + d * dangling / N;
    std::vector<double> l1_t(mm.n_threads(), 0.0);
    parallel_for_tid(0, iv, [&](int tid, size_t v) {
      double nv = base + d * ctx.acc[v].load(std::memory_order_relaxed);
      if (ctx.tol > 0) l1_t[tid] += std::fabs(nv - ctx.rank[v]);
      ctx.rank[v] = nv;
      ctx.acc[v].store(0.0, std::memory_order_relaxed);
    }, 4096);
    bool converged = false;
    if (ctx.tol > 0) {
      double l1 = 0;
      for (double x : l1_t) l1 += x;
      converged = mm.sum_double(l1) < ctx.tol;
    }
    if (!converged && ++ctx.iter < ctx.max_iters) mm.force_continue();
  }
};

}  // namespace grapehip
