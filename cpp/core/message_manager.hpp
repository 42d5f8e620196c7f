// grapehip — CPU message manager + BSP superstep worker.
//
// Reference parity: grape/parallel/{message_manager_base,
// default_message_manager,parallel_message_manager}.h and
// grape/worker/worker.h. Semantics kept exactly:
//   * per-round per-destination archives of (gid, msg) pairs;
//   * SyncStateOnOuterVertex / SendMsgThroughOEdges helpers;
//   * termination when a round moves zero bytes globally and no rank called
//     ForceContinue (worker.h:105-146, default_message_manager.h:324-350);
//   * per-thread channel buffers to avoid append contention
//     (thread_local_message_buffer.h).
// Transport is the TCP control plane (exchange_all) instead of MPI isend
// rings — the CPU path is the correctness oracle; the RCCL path lives in
// cpp/hip/.
#pragma once

#include <atomic>
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#include "archive.hpp"
#include "fragment.hpp"
#include "net.hpp"
#include "parallel.hpp"
#include "types.hpp"

namespace grapehip {

class MessageManager {
 public:
  void init(TcpComm* comm, const Fragment* frag, int n_threads) {
    comm_ = comm;
    frag_ = frag;
    fnum_ = frag->fnum();
    n_threads_ = n_threads;
    channels_.assign(n_threads, std::vector<InArchive>(fnum_));
    recv_.assign(fnum_, OutArchive());
    force_continue_ = false;
    terminated_ = false;
    round_ = 0;
    eager_bytes_.store(0, std::memory_order_relaxed);
    // compute/comm overlap (reference parallel_message_manager.h send
    // thread): channel blocks stream to peers over the data mesh WHILE
    // PEval/IncEval runs; the round boundary only drains stragglers
    if (comm_ && fnum_ > 1) comm_->enable_data_mesh();
  }

  void start_round() { force_continue_ = false; }

  // ---- send API (thread-safe via tid channel) --------------------------
  // blocks above this size flush mid-round to the background sender
  // (reference thread_local_message_buffer.h block flush)
  static constexpr size_t kFlushBlock = 32 << 10;

  template <typename MSG>
  void sync_state_on_outer_vertex(int tid, vid_t outer_lid, const MSG& msg) {
    vid_t gid = frag_->lid2gid(outer_lid);
    fid_t dst = frag_->parser().fid(gid);
    auto& ar = channels_[tid][dst];
    ar.add(gid);
    ar.add(msg);
    maybe_flush(ar, dst);
  }
  template <typename MSG>
  void send_to_fragment(int tid, fid_t dst, vid_t gid, const MSG& msg) {
    auto& ar = channels_[tid][dst];
    ar.add(gid);
    ar.add(msg);
    maybe_flush(ar, dst);
  }

  void force_continue() { force_continue_ = true; }

  // Cooperative abort (reference MessageManagerBase::ForceTerminate +
  // TerminateInfo, default_message_manager.h:156-166): the error string is
  // gathered on every rank at the next round boundary and thrown there,
  // so a failure on one fragment ends the whole query cleanly.
  void force_terminate(const std::string& info) {
    force_terminate_ = true;
    terminate_info_ = info;
  }

  // ---- round boundary --------------------------------------------------
  void finish_round() {
    // concatenate the channel tails per destination (most volume already
    // streamed mid-round by the data-mesh sender)
    std::vector<std::string> send(fnum_);
    for (int f = 0; f < fnum_; ++f) {
      size_t total = 0;
      for (int t = 0; t < n_threads_; ++t) total += channels_[t][f].size();
      send[f].reserve(total);
      for (int t = 0; t < n_threads_; ++t) {
        send[f].append(channels_[t][f].data(), channels_[t][f].size());
        channels_[t][f].clear();
      }
    }
    uint64_t moved = eager_bytes_.exchange(0, std::memory_order_relaxed);
    for (auto& s : send) moved += s.size();
    std::vector<std::string> recv =
        (comm_ && fnum_ > 1) ? comm_->data_round_end(std::move(send))
                             : std::move(send);
    for (int f = 0; f < fnum_; ++f) recv_[f].reset(std::move(recv[f]));

    uint64_t global_moved =
        comm_ ? comm_->allreduce_sum(moved + (force_continue_ ? 1 : 0))
              : moved + (force_continue_ ? 1 : 0);
    bool any_abort = comm_ ? comm_->allreduce_or(force_terminate_)
                           : force_terminate_;
    if (any_abort) {
      std::string info = terminate_info_;
      if (comm_) {
        std::vector<std::string> send(fnum_, terminate_info_);
        auto all = comm_->exchange_all(send);
        for (auto& blob : all)
          if (info.empty() && !blob.empty()) info = blob;
      }
      throw std::runtime_error("query force-terminated: " +
                               (info.empty() ? std::string("(no info)")
                                             : info));
    }
    terminated_ = (global_moved == 0);
    ++round_;
  }

  // bytes queued/streamed this round (message-size accounting parity:
  // reference GetMsgSize, default_message_manager.h:146)
  uint64_t bytes_sent() const {
    uint64_t n = eager_bytes_.load(std::memory_order_relaxed);
    for (auto& per_tid : channels_)
      for (auto& a : per_tid) n += a.size();
    return n;
  }

  bool to_terminate() const { return terminated_; }
  int round() const { return round_; }
  int n_threads() const { return n_threads_; }

  // App-level collectives (reference grape/communication/communicator.h) and
  // dense batch-shuffle exchanges (batch_shuffle_message_manager.h) go
  // straight over the control plane.
  TcpComm* comm() { return comm_; }
  double sum_double(double v) {
    if (!comm_ || fnum_ == 1) return v;
    std::vector<double> all(fnum_);
    comm_->allgather(&v, sizeof(double), all.data());
    double s = 0;
    for (double x : all) s += x;
    return s;
  }

  // ---- receive API -----------------------------------------------------
  // Decode all (gid, MSG) pairs from every source fragment; f(lid, msg).
  // Parallel over source archives and chunks within each archive.
  template <typename MSG, typename F>
  void process(F&& f) {
    const IdParser& P = frag_->parser();
    constexpr size_t unit = sizeof(vid_t) + sizeof(MSG);
    for (int src = 0; src < fnum_; ++src) {
      OutArchive& ar = recv_[src];
      size_t n = ar.remaining() / unit;
      const char* base = ar.cursor();
      parallel_for_tid(0, n, [&](int tid, size_t i) {
        vid_t gid;
        MSG m;
        std::memcpy(&gid, base + i * unit, sizeof(vid_t));
        std::memcpy(&m, base + i * unit + sizeof(vid_t), sizeof(MSG));
        f(tid, P.lid(gid), m);
      }, 2048);
      ar.skip(n * unit);
    }
  }

  // Raw variant: no gid->lid decode at all — the routing gid is passed
  // through untouched (payload-routed protocols like the GNN sampler).
  template <typename MSG, typename F>
  void process_raw(F&& f) {
    constexpr size_t unit = sizeof(vid_t) + sizeof(MSG);
    for (int src = 0; src < fnum_; ++src) {
      OutArchive& ar = recv_[src];
      size_t n = ar.remaining() / unit;
      const char* base = ar.cursor();
      parallel_for_tid(0, n, [&](int tid, size_t i) {
        vid_t gid;
        MSG m;
        std::memcpy(&gid, base + i * unit, sizeof(vid_t));
        std::memcpy(&m, base + i * unit + sizeof(vid_t), sizeof(MSG));
        f(tid, gid, m);
      }, 2048);
      ar.skip(n * unit);
    }
  }

  // Like process(), but the payload may target a NON-owned vertex of the
  // receiving fragment (owner pushing its inner state to mirrors, e.g.
  // core decomposition estimates / BC depth refresh). Decodes through the
  // fragment's full gid->lid map; drops gids this fragment doesn't hold.
  template <typename MSG, typename F>
  void process_any(F&& f) {
    constexpr size_t unit = sizeof(vid_t) + sizeof(MSG);
    for (int src = 0; src < fnum_; ++src) {
      OutArchive& ar = recv_[src];
      size_t n = ar.remaining() / unit;
      const char* base = ar.cursor();
      parallel_for_tid(0, n, [&](int tid, size_t i) {
        vid_t gid;
        MSG m;
        std::memcpy(&gid, base + i * unit, sizeof(vid_t));
        std::memcpy(&m, base + i * unit + sizeof(vid_t), sizeof(MSG));
        vid_t lid = frag_->gid2lid(gid);
        if (lid != kInvalidVid) f(tid, lid, m);
      }, 2048);
      ar.skip(n * unit);
    }
  }

 private:
  void maybe_flush(InArchive& ar, fid_t dst) {
    if (ar.size() < kFlushBlock || !comm_ || fnum_ <= 1 ||
        dst == frag_->fid())
      return;
    eager_bytes_.fetch_add(ar.size(), std::memory_order_relaxed);
    comm_->post_block(static_cast<int>(dst), ar.release());
    ar.clear();
  }

  TcpComm* comm_ = nullptr;
  const Fragment* frag_ = nullptr;
  int fnum_ = 1;
  int n_threads_ = 1;
  std::atomic<uint64_t> eager_bytes_{0};
  bool force_continue_ = false;
  bool force_terminate_ = false;
  std::string terminate_info_;
  bool terminated_ = false;
  int round_ = 0;
  std::vector<std::vector<InArchive>> channels_;  // [tid][fid]
  std::vector<OutArchive> recv_;                  // [src fid]
};

// BSP superstep driver (reference worker.h:105-146). GRAPEHIP_TRACE=1
// prints per-superstep wall time and bytes shipped (the reference's
// coordinator VLOG(1) round log, worker.h:120-139).
template <typename APP, typename CTX>
inline int RunWorker(APP& app, CTX& ctx, const Fragment& frag,
                     MessageManager& mm) {
  const bool trace = getenv("GRAPEHIP_TRACE") != nullptr;
  auto now = [] {
    return std::chrono::duration<double>(
               std::chrono::steady_clock::now().time_since_epoch())
        .count();
  };
  double t0 = trace ? now() : 0.0;
  mm.start_round();
  app.PEval(frag, ctx, mm);
  uint64_t sent = trace ? mm.bytes_sent() : 0;
  mm.finish_round();
  if (trace)
    fprintf(stderr, "[trace f%u] PEval  %8.2f ms  sent %zu B\n",
            frag.fid(), (now() - t0) * 1e3, static_cast<size_t>(sent));
  int rounds = 1;
  while (!mm.to_terminate()) {
    double tr = trace ? now() : 0.0;
    mm.start_round();
    app.IncEval(frag, ctx, mm);
    sent = trace ? mm.bytes_sent() : 0;
    mm.finish_round();
    if (trace)
      fprintf(stderr, "[trace f%u] round %-3d %6.2f ms  sent %zu B\n",
              frag.fid(), rounds, (now() - tr) * 1e3,
              static_cast<size_t>(sent));
    ++rounds;
  }
  if (trace)
    fprintf(stderr, "[trace f%u] done: %d supersteps, %.2f ms total\n",
            frag.fid(), rounds, (now() - t0) * 1e3);
  return rounds;
}

}  // namespace grapehip
