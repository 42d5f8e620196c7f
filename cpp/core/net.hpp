// grapehip — TCP control plane.
//
// The reference coordinates ranks with MPI (grape/communication/sync_comm.h).
// On an MI355X node the natural shape is one process per GPU bootstrapped the
// PyTorch-distributed way (MASTER_ADDR/MASTER_PORT env), with RCCL over xGMI
// as the data plane. This class is the control plane: full-mesh TCP sockets
// used for rendezvous, the per-round message-length matrix, termination
// allreduce, RCCL uniqueId broadcast, and the (test-only) CPU data path.
#pragma once

#include <condition_variable>
#include <cstdint>
#include <deque>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

namespace grapehip {

class TcpComm {
 public:
  TcpComm() = default;
  ~TcpComm();
  TcpComm(const TcpComm&) = delete;
  TcpComm& operator=(const TcpComm&) = delete;

  // Collective: every rank must call. master_port is used for rendezvous
  // only; per-pair sockets use ephemeral ports.
  void init(int rank, int world, const std::string& master_addr,
            int master_port);
  void finalize();

  int rank() const { return rank_; }
  int world() const { return world_; }

  // Point-to-point, explicitly sized (blocking).
  void send_bytes(int peer, const void* buf, size_t n);
  void recv_bytes(int peer, void* buf, size_t n);

  // Size-prefixed variants.
  void send_blob(int peer, const std::string& blob);
  std::string recv_blob(int peer);

  // Pairwise exchange of blobs with all peers; recv[i] = blob from rank i.
  // recv[rank] = send[rank] (self copy). Deadlock-free (sender thread).
  std::vector<std::string> exchange_all(const std::vector<std::string>& send);

  // n bytes contributed per rank; out has world*n bytes, rank-major.
  void allgather(const void* in, size_t n, void* out);

  uint64_t allreduce_sum(uint64_t v);
  uint64_t allreduce_max(uint64_t v);
  double allreduce_max_double(double v);
  bool allreduce_or(bool v);

  void bcast(void* buf, size_t n, int root);
  void barrier();

  // ---- async data plane (reference parallel_message_manager.h:396-434
  // send-thread parity) -------------------------------------------------
  // A SECOND full mesh carries framed message blocks streamed by a
  // background sender thread WHILE compute runs; control collectives stay
  // on the primary mesh, so mid-round allreduces never interleave with
  // payload frames. Collective to enable; rendezvous rides the existing
  // mesh (no extra master port).
  void enable_data_mesh();
  bool data_mesh_enabled() const { return !data_fds_.empty(); }
  // enqueue a block for peer (thread-safe; background thread streams it)
  void post_block(int peer, std::string blob);
  // flush tails + per-peer round markers, then drain every peer's frames
  // until its marker; returns the concatenated payload per source rank
  // (recv[rank] = self blocks + tail[rank]).
  std::vector<std::string> data_round_end(std::vector<std::string> tail);

 private:
  void sender_loop();

  int rank_ = 0;
  int world_ = 1;
  std::vector<int> fds_;  // fds_[peer]; -1 for self
  int listen_fd_ = -1;
  std::string master_addr_;
  // data mesh state
  std::vector<int> data_fds_;
  std::thread sender_;
  std::mutex q_mu_;
  std::condition_variable q_cv_;
  std::deque<std::pair<int, std::string>> q_;  // peer, blob ("" + peer<0: quit)
  bool sender_quit_ = false;
  std::string self_accum_;
  std::mutex self_mu_;
};

}  // namespace grapehip
