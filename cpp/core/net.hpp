// grapehip — TCP control plane.
//
// The reference coordinates ranks with MPI (grape/communication/sync_comm.h).
// On an MI355X node the natural shape is one process per GPU bootstrapped the
// PyTorch-distributed way (MASTER_ADDR/MASTER_PORT env), with RCCL over xGMI
// as the data plane. This class is the control plane: full-mesh TCP sockets
// used for rendezvous, the per-round message-length matrix, termination
// allreduce, RCCL uniqueId broadcast, and the (test-only) CPU data path.
#pragma once

#include <cstdint>
#include <string>
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

 private:
  int rank_ = 0;
  int world_ = 1;
  std::vector<int> fds_;  // fds_[peer]; -1 for self
  int listen_fd_ = -1;
};

}  // namespace grapehip
