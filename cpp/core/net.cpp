#include "net.hpp"

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <chrono>
#include <cstring>
#include <stdexcept>
#include <thread>

namespace grapehip {

namespace {

void throw_errno(const char* what) {
  throw std::runtime_error(std::string("TcpComm: ") + what + ": " +
                           std::strerror(errno));
}

void set_common_opts(int fd) {
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
}

int listen_on(uint16_t port, uint16_t* actual_port) {
  int fd = socket(AF_INET, SOCK_STREAM, 0);
  if (fd < 0) throw_errno("socket");
  int one = 1;
  setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_addr.s_addr = htonl(INADDR_ANY);
  addr.sin_port = htons(port);
  if (bind(fd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) < 0)
    throw_errno("bind");
  if (listen(fd, 64) < 0) throw_errno("listen");
  socklen_t len = sizeof(addr);
  getsockname(fd, reinterpret_cast<sockaddr*>(&addr), &len);
  *actual_port = ntohs(addr.sin_port);
  return fd;
}

int connect_to(const std::string& host, uint16_t port, int timeout_sec = 120) {
  addrinfo hints{}, *res = nullptr;
  hints.ai_family = AF_INET;
  hints.ai_socktype = SOCK_STREAM;
  std::string port_s = std::to_string(port);
  if (getaddrinfo(host.c_str(), port_s.c_str(), &hints, &res) != 0 || !res)
    throw std::runtime_error("TcpComm: getaddrinfo failed for " + host);
  auto deadline =
      std::chrono::steady_clock::now() + std::chrono::seconds(timeout_sec);
  int fd = -1;
  for (;;) {
    fd = socket(AF_INET, SOCK_STREAM, 0);
    if (fd < 0) throw_errno("socket");
    if (connect(fd, res->ai_addr, res->ai_addrlen) == 0) break;
    close(fd);
    fd = -1;
    if (std::chrono::steady_clock::now() > deadline) {
      freeaddrinfo(res);
      throw std::runtime_error("TcpComm: connect timeout to " + host + ":" +
                               port_s);
    }
    std::this_thread::sleep_for(std::chrono::milliseconds(50));
  }
  freeaddrinfo(res);
  set_common_opts(fd);
  return fd;
}

void write_all(int fd, const void* buf, size_t n) {
  const char* p = static_cast<const char*>(buf);
  while (n > 0) {
    ssize_t w = ::send(fd, p, n, MSG_NOSIGNAL);
    if (w < 0) {
      if (errno == EINTR) continue;
      throw_errno("send");
    }
    p += w;
    n -= static_cast<size_t>(w);
  }
}

void read_all(int fd, void* buf, size_t n) {
  char* p = static_cast<char*>(buf);
  while (n > 0) {
    ssize_t r = ::recv(fd, p, n, 0);
    if (r < 0) {
      if (errno == EINTR) continue;
      throw_errno("recv");
    }
    if (r == 0) throw std::runtime_error("TcpComm: peer closed connection");
    p += r;
    n -= static_cast<size_t>(r);
  }
}

}  // namespace

void TcpComm::init(int rank, int world, const std::string& master_addr,
                   int master_port) {
  rank_ = rank;
  world_ = world;
  master_addr_ = master_addr;
  fds_.assign(world, -1);
  if (world == 1) return;

  // Phase 1: every rank opens an ephemeral listener; rank 0 additionally
  // listens on master_port, collects everyone's listener port, and
  // broadcasts the port table.
  uint16_t my_port = 0;
  listen_fd_ = listen_on(0, &my_port);
  std::vector<uint16_t> ports(world, 0);
  ports[rank] = my_port;

  if (rank == 0) {
    uint16_t rport = 0;
    int rv_fd = listen_on(static_cast<uint16_t>(master_port), &rport);
    std::vector<int> tmp(world, -1);
    for (int i = 1; i < world; ++i) {
      int fd = accept(rv_fd, nullptr, nullptr);
      if (fd < 0) throw_errno("accept(rendezvous)");
      set_common_opts(fd);
      int32_t peer_rank;
      uint16_t peer_port;
      read_all(fd, &peer_rank, sizeof(peer_rank));
      read_all(fd, &peer_port, sizeof(peer_port));
      tmp[peer_rank] = fd;
      ports[peer_rank] = peer_port;
    }
    for (int i = 1; i < world; ++i) {
      write_all(tmp[i], ports.data(), sizeof(uint16_t) * world);
      close(tmp[i]);
    }
    close(rv_fd);
  } else {
    int fd = connect_to(master_addr, static_cast<uint16_t>(master_port));
    int32_t r32 = rank;
    write_all(fd, &r32, sizeof(r32));
    write_all(fd, &my_port, sizeof(my_port));
    read_all(fd, ports.data(), sizeof(uint16_t) * world);
    close(fd);
  }

  // Phase 2: full mesh. For pair (i, j) with i < j: j connects to i.
  // Accept loop runs concurrently in a thread to avoid ordering deadlocks.
  int expect_accepts = world - 1 - rank;
  std::thread acceptor([&] {
    for (int k = 0; k < expect_accepts; ++k) {
      int fd = accept(listen_fd_, nullptr, nullptr);
      if (fd < 0) throw_errno("accept(mesh)");
      set_common_opts(fd);
      int32_t peer_rank;
      read_all(fd, &peer_rank, sizeof(peer_rank));
      fds_[peer_rank] = fd;
    }
  });
  for (int i = 0; i < rank; ++i) {
    int fd = connect_to(master_addr, ports[i]);
    int32_t r32 = rank;
    write_all(fd, &r32, sizeof(r32));
    fds_[i] = fd;
  }
  acceptor.join();
  close(listen_fd_);
  listen_fd_ = -1;
}

void TcpComm::finalize() {
  if (sender_.joinable()) {
    {
      std::lock_guard<std::mutex> lk(q_mu_);
      sender_quit_ = true;
    }
    q_cv_.notify_all();
    sender_.join();
  }
  for (int& fd : data_fds_) {
    if (fd >= 0) close(fd);
    fd = -1;
  }
  data_fds_.clear();
  for (int& fd : fds_) {
    if (fd >= 0) close(fd);
    fd = -1;
  }
  if (listen_fd_ >= 0) {
    close(listen_fd_);
    listen_fd_ = -1;
  }
}

TcpComm::~TcpComm() { finalize(); }

void TcpComm::send_bytes(int peer, const void* buf, size_t n) {
  write_all(fds_[peer], buf, n);
}
void TcpComm::recv_bytes(int peer, void* buf, size_t n) {
  read_all(fds_[peer], buf, n);
}

void TcpComm::send_blob(int peer, const std::string& blob) {
  uint64_t n = blob.size();
  write_all(fds_[peer], &n, sizeof(n));
  if (n) write_all(fds_[peer], blob.data(), n);
}

std::string TcpComm::recv_blob(int peer) {
  uint64_t n = 0;
  read_all(fds_[peer], &n, sizeof(n));
  std::string blob(n, '\0');
  if (n) read_all(fds_[peer], blob.data(), n);
  return blob;
}

std::vector<std::string> TcpComm::exchange_all(
    const std::vector<std::string>& send) {
  std::vector<std::string> recv(world_);
  recv[rank_] = send[rank_];
  // Ring schedule: step i pairs (rank -> rank+i) and (rank-i -> rank).
  // Each of the 7 xGMI-analog steps talks to a distinct peer pair; a sender
  // thread makes each step deadlock-free regardless of payload size.
  for (int step = 1; step < world_; ++step) {
    int to = (rank_ + step) % world_;
    int from = (rank_ - step + world_) % world_;
    std::thread sender([&] { send_blob(to, send[to]); });
    recv[from] = recv_blob(from);
    sender.join();
  }
  return recv;
}

void TcpComm::allgather(const void* in, size_t n, void* out) {
  char* o = static_cast<char*>(out);
  std::memcpy(o + static_cast<size_t>(rank_) * n, in, n);
  for (int step = 1; step < world_; ++step) {
    int to = (rank_ + step) % world_;
    int from = (rank_ - step + world_) % world_;
    std::thread sender([&] { write_all(fds_[to], in, n); });
    read_all(fds_[from], o + static_cast<size_t>(from) * n, n);
    sender.join();
  }
}

uint64_t TcpComm::allreduce_sum(uint64_t v) {
  if (world_ == 1) return v;
  std::vector<uint64_t> all(world_);
  allgather(&v, sizeof(v), all.data());
  uint64_t s = 0;
  for (auto x : all) s += x;
  return s;
}

uint64_t TcpComm::allreduce_max(uint64_t v) {
  if (world_ == 1) return v;
  std::vector<uint64_t> all(world_);
  allgather(&v, sizeof(v), all.data());
  uint64_t m = 0;
  for (auto x : all) m = x > m ? x : m;
  return m;
}

double TcpComm::allreduce_max_double(double v) {
  if (world_ == 1) return v;
  std::vector<double> all(world_);
  allgather(&v, sizeof(v), all.data());
  double m = all[0];
  for (auto x : all) m = x > m ? x : m;
  return m;
}

bool TcpComm::allreduce_or(bool v) {
  return allreduce_sum(v ? 1 : 0) != 0;
}

void TcpComm::bcast(void* buf, size_t n, int root) {
  if (world_ == 1) return;
  if (rank_ == root) {
    std::vector<std::thread> senders;
    for (int i = 0; i < world_; ++i) {
      if (i == root) continue;
      senders.emplace_back([&, i] { write_all(fds_[i], buf, n); });
    }
    for (auto& t : senders) t.join();
  } else {
    read_all(fds_[root], buf, n);
  }
}

void TcpComm::barrier() {
  if (world_ == 1) return;
  char b = 0;
  std::vector<char> sink(world_);
  allgather(&b, 1, sink.data());
}

// ---------------------------------------------------------------------------
// Async data plane (second mesh + background sender; see net.hpp)
// ---------------------------------------------------------------------------

namespace {
constexpr uint32_t kRoundMarker = 0xFFFFFFFFu;
}

void TcpComm::enable_data_mesh() {
  if (world_ == 1 || !data_fds_.empty()) return;
  data_fds_.assign(world_, -1);
  uint16_t my_port = 0;
  int lfd = listen_on(0, &my_port);
  std::vector<uint16_t> ports(world_);
  allgather(&my_port, sizeof(uint16_t), ports.data());
  int expect = world_ - 1 - rank_;
  std::thread acceptor([&] {
    for (int k = 0; k < expect; ++k) {
      int fd = accept(lfd, nullptr, nullptr);
      if (fd < 0) throw_errno("accept(data mesh)");
      set_common_opts(fd);
      int32_t peer_rank;
      read_all(fd, &peer_rank, sizeof(peer_rank));
      data_fds_[peer_rank] = fd;
    }
  });
  for (int i = 0; i < rank_; ++i) {
    int fd = connect_to(master_addr_, ports[i]);
    int32_t r32 = rank_;
    write_all(fd, &r32, sizeof(r32));
    data_fds_[i] = fd;
  }
  acceptor.join();
  close(lfd);
  sender_quit_ = false;
  sender_ = std::thread([this] { sender_loop(); });
}

void TcpComm::sender_loop() {
  for (;;) {
    std::pair<int, std::string> item;
    {
      std::unique_lock<std::mutex> lk(q_mu_);
      q_cv_.wait(lk, [&] { return sender_quit_ || !q_.empty(); });
      if (q_.empty()) return;  // quit with drained queue
      item = std::move(q_.front());
      q_.pop_front();
    }
    if (item.first >= world_) {
      // round marker for peer (first - world_)
      write_all(data_fds_[item.first - world_], &kRoundMarker, 4);
    } else {
      uint32_t len = static_cast<uint32_t>(item.second.size());
      write_all(data_fds_[item.first], &len, 4);
      if (len) write_all(data_fds_[item.first], item.second.data(), len);
    }
  }
}

void TcpComm::post_block(int peer, std::string blob) {
  if (blob.empty()) return;
  if (peer == rank_) {
    std::lock_guard<std::mutex> lk(self_mu_);
    self_accum_.append(blob);
    return;
  }
  {
    std::lock_guard<std::mutex> lk(q_mu_);
    q_.emplace_back(peer, std::move(blob));
  }
  q_cv_.notify_one();
}

std::vector<std::string> TcpComm::data_round_end(
    std::vector<std::string> tail) {
  // flush tails + markers (the sender streams them after any pending
  // mid-round blocks — FIFO per construction)
  {
    std::lock_guard<std::mutex> lk(q_mu_);
    for (int p = 0; p < world_; ++p) {
      if (p == rank_) continue;
      if (!tail[p].empty()) q_.emplace_back(p, std::move(tail[p]));
      q_.emplace_back(world_ + p, std::string());
    }
  }
  q_cv_.notify_one();
  std::vector<std::string> recv(world_);
  {
    std::lock_guard<std::mutex> lk(self_mu_);
    recv[rank_] = std::move(self_accum_);
    self_accum_.clear();
  }
  recv[rank_].append(tail[rank_]);
  for (int p = 0; p < world_; ++p) {
    if (p == rank_) continue;
    for (;;) {
      uint32_t len = 0;
      read_all(data_fds_[p], &len, 4);
      if (len == kRoundMarker) break;
      size_t off = recv[p].size();
      recv[p].resize(off + len);
      if (len) read_all(data_fds_[p], recv[p].data() + off, len);
    }
  }
  return recv;
}

}  // namespace grapehip
