// grapehip — atomic bitset + dense vertex set (CPU frontier).
// Reference parity: grape/utils/bitset.h, grape/utils/vertex_set.h.
#pragma once

#include <atomic>
#include <cstdint>
#include <cstring>
#include <vector>

#include "parallel.hpp"
#include "types.hpp"

namespace grapehip {

class Bitset {
 public:
  Bitset() = default;
  explicit Bitset(size_t n) { init(n); }

  void init(size_t n) {
    n_ = n;
    words_ = std::vector<std::atomic<uint64_t>>((n + 63) / 64);
    clear();
  }
  void clear() {
    for (auto& w : words_) w.store(0, std::memory_order_relaxed);
  }
  void parallel_clear() {
    parallel_for(0, words_.size(), [&](size_t i) {
      words_[i].store(0, std::memory_order_relaxed);
    }, 4096);
  }

  bool set_bit_atomic(size_t i) {  // returns true if newly set
    uint64_t mask = 1ULL << (i & 63);
    uint64_t old = words_[i >> 6].fetch_or(mask, std::memory_order_relaxed);
    return (old & mask) == 0;
  }
  void set_bit(size_t i) {
    words_[i >> 6].store(
        words_[i >> 6].load(std::memory_order_relaxed) | (1ULL << (i & 63)),
        std::memory_order_relaxed);
  }
  bool get_bit(size_t i) const {
    return (words_[i >> 6].load(std::memory_order_relaxed) >> (i & 63)) & 1;
  }
  size_t count() const {
    size_t c = 0;
    for (auto& w : words_) c += __builtin_popcountll(w.load(std::memory_order_relaxed));
    return c;
  }
  size_t size() const { return n_; }
  size_t num_words() const { return words_.size(); }
  uint64_t word(size_t w) const { return words_[w].load(std::memory_order_relaxed); }

  void swap(Bitset& other) {
    words_.swap(other.words_);
    std::swap(n_, other.n_);
  }

  // Iterate set bits in [0, n) calling f(index). Parallel over words.
  template <typename F>
  void parallel_iterate(F&& f) const {
    parallel_for(0, words_.size(), [&](size_t w) {
      uint64_t bits = words_[w].load(std::memory_order_relaxed);
      while (bits) {
        int b = __builtin_ctzll(bits);
        bits &= bits - 1;
        f(static_cast<vid_t>((w << 6) + b));
      }
    }, 256);
  }

  // Same but f(tid, index) — for thread-local accumulators/channels.
  template <typename F>
  void parallel_iterate_tid(F&& f) const {
    parallel_for_tid(0, words_.size(), [&](int tid, size_t w) {
      uint64_t bits = words_[w].load(std::memory_order_relaxed);
      while (bits) {
        int b = __builtin_ctzll(bits);
        bits &= bits - 1;
        f(tid, static_cast<vid_t>((w << 6) + b));
      }
    }, 256);
  }

 private:
  size_t n_ = 0;
  std::vector<std::atomic<uint64_t>> words_;
};

// Frontier over inner-vertex lid range [0, n).
class DenseVertexSet {
 public:
  void init(size_t n) { bs_.init(n); }
  bool insert(vid_t v) { return bs_.set_bit_atomic(v); }
  bool exist(vid_t v) const { return bs_.get_bit(v); }
  size_t count() const { return bs_.count(); }
  bool empty() const { return bs_.count() == 0; }
  void clear() { bs_.parallel_clear(); }
  void swap(DenseVertexSet& o) { bs_.swap(o.bs_); }
  template <typename F>
  void parallel_iterate(F&& f) const { bs_.parallel_iterate(f); }
  template <typename F>
  void parallel_iterate_tid(F&& f) const { bs_.parallel_iterate_tid(f); }
  const Bitset& bitset() const { return bs_; }

 private:
  Bitset bs_;
};

}  // namespace grapehip
