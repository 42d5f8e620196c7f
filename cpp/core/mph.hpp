// grapehip — minimal perfect hash for vertex indexing.
// Reference parity: grape/vertex_map/idxers/pthash_idxer.h (+
// grape/utils/pthash_utils/): a memory-lean oid->lid index replacing the
// open-addressing hashmap (~40+ B/key) with ~4-5 bits/key of bitmaps.
// Algorithm: BBHash-style multi-level displacement (Limasset et al.) —
// per level, keys hashing to a slot nobody else hits get that slot;
// colliders retry at the next level; stragglers land in a tiny sorted
// fallback. Construction is deterministic, so every rank building from
// the same replicated key list derives identical lids.
#pragma once

#include <algorithm>
#include <cstdint>
#include <vector>

#include "types.hpp"

namespace grapehip {

class MinimalPerfectHash {
 public:
  void build(const std::vector<oid_t>& keys) {
    n_ = keys.size();
    levels_.clear();
    fallback_.clear();
    std::vector<oid_t> cur(keys);
    uint64_t base = 0;
    for (int lvl = 0; lvl < kMaxLevels && !cur.empty(); ++lvl) {
      Level L;
      uint64_t m_bits = cur.size() * kGamma;
      L.nwords = (m_bits + 63) / 64 + 1;
      L.bits.assign(L.nwords, 0);
      std::vector<uint64_t> collide(L.nwords, 0);
      uint64_t m = L.nwords * 64;
      for (oid_t k : cur) {
        uint64_t p = slot(k, lvl, m);
        uint64_t w = p >> 6, b = 1ull << (p & 63);
        if (L.bits[w] & b)
          collide[w] |= b;
        else
          L.bits[w] |= b;
      }
      for (size_t w = 0; w < L.nwords; ++w) L.bits[w] &= ~collide[w];
      // rank acceleration: prefix popcount per word
      L.rank.resize(L.nwords);
      uint64_t run = 0;
      for (size_t w = 0; w < L.nwords; ++w) {
        L.rank[w] = run;
        run += __builtin_popcountll(L.bits[w]);
      }
      L.base = base;
      base += run;
      std::vector<oid_t> next;
      for (oid_t k : cur) {
        uint64_t p = slot(k, lvl, m);
        if (!(L.bits[p >> 6] & (1ull << (p & 63)))) next.push_back(k);
      }
      levels_.push_back(std::move(L));
      cur.swap(next);
    }
    // stragglers: sorted (key, index) pairs, binary searched
    std::sort(cur.begin(), cur.end());
    fallback_.reserve(cur.size());
    for (oid_t k : cur) fallback_.push_back({k, base++});
  }

  // index in [0, n) for a built key; arbitrary value for others (callers
  // verify through their lid->oid array, like the reference's idxer)
  uint64_t lookup(oid_t key) const {
    for (size_t lvl = 0; lvl < levels_.size(); ++lvl) {
      const Level& L = levels_[lvl];
      uint64_t m = L.nwords * 64;
      uint64_t p = slot(key, static_cast<int>(lvl), m);
      uint64_t w = p >> 6;
      uint64_t mask = 1ull << (p & 63);
      if (L.bits[w] & mask) {
        uint64_t below = __builtin_popcountll(L.bits[w] & (mask - 1));
        return L.base + L.rank[w] + below;
      }
    }
    auto it = std::lower_bound(
        fallback_.begin(), fallback_.end(), key,
        [](const std::pair<oid_t, uint64_t>& a, oid_t k) {
          return a.first < k;
        });
    if (it != fallback_.end() && it->first == key) return it->second;
    return n_;  // definitely-absent sentinel
  }

  uint64_t size() const { return n_; }
  size_t memory_bytes() const {
    size_t b = fallback_.size() * sizeof(fallback_[0]);
    for (const Level& L : levels_)
      b += (L.bits.size() + L.rank.size()) * 8;
    return b;
  }

 private:
  static constexpr int kMaxLevels = 8;
  static constexpr uint64_t kGamma = 2;  // bits per remaining key per level

  struct Level {
    std::vector<uint64_t> bits;
    std::vector<uint64_t> rank;
    uint64_t nwords = 0;
    uint64_t base = 0;
  };

  static uint64_t slot(oid_t key, int level, uint64_t m) {
    uint64_t z = static_cast<uint64_t>(key) +
                 0x9e3779b97f4a7c15ULL * (level + 1);
    z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ULL;
    z = (z ^ (z >> 27)) * 0x94d049bb133111ebULL;
    z ^= z >> 31;
    return z % m;
  }

  uint64_t n_ = 0;
  std::vector<Level> levels_;
  std::vector<std::pair<oid_t, uint64_t>> fallback_;
};

}  // namespace grapehip
